#!/usr/bin/env python3
"""Extended randomized parity fuzz: many random chain configs (tap counts,
FFT lengths, stream sizes, both fused paths) vs the two-stage oracle.
Run on a GPU box: python tools/fuzz_chain.py [n_cases]
"""
import ctypes
import os
import sys

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
import futuresdr_amd as fa  # noqa: E402
import oracle  # noqa: E402


def main(n_cases=40):
    lib = fa.lib()
    r = np.random.default_rng(0xF00D)
    worst = 0.0
    for case in range(n_cases):
        nt1 = int(r.integers(2, 400))
        nt2 = int(r.integers(2, 400))
        fft_len = int(r.choice([64, 100, 256, 384, 512, 1000, 1024,
                                2048]))  # non-pow2 -> Bluestein split path
        n_in = int(r.integers(nt1 + nt2 + 4 * fft_len, 300_000))
        t1 = r.uniform(-1, 1, nt1).astype(np.float32)
        t2 = r.uniform(-1, 1, nt2).astype(np.float32)
        x = (r.uniform(-1, 1, (n_in, 2)) @ [1, 1j]).astype(np.complex64)
        d_in = ctypes.c_void_p()
        d_out = ctypes.c_void_p()
        assert lib.fsdr_dev_alloc(ctypes.byref(d_in), n_in * 8) == 0
        assert lib.fsdr_dev_alloc(ctypes.byref(d_out), n_in * 8) == 0
        try:
            lib.fsdr_memcpy_h2d(d_in, ctypes.c_void_p(x.ctypes.data),
                                n_in * 8)
            ch = fa.Chain(t1, t2, 4, fft_len)
            cons, prod = ch.run_dev(d_in.value, n_in, d_out.value, n_in)
            fa.synchronize()
            ref, cons_ref = oracle.chain_cf32(t1, t2, 4, fft_len, x)
            assert (cons, prod) == (cons_ref, ref.size), \
                (case, nt1, nt2, fft_len, n_in, cons, prod, cons_ref,
                 ref.size)
            if prod:
                got = np.zeros(prod, np.complex64)
                lib.fsdr_memcpy_d2h(ctypes.c_void_p(got.ctypes.data),
                                    d_out, prod * 8)
                rel = float(np.linalg.norm(got - ref) /
                            max(np.linalg.norm(ref), 1e-30))
                worst = max(worst, rel)
                assert rel < 3e-4, (case, nt1, nt2, fft_len, n_in, rel)
        finally:
            lib.fsdr_dev_free(d_in)
            lib.fsdr_dev_free(d_out)
    print(f"fuzz OK: {n_cases} cases, worst rel l2 = {worst:.2e}")


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 40)
