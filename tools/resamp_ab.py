#!/usr/bin/env python3
"""A/B the LDS-tiled resampler kernel vs the naive gather kernel across
interp/decim shapes (FSDR_RESAMP_TILED=0 forces the naive path)."""
import ctypes
import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import futuresdr_amd as fa  # noqa: E402


def main():
    lib = fa.lib()
    fa.set_device(0)
    st = torch.cuda.current_stream()
    beta = fa.kaiser_beta(1e-4)
    S = 1 << 26
    d_in = ctypes.c_void_p()
    d_out = ctypes.c_void_p()
    assert lib.fsdr_dev_alloc(ctypes.byref(d_in), S * 8) == 0
    assert lib.fsdr_dev_alloc(ctypes.byref(d_out), S * 8) == 0
    fa.fill_uniform_dev(d_in.value, S, seed=3, stream=st.cuda_stream)
    for (L, M) in [(1, 4), (3, 2), (2, 3), (5, 8)]:
        nt = 128 * L
        taps = fa.lowpass_kaiser_n(nt, beta, 0.4 / max(L, M))
        for env in ("1", "0"):
            os.environ["FSDR_RESAMP_TILED"] = env
            rs = fa.Resampler(L, M, taps)
            for _ in range(3):
                rs.filter_dev(d_in.value, S, d_out.value, S,
                              stream=st.cuda_stream)
            torch.cuda.synchronize()
            e0 = torch.cuda.Event(enable_timing=True)
            e1 = torch.cuda.Event(enable_timing=True)
            e0.record(st)
            p = 0
            for _ in range(10):
                c, p, s = rs.filter_dev(d_in.value, S, d_out.value, S,
                                        stream=st.cuda_stream)
            e1.record(st)
            torch.cuda.synchronize()
            ms = e0.elapsed_time(e1) / 10
            tf = p * (nt // L) * 4 / (ms * 1e-3) / 1e12
            print(f"L={L} M={M} tiled={env}: {ms:8.3f} ms  prod={p}  "
                  f"{tf:6.1f} TF/s ({tf / 157.3:.3f} of fp32 peak)")
    lib.fsdr_dev_free(d_in)
    lib.fsdr_dev_free(d_out)


if __name__ == "__main__":
    main()
