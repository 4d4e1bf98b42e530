"""GPU parity tests: HIP kernels vs the KAT-pinned CPU oracle.

All marked @pytest.mark.gpu — they run on a real MI355X (gpurun / driver).
Everything compares through the C-ABI (the drop-in boundary); nothing here
reads /root/reference at run time.

Tolerance bar: consumed/produced/status bit-exact; float
results |gpu - oracle|_inf <= 1e-5 * max(1, |oracle|_inf) for FIR-class
kernels (same-order fp32 sums, GPU uses fma), 1e-4 relative l2 for FFT.
"""
import ctypes
import os

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

GOLDEN = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden",
                      "golden.npz")


def rng(seed=0):
    return np.random.default_rng(seed)


def cplx(r, n):
    return (r.uniform(-1, 1, (n, 2)) @ [1, 1j]).astype(np.complex64)


def assert_close(got, ref, tol=1e-5):
    ref = np.asarray(ref)
    got = np.asarray(got)
    assert got.shape == ref.shape
    scale = max(1.0, float(np.abs(ref).max()) if ref.size else 1.0)
    err = float(np.abs(got - ref).max()) if ref.size else 0.0
    assert err <= tol * scale, f"max err {err} > {tol}*{scale}"


# ---------------- FIR cf32 (the dominant kernel) ----------------------

@pytest.mark.parametrize("n_in", [1, 126, 127, 128, 300, 1149, 1150, 1151,
                                  2048, 4096, 12345, 100000])
def test_fir_cf32_parity_sizes(gpu, oracle_lib, n_in):
    r = rng(n_in)
    taps = r.uniform(-1, 1, 127).astype(np.float32)
    x = cplx(r, n_in)
    f = gpu.Fir(taps)
    assert f.length == 127
    got, c, p, s = f.filter(x, n_in)
    ref, co, po, so = oracle_lib.fir_cf32(taps, x, n_in)
    assert (c, p, s) == (co, po, so)
    assert_close(got, ref)


@pytest.mark.parametrize("n_taps", [1, 2, 3, 5, 6, 7, 63, 64, 121, 127, 128,
                                    250])
def test_fir_cf32_parity_tap_counts(gpu, oracle_lib, n_taps):
    r = rng(n_taps + 1000)
    taps = r.uniform(-1, 1, n_taps).astype(np.float32)
    x = cplx(r, 3000)
    got, c, p, s = gpu.Fir(taps).filter(x, 4000)
    ref, co, po, so = oracle_lib.fir_cf32(taps, x, 4000)
    assert (c, p, s) == (co, po, so)
    assert_close(got, ref)


def test_fir_cf32_statuses(gpu, oracle_lib):
    r = rng(7)
    taps = r.uniform(-1, 1, 3).astype(np.float32)
    x = cplx(r, 5)
    f = gpu.Fir(taps)
    # InsufficientOutput (fir.rs:71)
    got, c, p, s = f.filter(x, 2)
    assert (c, p, s) == (2, 2, gpu.INSUFFICIENT_OUTPUT)
    # BothSufficient (fir.rs:72)
    got, c, p, s = f.filter(x, 3)
    assert (c, p, s) == (3, 3, gpu.BOTH_SUFFICIENT)
    # InsufficientInput (fir.rs:73)
    got, c, p, s = f.filter(x, 10)
    assert (c, p, s) == (3, 3, gpu.INSUFFICIENT_INPUT)
    # empty input / empty output
    got, c, p, s = f.filter(x[:0], 10)
    assert (c, p, s) == (0, 0, gpu.INSUFFICIENT_INPUT)
    got, c, p, s = f.filter(x, 0)
    assert (c, p, s) == (0, 0, gpu.INSUFFICIENT_OUTPUT)
    # input shorter than taps
    got, c, p, s = f.filter(x[:2], 10)
    assert (c, p, s) == (0, 0, gpu.INSUFFICIENT_INPUT)


def test_fir_cf32_impulse_exact(gpu):
    """Impulse train -> each output is exactly one tap (bit-exact)."""
    taps = rng(9).uniform(-1, 1, 127).astype(np.float32)
    n = 8192
    x = np.zeros(n, np.complex64)
    x[::512] = 1.0 + 0.0j
    got, c, p, s = gpu.Fir(taps).filter(x, n)
    # y[k] = sum_t x[k+t] h[126-t]; impulses at k+t = 512*m
    ref = np.zeros(p, np.complex64)
    for m in range(0, n, 512):
        for k in range(max(0, m - 126), min(p, m + 1)):
            ref[k] += taps[126 - (m - k)]
    np.testing.assert_array_equal(got, ref)


def test_fir_cf32_vs_golden(gpu):
    g = np.load(GOLDEN)
    got, c, p, s = gpu.Fir(g["fir127_taps"]).filter(g["fir127_in"], 10 ** 6)
    assert p == g["fir127_out"].size
    assert_close(got, g["fir127_out"].astype(np.complex64))


def test_fir_f32_parity(gpu, oracle_lib):
    r = rng(11)
    taps = r.uniform(-1, 1, 64).astype(np.float32)  # perf/fir shape
    x = r.uniform(-1, 1, 15000).astype(np.float32)
    got, c, p, s = gpu.FirF32(taps).filter(x, 15000)
    ref, co, po, so = oracle_lib.fir_f32(taps, x, 15000)
    assert (c, p, s) == (co, po, so)
    assert_close(got, ref)


# ---------------- Decimating FIR --------------------------------------

@pytest.mark.parametrize("decim,n_taps,n_in", [
    (4, 127, 50000),   # fast path, metric shape
    (4, 127, 4222),    # one tile exactly
    (4, 127, 4223), (4, 127, 9000), (4, 64, 10000), (4, 1, 1000),
    (2, 127, 10000),   # generic path
    (3, 33, 5000), (8, 127, 30000), (1, 127, 5000),
])
def test_decim_fir_parity(gpu, oracle_lib, decim, n_taps, n_in):
    r = rng(decim * 1000 + n_taps)
    taps = r.uniform(-1, 1, n_taps).astype(np.float32)
    x = cplx(r, n_in)
    got, c, p, s = gpu.DecimFir(decim, taps).filter(x, n_in)
    ref, co, po, so = oracle_lib.decim_fir_cf32(decim, taps, x, n_in)
    assert (c, p, s) == (co, po, so)
    assert_close(got, ref)


def test_decim_fir_statuses(gpu, oracle_lib):
    r = rng(13)
    taps = r.uniform(-1, 1, 3).astype(np.float32)
    f = gpu.DecimFir(2, taps)
    for n_in, n_out in [(4, 3), (5, 3), (5, 1), (6, 1), (6, 3), (6, 0),
                        (0, 4), (2, 4)]:
        x = cplx(r, n_in)
        got, c, p, s = f.filter(x, n_out)
        ref, co, po, so = oracle_lib.decim_fir_cf32(2, taps, x, n_out)
        assert (c, p, s) == (co, po, so), (n_in, n_out)
        assert_close(got, ref)


# ---------------- Polyphase resampler ---------------------------------

@pytest.mark.parametrize("interp,decim,n_taps", [
    (3, 2, 6), (2, 1, 2), (1, 3, 2), (1, 4, 128), (5, 3, 40), (4, 7, 48),
])
def test_resampler_parity(gpu, oracle_lib, interp, decim, n_taps):
    r = rng(interp * 100 + decim)
    taps = r.uniform(-1, 1, n_taps).astype(np.float32)
    x = cplx(r, 5000)
    got, c, p, s = gpu.Resampler(interp, decim, taps).filter(x, 20000)
    ref, co, po, so = oracle_lib.resamp_cf32(interp, decim, taps, x, 20000)
    assert (c, p, s) == (co, po, so)
    assert_close(got, ref)


def test_resampler_output_multiple_of_interp(gpu, oracle_lib):
    # InsufficientOutput rounds produced down to a multiple of interp
    # (polyphase_resampling_fir.rs:96-100)
    r = rng(17)
    taps = r.uniform(-1, 1, 6).astype(np.float32)
    x = cplx(r, 100)
    got, c, p, s = gpu.Resampler(3, 2, taps).filter(x, 7)
    ref, co, po, so = oracle_lib.resamp_cf32(3, 2, taps, x, 7)
    assert (c, p, s) == (co, po, so)
    assert p % 3 == 0
    assert_close(got, ref)


# ---------------- FFT ---------------------------------------------------

@pytest.mark.parametrize("n", [16, 64, 256, 1024, 2048])
def test_fft_parity_vs_oracle(gpu, oracle_lib, n):
    r = rng(n)
    frames = 8
    x = cplx(r, n * frames)
    got, c, p, s = gpu.Fft(n).filter(x, n * frames)
    assert (c, p) == (n * frames, n * frames)
    ref = np.concatenate([
        oracle_lib.dft_cf32(x[i * n:(i + 1) * n]) for i in range(frames)])
    rel = np.linalg.norm(got - ref) / np.linalg.norm(ref)
    assert rel < 1e-4, rel


def test_fft_vs_golden(gpu):
    g = np.load(GOLDEN)
    # pow2 -> Stockham kernel; non-pow2 -> Bluestein path
    for n in (64, 256, 1024, 60, 100, 1000):
        xs, refs = g[f"fft{n}_in"], g[f"fft{n}_out"]
        got, c, p, s = gpu.Fft(n).filter(xs.ravel(), xs.size)
        got = got.reshape(xs.shape)
        rel = np.linalg.norm(got - refs) / np.linalg.norm(refs)
        assert rel < 1e-4, (n, rel)


def test_fft_m_rounding_and_cap(gpu, oracle_lib):
    r = rng(31)
    x = cplx(r, 1024 * 40)
    f = gpu.Fft(1024)
    got, c, p, s = f.filter(x, 1024 * 40)
    assert c == p == 1024 * 32  # BUFF_FFTS cap (fft.rs:56,171)
    got, c, p, s = f.filter(x[:2500], 4096)
    assert c == p == 2048  # rounded down to multiple of len
    got, c, p, s = f.filter(x[:1000], 4096)
    assert c == p == 0


def test_fft_inverse_shift_normalize(gpu, oracle_lib):
    r = rng(37)
    n = 256
    x = cplx(r, n * 2)
    for kw in [dict(inverse=True), dict(fft_shift=True),
               dict(inverse=True, fft_shift=True),
               dict(normalize=1.0 / n), dict(fft_shift=True,
                                             normalize=0.5)]:
        got, c, p, s = gpu.Fft(n, **kw).filter(x, n * 2)
        ref, m = oracle_lib.fft_block(n, x, n * 2, **kw)
        assert p == m
        rel = np.linalg.norm(got - ref) / max(np.linalg.norm(ref), 1e-30)
        assert rel < 1e-4, (kw, rel)


def test_fft_roundtrip(gpu):
    r = rng(41)
    n = 1024
    x = cplx(r, n * 4)
    X, _, _, _ = gpu.Fft(n).filter(x, x.size)
    back, _, _, _ = gpu.Fft(n, inverse=True, normalize=1.0 / n).filter(
        X, X.size)
    assert_close(back, x, 1e-4)


def test_fft_parseval(gpu):
    r = rng(43)
    n = 1024
    x = cplx(r, n * 16)
    X, _, _, _ = gpu.Fft(n).filter(x, x.size)
    for f in range(16):
        e_t = np.sum(np.abs(x[f * n:(f + 1) * n]) ** 2)
        e_f = np.sum(np.abs(X[f * n:(f + 1) * n]) ** 2) / n
        assert abs(e_t - e_f) / e_t < 1e-5


@pytest.mark.parametrize("n", [6, 12, 60, 100, 384, 1000, 2000])
def test_fft_bluestein_parity_vs_oracle(gpu, oracle_lib, n):
    """Non-pow2 lengths (the reference Fft is generic over rustfft plan
    lengths, fft.rs:98-103): Bluestein path vs the f64 oracle DFT."""
    r = rng(4000 + n)
    frames = 5
    x = cplx(r, n * frames)
    got, c, p, s = gpu.Fft(n).filter(x, n * frames)
    assert (c, p) == (n * frames, n * frames)
    ref = np.concatenate([
        oracle_lib.dft_cf32(x[i * n:(i + 1) * n]) for i in range(frames)])
    rel = np.linalg.norm(got - ref) / np.linalg.norm(ref)
    assert rel < 2e-4, rel


def test_fft_bluestein_inverse_shift_normalize(gpu, oracle_lib):
    r = rng(4100)
    n = 100
    x = cplx(r, n * 3)
    for kw in [dict(inverse=True), dict(fft_shift=True),
               dict(inverse=True, fft_shift=True),
               dict(normalize=1.0 / n)]:
        got, c, p, s = gpu.Fft(n, **kw).filter(x, n * 3)
        ref, m = oracle_lib.fft_block(n, x, n * 3, **kw)
        assert p == m
        rel = np.linalg.norm(got - ref) / max(np.linalg.norm(ref), 1e-30)
        assert rel < 2e-4, (kw, rel)


def test_chain_kernel_variants_agree(gpu, oracle_lib):
    """The three fused-kernel variants — ws (software-pipelined FFT,
    the aligned-1024 default), ap (all-phase aligned-group staging) and
    halves (the unaligned ring fallback) — compute the same MFMA/K and
    butterfly order: outputs must agree to float equality on identical
    input."""
    r = rng(4400)
    t1 = r.uniform(-1, 1, 127).astype(np.float32)
    t2 = r.uniform(-1, 1, 127).astype(np.float32)
    n_in = 300000
    x = cplx(r, n_in)
    lib = gpu.lib()
    d_in = ctypes.c_void_p()
    d_out = ctypes.c_void_p()
    assert lib.fsdr_dev_alloc(ctypes.byref(d_in), n_in * 8) == 0
    assert lib.fsdr_dev_alloc(ctypes.byref(d_out), n_in * 8) == 0
    variants = {
        "ws": {},
        "ap": {"FSDR_CHAIN_WS": "0"},
        "halves": {"FSDR_CHAIN_WS": "0", "FSDR_CHAIN_ALLPHASE": "0"},
    }
    try:
        lib.fsdr_memcpy_h2d(d_in, ctypes.c_void_p(x.ctypes.data), n_in * 8)
        outs = {}
        for name, env in variants.items():
            for k in ("FSDR_CHAIN_WS", "FSDR_CHAIN_ALLPHASE"):
                os.environ.pop(k, None)
            os.environ.update(env)
            ch = gpu.Chain(t1, t2, 4, 1024)
            cons, prod = ch.run_dev(d_in.value, n_in, d_out.value, n_in)
            gpu.synchronize()
            h = np.zeros(prod, np.complex64)
            lib.fsdr_memcpy_d2h(ctypes.c_void_p(h.ctypes.data), d_out,
                                prod * 8)
            outs[name] = h
        for k in ("FSDR_CHAIN_WS", "FSDR_CHAIN_ALLPHASE"):
            os.environ.pop(k, None)
        assert outs["ws"].size == outs["ap"].size == outs["halves"].size
        assert_close(outs["ws"], outs["ap"], 1e-6)
        assert_close(outs["ap"], outs["halves"], 1e-6)
    finally:
        lib.fsdr_dev_free(d_in)
        lib.fsdr_dev_free(d_out)


def test_chain_non_pow2_fft_vs_oracle(gpu, oracle_lib):
    """Chain with a non-pow2 FFT length: the split path runs the
    Bluestein FFT; outputs equal the two-stage oracle (which falls back
    to an exact f64 DFT per frame for non-pow2)."""
    r = rng(4300)
    t1 = r.uniform(-1, 1, 63).astype(np.float32)
    t2 = r.uniform(-1, 1, 65).astype(np.float32)
    fft_len, n_in = 100, 30000
    x = cplx(r, n_in)
    lib = gpu.lib()
    d_in = ctypes.c_void_p()
    d_out = ctypes.c_void_p()
    assert lib.fsdr_dev_alloc(ctypes.byref(d_in), n_in * 8) == 0
    assert lib.fsdr_dev_alloc(ctypes.byref(d_out), n_in * 8) == 0
    try:
        lib.fsdr_memcpy_h2d(d_in, ctypes.c_void_p(x.ctypes.data), n_in * 8)
        ch = gpu.Chain(t1, t2, 4, fft_len)
        cons, prod = ch.run_dev(d_in.value, n_in, d_out.value, n_in)
        ref, cons_ref = oracle_lib.chain_cf32(t1, t2, 4, fft_len, x)
        assert (cons, prod) == (cons_ref, ref.size)
        gpu.synchronize()
        got = np.zeros(prod, np.complex64)
        lib.fsdr_memcpy_d2h(ctypes.c_void_p(got.ctypes.data), d_out,
                            prod * 8)
        rel = np.linalg.norm(got - ref) / np.linalg.norm(ref)
        assert rel < 2e-4, rel
    finally:
        lib.fsdr_dev_free(d_in)
        lib.fsdr_dev_free(d_out)


def test_fft_bluestein_roundtrip(gpu):
    r = rng(4200)
    n = 60
    x = cplx(r, n * 4)
    X, _, _, _ = gpu.Fft(n).filter(x, x.size)
    back, _, _, _ = gpu.Fft(n, inverse=True, normalize=1.0 / n).filter(
        X, X.size)
    assert_close(back, x, 1e-4)


# ---------------- element-wise + synthetic source ----------------------

def test_mag2_parity(gpu, oracle_lib):
    x = cplx(rng(47), 10000)
    got, c, p, s = gpu.Mag2().filter(x, 10000)
    assert_close(got, oracle_lib.mag2(x), 1e-6)


def test_cmul_parity(gpu, oracle_lib):
    r = rng(53)
    a, b = cplx(r, 5000), cplx(r, 4000)
    got = gpu.cmul_host(a, b)
    ref = oracle_lib.cmul(a, b)
    assert got.size == ref.size == 4000
    assert_close(got, ref, 1e-6)


def test_fill_uniform_deterministic(gpu):
    lib = gpu.lib()
    n = 100000
    ptr = ctypes.c_void_p()
    assert lib.fsdr_dev_alloc(ctypes.byref(ptr), n * 8) == 0
    try:
        gpu.fill_uniform_dev(ptr.value, n, seed=123)
        h1 = np.zeros(n, np.complex64)
        lib.fsdr_memcpy_d2h(ctypes.c_void_p(h1.ctypes.data), ptr, n * 8)
        gpu.fill_uniform_dev(ptr.value, n, seed=123)
        h2 = np.zeros(n, np.complex64)
        lib.fsdr_memcpy_d2h(ctypes.c_void_p(h2.ctypes.data), ptr, n * 8)
        np.testing.assert_array_equal(h1, h2)
        v = h1.view(np.float32)
        assert v.min() >= -1.0 and v.max() < 1.0
        assert abs(v.mean()) < 0.01 and abs(v.std() - 0.577) < 0.01
        # different seed differs
        gpu.fill_uniform_dev(ptr.value, n, seed=124)
        lib.fsdr_memcpy_d2h(ctypes.c_void_p(h2.ctypes.data), ptr, n * 8)
        assert not np.array_equal(h1, h2)
    finally:
        lib.fsdr_dev_free(ptr)


# ---------------- chain (the bench hot path) ---------------------------

def test_chain_parity_vs_oracle(gpu, oracle_lib):
    r = rng(59)
    n_in = 4 * 1024 * 8 + 4 * 127 + 500  # ~8 frames + slack
    x = cplx(r, n_in)
    taps1 = r.uniform(-1, 1, 127).astype(np.float32)
    taps2 = r.uniform(-1, 1, 127).astype(np.float32)
    lib = gpu.lib()
    d_in = ctypes.c_void_p()
    d_out = ctypes.c_void_p()
    d_mag = ctypes.c_void_p()
    assert lib.fsdr_dev_alloc(ctypes.byref(d_in), n_in * 8) == 0
    assert lib.fsdr_dev_alloc(ctypes.byref(d_out), n_in * 8) == 0
    assert lib.fsdr_dev_alloc(ctypes.byref(d_mag), n_in * 4) == 0
    try:
        lib.fsdr_memcpy_h2d(d_in, ctypes.c_void_p(x.ctypes.data), n_in * 8)
        ch = gpu.Chain(taps1, taps2, 4, 1024)
        cons, prod = ch.run_dev(d_in.value, n_in, d_out.value, n_in,
                                d_mag.value, n_in)
        gpu.synchronize()
        ref, cons_ref = oracle_lib.chain_cf32(taps1, taps2, 4, 1024, x)
        assert prod == ref.size and cons == cons_ref
        got = np.zeros(prod, np.complex64)
        lib.fsdr_memcpy_d2h(ctypes.c_void_p(got.ctypes.data), d_out,
                            prod * 8)
        rel = np.linalg.norm(got - ref) / np.linalg.norm(ref)
        assert rel < 1e-4, rel
        mag = np.zeros(prod, np.float32)
        lib.fsdr_memcpy_d2h(ctypes.c_void_p(mag.ctypes.data), d_mag,
                            prod * 4)
        assert_close(mag, np.abs(got) ** 2, 1e-6)
    finally:
        lib.fsdr_dev_free(d_in)
        lib.fsdr_dev_free(d_out)
        lib.fsdr_dev_free(d_mag)


def test_chain_full_size_properties(gpu):
    """BASELINE-scale run (2^22 samples): determinism (checksum of two
    runs identical) + per-frame Parseval between stage-2 output computed
    through the filter ABI and the chain's spectra."""
    r = rng(61)
    lib = gpu.lib()
    n_in = 1 << 22
    taps1 = gpu.kaiser_lowpass(0.1, 0.02, 1e-4)
    taps2 = gpu.lowpass_kaiser_n(127, gpu.kaiser_beta(1e-4), 0.11)
    d_in = ctypes.c_void_p()
    d_out = ctypes.c_void_p()
    assert lib.fsdr_dev_alloc(ctypes.byref(d_in), n_in * 8) == 0
    assert lib.fsdr_dev_alloc(ctypes.byref(d_out), n_in * 8) == 0
    try:
        gpu.fill_uniform_dev(d_in.value, n_in, seed=0x5D5D5D5D)
        ch = gpu.Chain(taps1, taps2, 4, 1024)
        cons, prod = ch.run_dev(d_in.value, n_in, d_out.value, n_in)
        gpu.synchronize()
        assert prod > 0
        out1 = np.zeros(prod, np.complex64)
        lib.fsdr_memcpy_d2h(ctypes.c_void_p(out1.ctypes.data), d_out,
                            prod * 8)
        # determinism
        cons2, prod2 = ch.run_dev(d_in.value, n_in, d_out.value, n_in)
        gpu.synchronize()
        out2 = np.zeros(prod, np.complex64)
        lib.fsdr_memcpy_d2h(ctypes.c_void_p(out2.ctypes.data), d_out,
                            prod * 8)
        assert (cons, prod) == (cons2, prod2)
        np.testing.assert_array_equal(out1, out2)
        # Parseval vs the composed per-stage path at full size
        fir1 = gpu.Fir(taps1)
        d_y1 = ctypes.c_void_p()
        assert lib.fsdr_dev_alloc(ctypes.byref(d_y1), n_in * 8) == 0
        try:
            c1, p1, _ = fir1.filter_dev(d_in.value, n_in, d_y1.value, n_in)
            fir2 = gpu.DecimFir(4, taps2)
            d_y2 = ctypes.c_void_p()
            assert lib.fsdr_dev_alloc(ctypes.byref(d_y2), n_in * 2) == 0
            try:
                c2, p2, _ = fir2.filter_dev(d_y1.value, p1, d_y2.value,
                                            n_in // 4)
                gpu.synchronize()
                y2 = np.zeros(p2, np.complex64)
                lib.fsdr_memcpy_d2h(ctypes.c_void_p(y2.ctypes.data), d_y2,
                                    p2 * 8)
                frames = prod // 1024
                for f in range(0, frames, max(1, frames // 7)):
                    e_t = np.sum(np.abs(y2[f * 1024:(f + 1) * 1024]) ** 2)
                    e_f = np.sum(np.abs(out1[f * 1024:(f + 1) * 1024]) ** 2)
                    assert abs(e_f / 1024 - e_t) / max(e_t, 1e-30) < 1e-4
            finally:
                lib.fsdr_dev_free(d_y2)
        finally:
            lib.fsdr_dev_free(d_y1)
    finally:
        lib.fsdr_dev_free(d_in)
        lib.fsdr_dev_free(d_out)


# ---------------- ring (Slab-style) ------------------------------------

def _stream_through_ring(gpu, lib, chunks, reserved, consume):
    """Push `chunks` (list of complex64 arrays) through the ring; call
    `consume(dev_ptr, n_presented) -> (consumed, output_array)` per
    acquire; release_consumed carries the leftover slab-style. Returns
    the concatenated outputs."""
    max_chunk = max(c.size for c in chunks)
    ring = lib.fsdr_ring_create(4, max_chunk, 8, reserved)
    assert ring, lib.fsdr_last_error().decode()
    outs = []
    try:
        for x in chunks:
            hp = ctypes.c_void_p()
            items = ctypes.c_size_t()
            assert lib.fsdr_ring_writer_acquire(
                ring, ctypes.byref(hp), ctypes.byref(items)) == 0
            assert items.value >= x.size
            ctypes.memmove(hp, ctypes.c_void_p(x.ctypes.data), x.size * 8)
            assert lib.fsdr_ring_writer_commit(ring, x.size) == 0
            dp = ctypes.c_void_p()
            got_items = ctypes.c_size_t()
            assert lib.fsdr_ring_reader_acquire(
                ring, ctypes.byref(dp), ctypes.byref(got_items)) == 0
            cons, out = consume(dp.value, got_items.value)
            outs.append(out)
            assert lib.fsdr_ring_reader_release_consumed(
                ring, cons, None) == 0, lib.fsdr_last_error().decode()
    finally:
        lib.fsdr_ring_destroy(ring)
    return np.concatenate(outs)


def test_ring_history_prefix(gpu, oracle_lib):
    """Stream a long signal through the pinned ring in chunks and run the
    FIR on each acquired device buffer; the carried unconsumed tail
    (slab.rs:369-399 semantics) makes the concatenated outputs equal the
    single-shot oracle on the raw stream — no zero prologue, exactly like
    the reference block consuming less on its first work()."""
    lib = gpu.lib()
    r = rng(67)
    taps = r.uniform(-1, 1, 127).astype(np.float32)
    n_total, chunk = 40960, 4096
    x = cplx(r, n_total)
    fir = gpu.Fir(taps)
    d_out = ctypes.c_void_p()
    assert lib.fsdr_dev_alloc(ctypes.byref(d_out), 2 * chunk * 8) == 0
    try:
        def consume(dp, n_in):
            c, p, s = fir.filter_dev(dp, n_in, d_out.value, 2 * chunk)
            gpu.synchronize()
            h = np.zeros(p, np.complex64)
            lib.fsdr_memcpy_d2h(ctypes.c_void_p(h.ctypes.data), d_out,
                                p * 8)
            return c, h
        chunks = [x[o:o + chunk] for o in range(0, n_total, chunk)]
        got = _stream_through_ring(gpu, lib, chunks, taps.size - 1, consume)
        ref, c, p, s = oracle_lib.fir_cf32(taps, x, n_total)
        assert p == got.size
        assert_close(got, ref[:got.size])
    finally:
        lib.fsdr_dev_free(d_out)


# ---------------- dev path on a torch stream ---------------------------

def test_filter_dev_on_torch_stream(gpu):
    torch = pytest.importorskip("torch")
    if not torch.cuda.is_available():
        pytest.fail("torch.cuda not available on GPU box")
    r = rng(71)
    taps = r.uniform(-1, 1, 127).astype(np.float32)
    x = torch.from_numpy(cplx(r, 50000).view(np.float32)).cuda().view(-1, 2)
    y = torch.empty_like(x)
    f = gpu.Fir(taps)
    st = torch.cuda.current_stream().cuda_stream
    c, p, s = f.filter_dev(x.data_ptr(), x.shape[0], y.data_ptr(),
                           y.shape[0], stream=st)
    torch.cuda.synchronize()
    got = y[:p].cpu().numpy().view(np.complex64).ravel()
    ref, _, _, _ = __import__("oracle").fir_cf32(
        taps, x.cpu().numpy().view(np.complex64).ravel(), y.shape[0])
    assert_close(got, ref)


# ---------------- complex-taps FIR (WLAN correlator core) --------------

@pytest.mark.parametrize("n_taps,n_in", [(64, 320), (64, 10000), (3, 50),
                                         (127, 5000)])
def test_fir_ccf32_parity(gpu, oracle_lib, n_taps, n_in):
    r = rng(n_taps * 7 + n_in)
    taps = cplx(r, n_taps)
    x = cplx(r, n_in)
    got, c, p, s = gpu.FirCC(taps).filter(x, n_in)
    ref, co, po, so = oracle_lib.fir_ccf32(taps, x, n_in)
    assert (c, p, s) == (co, po, so)
    assert_close(got, ref)


def test_fir_ccf32_wlan_sync_long_shape(gpu, oracle_lib):
    """64-tap complex correlator over SEARCH_WINDOW=320 samples — the
    SyncLong shape (examples/wlan/src/sync_long.rs:18-50): correlate,
    mag^2, and find the top-2 peak indices; compare against the oracle
    pipeline end-to-end."""
    r = rng(802)
    # synthetic 'LTF-like' training sequence embedded at a known offset
    ltf = cplx(r, 64)
    sig = cplx(r, 320) * 0.05
    off = 123
    sig[off:off + 64] += ltf
    sig[off + 64:off + 128] += ltf  # repeated symbol -> two peaks
    taps = np.conj(ltf[::-1])  # matched filter
    got, _, p, _ = gpu.FirCC(taps).filter(sig, 320)
    ref, _, po, _ = oracle_lib.fir_ccf32(taps, sig, 320)
    assert p == po
    assert_close(got, ref)
    mags = np.abs(got) ** 2
    top2 = np.sort(np.argsort(mags)[-2:])
    assert abs(int(top2[1]) - int(top2[0])) == 64  # peak spacing = symbol


# ---------------- rotator ----------------------------------------------

def test_rotator_vs_ideal_and_oracle(gpu, oracle_lib):
    import cmath
    r = rng(90)
    n = 2048
    x = cplx(r, n)
    theta = 0.31
    got, final = gpu.rotator_host(x, theta)
    # vs ideal f64 rotation (the GPU computes closed-form phases)
    ideal = x * np.exp(1j * theta * (np.arange(n, dtype=np.float64) + 1))
    assert np.abs(got - ideal).max() < 1e-4
    # vs the oracle (reference semantics: iterated multiply, which drifts
    # from ideal by O(n*eps) — bound the comparison accordingly)
    ref, ref_phase = oracle_lib.rotator(theta, x)
    assert np.abs(got - ref).max() < 5e-4
    assert abs(final - cmath.exp(1j * theta * n)) < 1e-5


# ---------------- MovingAvg --------------------------------------------

def test_moving_avg_parity(gpu, oracle_lib):
    r = rng(97)
    w, d, h = 256, 0.1, 3
    frames = 32
    x = r.uniform(-1, 1, frames * w).astype(np.float32)
    f = gpu.MovingAvg(w, d, h)
    got, c, p, s = f.filter(x, frames * w)
    ref, co, po, avg, i = oracle_lib.moving_avg(w, d, h, x, frames * w)
    assert (c, p) == (co, po)
    assert_close(got, ref, 1e-6)
    # state carries across calls (block is stateful)
    x2 = r.uniform(-1, 1, 5 * w).astype(np.float32)
    got2, c2, p2, s2 = f.filter(x2, 5 * w)
    ref2, co2, po2, avg, i = oracle_lib.moving_avg(w, d, h, x2, 5 * w,
                                                   avg=avg, i_state=i)
    assert (c2, p2) == (co2, po2)
    assert_close(got2, ref2, 1e-6)


# ---------------- XlatingFir -------------------------------------------

def test_xlating_fir_parity(gpu, oracle_lib):
    """xlating_fir.rs semantics: bpf-tap decimating FIR + output rotator.
    The oracle pipeline composes oracle_decim_fir_ccf32 with the
    reference's iterated rotator; the GPU uses closed-form phases, so the
    comparison tolerance covers the oracle's own O(n*eps) phase drift."""
    r = rng(117)
    decim, offset, fs = 4, 12_000.0, 1_000_000.0
    taps = r.uniform(-1, 1, 63).astype(np.float32)
    x = cplx(r, 20000)
    f = gpu.XlatingFir(taps, decim, offset, fs)
    got, c, p, s = f.filter(x, 20000)
    # oracle composition
    i = np.arange(taps.size, dtype=np.float32)
    ang = i * np.float32(2 * np.pi) * np.float32(offset) / np.float32(fs)
    bpf = (np.cos(ang) + 1j * np.sin(ang)).astype(np.complex64) * taps
    ref, co, po, so = oracle_lib.decim_fir_ccf32(decim, bpf, x, 20000)
    theta = -2 * np.pi * offset * decim / fs
    ref_rot, _ = oracle_lib.rotator(theta, ref)
    assert (c, p, s) == (co, po, so)
    assert_close(got, ref_rot, 5e-4)
    # stateful phase across calls
    got2, c2, p2, s2 = f.filter(x, 20000)
    ideal2 = ref * np.exp(1j * theta * (np.arange(p) + 1 + p)).astype(
        np.complex64)
    assert_close(got2, ideal2, 5e-4)


# ---------------- PFB channelizer --------------------------------------

@pytest.mark.parametrize("N,n_taps,n_in", [(8, 64, 4096), (4, 37, 2000),
                                           (16, 128, 8192)])
def test_pfb_channelizer_parity(gpu, oracle_lib, N, n_taps, n_in):
    r = rng(N * 1000 + n_taps)
    taps = r.uniform(-1, 1, n_taps).astype(np.float32)
    x = cplx(r, n_in)
    got = gpu.PfbChannelizer(N, taps).run(x)
    ref = oracle_lib.pfb_channelizer(N, N, taps, x,
                                     max(1, (n_in) // N))
    assert got.shape == ref.shape
    assert_close(got, ref, 1e-4)


def test_pfb_channelizer_tone_isolation_gpu(gpu, oracle_lib):
    N = 8
    taps = gpu.kaiser_lowpass(1.0 / (2 * N), 0.05, 1e-3)
    m = np.arange(8192, dtype=np.float64)
    x = np.exp(2j * np.pi * (3 / N) * m).astype(np.complex64)
    ch = gpu.PfbChannelizer(N, taps).run(x)
    e = (np.abs(ch[:, 10:]) ** 2).sum(axis=1)
    assert e[3] / e.sum() > 0.95


@pytest.mark.parametrize("oversample", [2.0, 4.0])
def test_pfb_channelizer_oversample_parity(gpu, oracle_lib, oversample):
    """Oversampled (D = N/oversample < N): the round-robin base rotates
    every step (channelizer.rs:126-135) — GPU bulk closed form vs the
    oracle restatement with the same D."""
    N = 8
    D = int(N / oversample)
    r = rng(163 + D)
    taps = r.uniform(-1, 1, 64).astype(np.float32)
    x = cplx(r, 4096)
    got = gpu.PfbChannelizer(N, taps, oversample_rate=oversample).run(x)
    ref = oracle_lib.pfb_channelizer(N, D, taps, x, got.shape[1] + 8)
    assert got.shape[1] == ref.shape[1]
    assert_close(got, ref, 1e-5)


@pytest.mark.parametrize("over", [1.0, 2.0, 4.0])
def test_pfb_channelizer_streaming(gpu, oracle_lib, over):
    """Arbitrary chunk sizes through the stateful streaming path; the
    caller re-presents the unconsumed tail (slab semantics). The
    concatenated channel outputs equal the one-shot oracle."""
    N = 8
    D = int(N / over)
    r = rng(167 + D)
    taps = r.uniform(-1, 1, 64).astype(np.float32)
    n_total = 6000
    x = cplx(r, n_total)
    ch = gpu.PfbChannelizer(N, taps, oversample_rate=over)
    outs = []
    leftover = np.zeros(0, np.complex64)
    off = 0
    while off < n_total or leftover.size:
        take = int(r.integers(7, 700))
        buf = np.concatenate([leftover, x[off:off + take]])
        off += take
        out, cons = ch.stream(buf)
        assert cons <= buf.size
        if out.shape[1]:
            outs.append(out)
        leftover = buf[cons:]
        if off >= n_total and cons == 0:
            break
    got = np.concatenate(outs, axis=1)
    ref = oracle_lib.pfb_channelizer(N, D, taps, x, got.shape[1] + 8)
    assert got.shape[1] <= ref.shape[1]
    assert_close(got, ref[:, :got.shape[1]], 1e-5)


# ---------------- WLAN sync-short autocorrelation chain ----------------

def test_wlan_ops_parity(gpu, oracle_lib):
    r = rng(131)
    a, b = cplx(r, 3000), cplx(r, 3000)
    got = gpu.cmul_conj_host(a, b)
    ref = oracle_lib.cmul_conj(a, b)
    assert_close(got, ref, 1e-6)
    x = cplx(r, 5000)
    got = gpu.wlan_moving_sum_host(x, 48)
    ref = oracle_lib.wlan_moving_sum(x, 48)
    assert got.size == ref.size
    assert_close(got, ref, 1e-5)
    xf = r.uniform(-1, 1, 5000).astype(np.float32)
    got = gpu.wlan_moving_sum_host(xf, 64)
    ref = oracle_lib.wlan_moving_sum(xf, 64)
    assert_close(got, ref, 1e-5)


def test_wlan_sync_short_metric_chain(gpu, oracle_lib):
    """The rx.rs:73-96 autocorrelation front end on a synthesized 802.11
    STF-like preamble (16-sample periodic pattern): the correlation
    metric |movsum48(x*conj(delay16(x)))| / movsum64(|x|^2) plateaus
    near 1 inside the preamble and stays low in noise. All compute
    stages on GPU, composed through the ABI."""
    r = rng(137)
    stf = cplx(r, 16)  # one short-training symbol
    preamble = np.tile(stf, 10)  # 160-sample STF
    noise = 0.1 * cplx(r, 400)
    sig = np.concatenate([noise[:200], preamble, noise[200:]])
    n = sig.size
    # GPU chain
    delayed = np.concatenate([np.zeros(16, np.complex64), sig[:-16]])
    prod = gpu.cmul_conj_host(sig, delayed)
    corr = gpu.wlan_moving_sum_host(prod, 48)[: n]
    mag = np.abs(sig) ** 2
    power = gpu.wlan_moving_sum_host(mag.astype(np.float32), 64)[: n]
    metric = np.abs(corr) / np.maximum(power, 1e-9)
    # inside the plateau (after warmup of delay+windows)
    inside = metric[200 + 80:200 + 150]
    outside = metric[:150]
    assert inside.min() > 0.6, inside.min()
    assert np.median(outside) < 0.4, np.median(outside)


# ---------------- randomized chain fuzz --------------------------------

@pytest.mark.parametrize("seed", range(10))
def test_chain_fuzz_vs_oracle(gpu, oracle_lib, seed):
    """Randomized tap counts / FFT lengths / stream sizes through the
    fused chain vs the two-stage oracle. The single-kernel fused path now
    engages for every pow2 fft_len 64..1024 and every MFMA K template
    (20/32/48/80/144), so this fuzz covers the generalized kernel, not
    just the 1024/K=80 bench shape."""
    r = rng(10_000 + seed)
    nt1 = int(r.integers(8, 250))
    nt2 = int(r.integers(8, 250))
    fft_len = int(r.choice([64, 128, 256, 512, 1024]))
    n_in = int(r.integers(4 * fft_len * 2 + nt1 + nt2, 200_000))
    t1 = r.uniform(-1, 1, nt1).astype(np.float32)
    t2 = r.uniform(-1, 1, nt2).astype(np.float32)
    x = cplx(r, n_in)
    lib = gpu.lib()
    d_in = ctypes.c_void_p()
    d_out = ctypes.c_void_p()
    assert lib.fsdr_dev_alloc(ctypes.byref(d_in), n_in * 8) == 0
    assert lib.fsdr_dev_alloc(ctypes.byref(d_out), n_in * 8) == 0
    try:
        lib.fsdr_memcpy_h2d(d_in, ctypes.c_void_p(x.ctypes.data), n_in * 8)
        ch = gpu.Chain(t1, t2, 4, fft_len)
        cons, prod = ch.run_dev(d_in.value, n_in, d_out.value, n_in)
        gpu.synchronize()
        ref, cons_ref = oracle_lib.chain_cf32(t1, t2, 4, fft_len, x)
        assert (cons, prod) == (cons_ref, ref.size), (nt1, nt2, fft_len)
        got = np.zeros(prod, np.complex64)
        lib.fsdr_memcpy_d2h(ctypes.c_void_p(got.ctypes.data), d_out,
                            prod * 8)
        if prod:
            rel = np.linalg.norm(got - ref) / np.linalg.norm(ref)
            assert rel < 2e-4, (nt1, nt2, fft_len, rel)
    finally:
        lib.fsdr_dev_free(d_in)
        lib.fsdr_dev_free(d_out)


def test_divide_mag_parity(gpu):
    r = rng(139)
    a = cplx(r, 4000)
    b = (r.uniform(0.1, 2.0, 4000)).astype(np.float32)
    got = gpu.divide_mag_host(a, b)
    ref = (np.abs(a) / b).astype(np.float32)
    assert_close(got, ref, 1e-5)


def _chain_stream_case(gpu, oracle_lib, fft_len, chunk_sizes, seed,
                       decim=4):
    """Stream arbitrary chunk sizes through ring -> fused chain; the
    carried unconsumed tail (input remainder AND non-frame-aligned
    decimated leftovers, which stay upstream as unconsumed input) makes
    the concatenated spectra equal the one-shot oracle chain."""
    lib = gpu.lib()
    r = rng(seed)
    t1 = r.uniform(-1, 1, 127).astype(np.float32)
    t2 = r.uniform(-1, 1, 127).astype(np.float32)
    n_total = int(sum(chunk_sizes))
    x = cplx(r, n_total)
    g_len = t1.size + t2.size - 1
    # worst-case leftover: g_len-1 window history + an unfilled frame
    reserved = g_len - 1 + decim * fft_len + decim
    chain = gpu.Chain(t1, t2, decim, fft_len)
    max_chunk = max(chunk_sizes)
    out_cap = (reserved + max_chunk) // decim + fft_len
    d_out = ctypes.c_void_p()
    assert lib.fsdr_dev_alloc(ctypes.byref(d_out), out_cap * 8) == 0
    try:
        def consume(dp, n_in):
            cons, prod = chain.run_dev(dp, n_in, d_out.value, out_cap)
            gpu.synchronize()
            h = np.zeros(prod, np.complex64)
            lib.fsdr_memcpy_d2h(ctypes.c_void_p(h.ctypes.data), d_out,
                                prod * 8)
            return cons, h
        offs = np.concatenate([[0], np.cumsum(chunk_sizes)]).astype(int)
        chunks = [x[offs[i]:offs[i + 1]] for i in range(len(chunk_sizes))]
        got = _stream_through_ring(gpu, lib, chunks, reserved, consume)
        ref, _ = oracle_lib.chain_cf32(t1, t2, decim, fft_len, x)
        assert got.size == ref.size, (got.size, ref.size)
        rel = np.linalg.norm(got - ref) / np.linalg.norm(ref)
        assert rel < 2e-4, rel
    finally:
        lib.fsdr_dev_free(d_out)


def test_ring_chain_streaming(gpu, oracle_lib):
    """Production streaming path at an aligned chunk size (every chunk's
    decimated output is whole frames)."""
    fft_len = 256
    chunk = 4 * fft_len * 16
    _chain_stream_case(gpu, oracle_lib, fft_len, [chunk] * 8, 149)


def test_ring_chain_streaming_unaligned(gpu, oracle_lib):
    """Arbitrary (random) chunk sizes: non-frame-aligned leftovers must
    carry across fsdr_chain_run_dev calls like slab does."""
    r = rng(151)
    sizes = r.integers(257, 20000, size=12).tolist()
    _chain_stream_case(gpu, oracle_lib, 256, sizes, 153)


def test_ring_chain_streaming_tiny_chunks(gpu, oracle_lib):
    """Chunks smaller than the filter window: several acquires produce
    nothing until enough history accumulates."""
    r = rng(157)
    sizes = r.integers(40, 900, size=30).tolist()
    _chain_stream_case(gpu, oracle_lib, 64, sizes, 159)


def test_ring_full_loop_h2d_chain_d2h(gpu, oracle_lib):
    """The complete streaming loop: host chunks -> pinned H2D ring ->
    fused chain on the compute stream -> D2H return ring -> host
    consumer. Results equal the one-shot oracle chain. Exercises the
    d2h ring's event ordering (producer-stream wait, copy-stream D2H)."""
    lib = gpu.lib()
    r = rng(171)
    t1 = r.uniform(-1, 1, 127).astype(np.float32)
    t2 = r.uniform(-1, 1, 127).astype(np.float32)
    fft_len = 256
    chunk = 9000
    n_total = chunk * 6
    x = cplx(r, n_total)
    g_len = t1.size + t2.size - 1
    reserved = g_len - 1 + 4 * fft_len + 4
    out_cap = (reserved + chunk) // 4 + fft_len
    chain = gpu.Chain(t1, t2, 4, fft_len)
    ring = lib.fsdr_ring_create(4, chunk, 8, reserved)
    d2h = lib.fsdr_ring_d2h_create(4, out_cap, 8)
    assert ring and d2h
    outs = []
    try:
        for off in range(0, n_total, chunk):
            hp = ctypes.c_void_p()
            items = ctypes.c_size_t()
            assert lib.fsdr_ring_writer_acquire(
                ring, ctypes.byref(hp), ctypes.byref(items)) == 0
            ctypes.memmove(hp, ctypes.c_void_p(x[off:off + chunk]
                                               .ctypes.data), chunk * 8)
            assert lib.fsdr_ring_writer_commit(ring, chunk) == 0
            dp = ctypes.c_void_p()
            got_items = ctypes.c_size_t()
            assert lib.fsdr_ring_reader_acquire(
                ring, ctypes.byref(dp), ctypes.byref(got_items)) == 0
            op = ctypes.c_void_p()
            ocap = ctypes.c_size_t()
            assert lib.fsdr_ring_d2h_writer_acquire(
                d2h, ctypes.byref(op), ctypes.byref(ocap), None) == 0
            cons, prod = chain.run_dev(dp.value, got_items.value,
                                       op.value, ocap.value)
            assert lib.fsdr_ring_d2h_writer_commit(d2h, prod, None) == 0
            assert lib.fsdr_ring_reader_release_consumed(
                ring, cons, None) == 0
            hp2 = ctypes.c_void_p()
            n2 = ctypes.c_size_t()
            assert lib.fsdr_ring_d2h_reader_acquire(
                d2h, ctypes.byref(hp2), ctypes.byref(n2)) == 0
            if n2.value:
                h = np.zeros(n2.value, np.complex64)
                ctypes.memmove(ctypes.c_void_p(h.ctypes.data), hp2,
                               n2.value * 8)
                outs.append(h)
            assert lib.fsdr_ring_d2h_reader_release(d2h) == 0
        got = np.concatenate(outs)
        ref, _ = oracle_lib.chain_cf32(t1, t2, 4, fft_len, x)
        assert got.size == ref.size
        rel = np.linalg.norm(got - ref) / np.linalg.norm(ref)
        assert rel < 2e-4, rel
    finally:
        lib.fsdr_ring_destroy(ring)
        lib.fsdr_ring_d2h_destroy(d2h)


def test_moving_avg_single_emission_fast_path(gpu, oracle_lib):
    """history == frames triggers the parallel chunked-EMA path; the
    chunk composition reorders f32 ops, so tolerance-compared."""
    r = rng(151)
    w, d = 1024, 0.1
    frames = 1000
    x = r.uniform(-1, 1, frames * w).astype(np.float32)
    f = gpu.MovingAvg(w, d, frames)
    got, c, p, s = f.filter(x, w)
    ref, co, po, avg, i = oracle_lib.moving_avg(w, d, frames, x, w)
    assert (c, p) == (co, po) and p == w
    assert_close(got, ref, 1e-4)
    # second step continues the EMA state
    x2 = r.uniform(-1, 1, frames * w).astype(np.float32)
    got2, c2, p2, _ = f.filter(x2, w)
    ref2, _, _, _, _ = oracle_lib.moving_avg(w, d, frames, x2, w,
                                             avg=avg, i_state=i)
    assert_close(got2, ref2, 1e-4)
