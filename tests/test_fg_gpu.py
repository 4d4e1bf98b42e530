"""Flowgraph-driver tests (GPU): the native C++ harness reproduces the
reference runtime behavior for chains, including perf/fir's correctness
guard `n_received == samples - stages*(taps-1)` (perf/fir/fir.rs:97).
"""
import numpy as np
import pytest

pytestmark = pytest.mark.gpu


def rng(s=0):
    return np.random.default_rng(s)


def cplx(r, n):
    return (r.uniform(-1, 1, (n, 2)) @ [1, 1j]).astype(np.complex64)


def assert_close(got, ref, tol=1e-5):
    got, ref = np.asarray(got), np.asarray(ref)
    scale = max(1.0, float(np.abs(ref).max()) if ref.size else 1.0)
    err = float(np.abs(got - ref).max()) if ref.size else 0.0
    assert err <= tol * scale, f"max err {err} > {tol}*{scale}"


def test_fg_perf_fir_shape(gpu):
    """NullSource -> Head(samples) -> Fir(64 taps) x stages -> NullSink:
    the perf/fir flowgraph (perf/fir/fir.rs:50-74) with its correctness
    guard n_received == samples - stages*63 (:97)."""
    samples, stages = 1_500_000, 3
    taps = rng(1).uniform(-1, 1, 64).astype(np.float32)
    fg = gpu.Flowgraph()
    src = fg.null_source()
    head = fg.head(samples)
    fg.stream(src, head)
    last = head
    for _ in range(stages):
        f = fg.filter(gpu.Fir(taps))
        fg.stream(last, f)
        last = f
    snk = fg.null_sink()
    fg.stream(last, snk)
    fg.run()
    assert fg.n_received(snk) == samples - stages * 63


def test_fg_vector_roundtrip_vs_oracle(gpu, oracle_lib):
    r = rng(2)
    x = cplx(r, 300000)
    taps = r.uniform(-1, 1, 127).astype(np.float32)
    fg = gpu.Flowgraph()
    src = fg.vector_source(x)
    f = fg.filter(gpu.Fir(taps))
    snk = fg.vector_sink()
    fg.connect(src, f, snk)
    fg.run()
    got = fg.sink_data(snk)
    ref, c, p, s = oracle_lib.fir_cf32(taps, x, x.size)
    assert got.size == p
    err = np.abs(got - ref).max() / max(1.0, np.abs(ref).max())
    assert err < 1e-5


def test_fg_full_chain_vs_oracle(gpu, oracle_lib):
    """VectorSource -> Fir -> DecimFir(4) -> Fft(256) -> Mag2 -> VectorSink
    (the spectrum chain, examples/spectrum/src/bin/cpu.rs:21-28 shape)."""
    r = rng(3)
    x = cplx(r, 4 * 256 * 6 + 600)
    t1 = r.uniform(-1, 1, 127).astype(np.float32)
    t2 = r.uniform(-1, 1, 127).astype(np.float32)
    fg = gpu.Flowgraph()
    src = fg.vector_source(x)
    f1 = fg.filter(gpu.Fir(t1))
    f2 = fg.filter(gpu.DecimFir(4, t2))
    f3 = fg.filter(gpu.Fft(256))
    f4 = fg.filter(gpu.Mag2())
    snk = fg.vector_sink()
    fg.connect(src, f1, f2, f3, f4, snk)
    fg.run()
    got = fg.sink_data(snk, np.float32)
    spectra, _ = oracle_lib.chain_cf32(t1, t2, 4, 256, x)
    ref = np.abs(spectra) ** 2
    assert got.size == ref.size
    err = np.abs(got - ref).max() / max(1.0, ref.max())
    assert err < 1e-4


def test_fg_small_buffer_many_chunks(gpu, oracle_lib):
    """Stream much more data than one edge buffer holds — exercises the
    compaction (slab tail-copy) path repeatedly."""
    r = rng(4)
    x = cplx(r, 1 << 20)  # 4x the default edge capacity
    taps = r.uniform(-1, 1, 63).astype(np.float32)
    fg = gpu.Flowgraph()
    src = fg.vector_source(x)
    f = fg.filter(gpu.Fir(taps))
    snk = fg.vector_sink()
    fg.connect(src, f, snk)
    fg.run()
    got = fg.sink_data(snk)
    assert got.size == x.size - 62
    ref, _, p, _ = oracle_lib.fir_cf32(taps, x, x.size)
    err = np.abs(got - ref).max() / max(1.0, np.abs(ref).max())
    assert err < 1e-5


def test_fg_spectrum_sink_with_moving_avg(gpu, oracle_lib):
    """Full spectrum pipeline incl. the averaged sink:
    VectorSource -> Fir -> DecimFir(4) -> Fft(64) -> Mag2 ->
    MovingAvg(64, 0.1, 2) -> VectorSink (examples/spectrum shape with
    moving_avg.rs smoothing)."""
    r = rng(5)
    x = cplx(r, 4 * 64 * 9 + 600)
    t1 = r.uniform(-1, 1, 63).astype(np.float32)
    t2 = r.uniform(-1, 1, 63).astype(np.float32)
    fg = gpu.Flowgraph()
    src = fg.vector_source(x)
    f1 = fg.filter(gpu.Fir(t1))
    f2 = fg.filter(gpu.DecimFir(4, t2))
    f3 = fg.filter(gpu.Fft(64))
    f4 = fg.filter(gpu.Mag2())
    f5 = fg.filter(gpu.MovingAvg(64, 0.1, 2))
    snk = fg.vector_sink()
    fg.connect(src, f1, f2, f3, f4, f5, snk)
    fg.run()
    got = fg.sink_data(snk, np.float32)
    import oracle as o
    y1, _, _, _ = o.fir_cf32(t1, x, x.size)
    y2, _, _, _ = o.decim_fir_cf32(4, t2, y1, y1.size)
    frames = y2.size // 64
    spec = np.concatenate([o.dft_cf32(y2[i * 64:(i + 1) * 64])
                           for i in range(frames)])
    mags = o.mag2(spec)
    ref, _, _, _, _ = o.moving_avg(64, 0.1, 2, mags, mags.size)
    assert got.size == ref.size
    assert_close(got, ref, 1e-4)


def test_fg_xlating_fir_in_graph(gpu, oracle_lib):
    """VectorSource -> XlatingFir -> VectorSink through the native actor
    loop: the block's stateful rotator phase must advance correctly
    across the driver's chunked work() calls, matching the single-shot
    oracle composition (xlating_fir.rs:27-94 semantics)."""
    r = rng(191)
    decim, offset, fs = 4, 12_000.0, 1_000_000.0
    taps = r.uniform(-1, 1, 63).astype(np.float32)
    n = 120_000
    x = cplx(r, n)
    import futuresdr_amd as fa
    fg = fa.Flowgraph()
    src = fg.vector_source(x)
    xl = fg.filter(fa.XlatingFir(taps, decim, offset, fs))
    snk = fg.vector_sink()
    fg.connect(src, xl, snk)
    fg.run()
    got = fg.sink_data(snk)
    i = np.arange(taps.size, dtype=np.float32)
    ang = i * np.float32(2 * np.pi) * np.float32(offset) / np.float32(fs)
    bpf = (np.cos(ang) + 1j * np.sin(ang)).astype(np.complex64) * taps
    ref, co, po, so = oracle_lib.decim_fir_ccf32(decim, bpf, x, n)
    # the block stores the f32-rounded phase increment (like the
    # reference's f32 Rotator) and rotates ideally BY THAT increment;
    # pin against the f32-theta ideal — an f64-theta ideal diverges by
    # k*ulp(theta) ~ 5e-4 rad by output 30k, and the oracle's iterated
    # rotator drifts even more over this span.
    # replicate the create's sequential f32 arithmetic exactly
    theta32 = np.float32(np.float32(np.float32(
        np.float32(-6.2831853071795864769) * np.float32(offset))
        * np.float32(decim)) / np.float32(fs))
    ref_rot = ref * np.exp(
        1j * float(theta32) * (np.arange(ref.size, dtype=np.float64) + 1))
    assert got.size == ref_rot.size
    assert_close(got, ref_rot.astype(np.complex64), 5e-4)


def test_fg_config3_resampler_chain(gpu, oracle_lib):
    """BASELINE configs[2] shape with the polyphase resampler:
    VectorSource -> Fir -> Resampler(1,4) -> Fft(256) -> VectorSink."""
    r = rng(6)
    x = cplx(r, 4 * 256 * 5 + 800)
    t1 = r.uniform(-1, 1, 127).astype(np.float32)
    t2 = r.uniform(-1, 1, 128).astype(np.float32)
    fg = gpu.Flowgraph()
    src = fg.vector_source(x)
    f1 = fg.filter(gpu.Fir(t1))
    f2 = fg.filter(gpu.Resampler(1, 4, t2))
    f3 = fg.filter(gpu.Fft(256))
    snk = fg.vector_sink()
    fg.connect(src, f1, f2, f3, snk)
    fg.run()
    got = fg.sink_data(snk)
    import oracle as o
    y1, _, _, _ = o.fir_cf32(t1, x, x.size)
    y2, _, _, _ = o.resamp_cf32(1, 4, t2, y1, y1.size)
    frames = y2.size // 256
    ref = np.concatenate([o.dft_cf32(y2[i * 256:(i + 1) * 256])
                          for i in range(frames)])
    assert got.size == ref.size
    rel = np.linalg.norm(got - ref) / np.linalg.norm(ref)
    assert rel < 1e-4
