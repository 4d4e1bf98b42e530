"""Constructor-argument validation (mirrors the reference's asserts).
Runs on CPU: semantic validation precedes the GPU check in every create,
so the error message distinguishes bad arguments from missing hardware.
"""


def _err(fsdr):
    return fsdr.lib().fsdr_last_error().decode()


def test_resampler_taps_divisibility(fsdr):
    # polyphase_resampling_fir.rs:54-56 assert
    import numpy as np
    taps = np.ones(7, np.float32)
    h = fsdr.lib().fsdr_resamp_cf32_create(
        3, 2, taps.ctypes.data_as(
            __import__("ctypes").POINTER(__import__("ctypes").c_float)), 7)
    assert not h
    assert "multiple" in _err(fsdr) or "interp" in _err(fsdr)


def test_moving_avg_decay_range(fsdr):
    # moving_avg.rs:59-62 assert
    h = fsdr.lib().fsdr_moving_avg_create(16, 1.5, 3)
    assert not h and "decay" in _err(fsdr)


def test_xlating_decimation_min(fsdr):
    # xlating_fir.rs:44 assert
    import ctypes
    import numpy as np
    taps = np.ones(8, np.float32)
    h = fsdr.lib().fsdr_xlating_fir_cf32_create(
        taps.ctypes.data_as(ctypes.POINTER(ctypes.c_float)), 8, 1, 0.0,
        1e6)
    assert not h and "decimation" in _err(fsdr)


def test_pfb_constraints(fsdr):
    # channelizer.rs:94-106 asserts + GPU-path constraints
    import ctypes
    import numpy as np
    taps = np.ones(32, np.float32)
    p = taps.ctypes.data_as(ctypes.POINTER(ctypes.c_float))
    assert not fsdr.lib().fsdr_pfb_channelizer_create(2, p, 32, 1.0)
    assert not fsdr.lib().fsdr_pfb_channelizer_create(8, p, 4, 1.0)
    assert not fsdr.lib().fsdr_pfb_channelizer_create(8, p, 32, 2.0)


def test_fft_len_gate(fsdr):
    # pow2 lengths: [4,4096]; any other length: [2,2048] via Bluestein
    assert not fsdr.lib().fsdr_fft_cf32_create(1, 0, 0, None)
    assert not fsdr.lib().fsdr_fft_cf32_create(8192, 0, 0, None)
    assert not fsdr.lib().fsdr_fft_cf32_create(3000, 0, 0, None)
    assert "Bluestein" in _err(fsdr)
