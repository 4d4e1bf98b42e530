"""Product firdes (host-side C++ in the ABI) vs oracle — runs on CPU.

The tap designers are host code in the reference too (firdes/basic.rs),
so the product ships its own; parity against the KAT-pinned oracle.
"""
import numpy as np


def test_product_kaiser_lowpass_matches_oracle(fsdr, oracle_lib):
    for (c, t, r) in [(0.2, 0.05, 0.01), (0.1, 0.02, 1e-4),
                      (0.05, 0.01, 1e-3)]:
        tp = fsdr.kaiser_lowpass(c, t, r)
        to = oracle_lib.kaiser_lowpass_f32(c, t, r)
        assert tp.size == to.size
        np.testing.assert_array_equal(tp, to)


def test_product_lowpass_kaiser_n_matches_oracle(fsdr, oracle_lib):
    beta = fsdr.kaiser_beta(1e-4)
    assert abs(beta - 0.1102 * (80.0 - 8.7)) < 1e-9  # basic.rs:448
    tp = fsdr.lowpass_kaiser_n(127, beta, 0.1)
    to = np.float32(oracle_lib.firdes_lowpass(
        0.1, oracle_lib.kaiser_window(127, beta)))
    np.testing.assert_array_equal(tp, to)
    np.testing.assert_allclose(tp, tp[::-1])  # linear phase
    assert abs(tp.sum() - 1.0) < 1e-4  # unit DC gain lowpass
