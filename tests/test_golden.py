"""Oracle vs committed golden vectors (tests/golden/golden.npz).

These run on CPU everywhere (no /root/reference access at run time).
"""
import os

import numpy as np
import pytest

GOLDEN = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden",
                      "golden.npz")


@pytest.fixture(scope="module")
def golden():
    return np.load(GOLDEN)


@pytest.mark.parametrize("n", [64, 256, 1024, 60, 100, 1000])
def test_oracle_fft_vs_golden(oracle_lib, golden, n):
    o = oracle_lib
    xs = golden[f"fft{n}_in"]
    refs = golden[f"fft{n}_out"]
    for x, ref in zip(xs, refs):
        got = o.dft_cf32(x)
        assert np.linalg.norm(got - ref) / np.linalg.norm(ref) < 1e-6


def test_oracle_fir_vs_golden(oracle_lib, golden):
    o = oracle_lib
    taps = golden["fir127_taps"]
    xin = golden["fir127_in"]
    ref = golden["fir127_out"]
    out, c, p, s = o.fir_cf32(taps, xin, ref.size)
    assert p == ref.size
    err = np.abs(out - ref).max() / np.abs(ref).max()
    assert err < 1e-5  # f32 oracle vs f64 reference
