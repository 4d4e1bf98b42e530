"""Generate committed golden vectors for the FFT parity pin.

The reference's FFT math lives in third-party rustfft 6.4 (Cargo.toml:82),
which has no in-repo KAT; SURVEY.md §8c pins FFT parity to the
unnormalized-DFT definition via numpy instead. This script (run in the
build container, where numpy is available) produces small fixtures that
travel with the repo; GPU-box tests compare against the .npz, never
against /root/reference.

Run: python tests/golden/generate.py
"""
import os

import numpy as np

HERE = os.path.dirname(os.path.abspath(__file__))


def main():
    rng = np.random.default_rng(0x5D5D5D5D)
    out = {}
    # pow2 (Stockham kernel) and non-pow2 (Bluestein path) lengths
    for n in (64, 256, 1024, 60, 100, 1000):
        x = (rng.uniform(-1, 1, (4, n)) +
             1j * rng.uniform(-1, 1, (4, n))).astype(np.complex64)
        X = np.fft.fft(x.astype(np.complex128), axis=1)
        out[f"fft{n}_in"] = x
        out[f"fft{n}_out"] = X.astype(np.complex128)
    # a small FIR golden set as well (f64 reference of the cf32 FIR)
    taps = rng.uniform(-1, 1, 127).astype(np.float32)
    xin = (rng.uniform(-1, 1, 2000) +
           1j * rng.uniform(-1, 1, 2000)).astype(np.complex64)
    n_out = xin.size + 1 - taps.size
    y = np.array([
        np.dot(xin[k:k + taps.size].astype(np.complex128), taps[::-1])
        for k in range(n_out)])
    out["fir127_taps"] = taps
    out["fir127_in"] = xin
    out["fir127_out"] = y
    np.savez_compressed(os.path.join(HERE, "golden.npz"), **out)
    print("wrote", os.path.join(HERE, "golden.npz"))


if __name__ == "__main__":
    main()
