#!/usr/bin/env python3
"""Spectrum analyzer pipeline — the MI355X counterpart of the reference's
examples/spectrum (src/bin/cpu.rs:21-28 shape):

    source -> Fir(lowpass) -> DecimFir(4) -> Fft(1024) -> |X|^2
           -> MovingAvg -> sink

Run on a box with an MI355X:  python examples/spectrum.py
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import futuresdr_amd as fa  # noqa: E402


def main():
    if fa.device_count() < 1:
        raise SystemExit("needs a HIP device (MI355X)")
    fs = 1.0  # normalized sample rate
    n = 4 * 1024 * 64 + 4096
    rng = np.random.default_rng(0)
    # two tones + noise
    t = np.arange(n)
    x = (0.5 * np.exp(2j * np.pi * 0.0123 * t)
         + 0.25 * np.exp(2j * np.pi * 0.0441 * t)
         + 0.05 * (rng.standard_normal(n) + 1j * rng.standard_normal(n))
         ).astype(np.complex64)

    taps1 = fa.kaiser_lowpass(0.10, 0.02, 1e-4)
    taps2 = fa.lowpass_kaiser_n(127, fa.kaiser_beta(1e-4), 0.11)

    fg = fa.Flowgraph()
    src = fg.vector_source(x)
    f1 = fg.filter(fa.Fir(taps1))
    f2 = fg.filter(fa.DecimFir(4, taps2))
    f3 = fg.filter(fa.Fft(1024))
    f4 = fg.filter(fa.Mag2())
    f5 = fg.filter(fa.MovingAvg(1024, 0.1, 16))
    snk = fg.vector_sink()
    fg.connect(src, f1, f2, f3, f4, f5, snk)
    fg.run()

    frames = fg.sink_data(snk, np.float32).reshape(-1, 1024)
    spec = frames[-1]
    # decimate-by-4: input tone f appears at bin f*4*1024
    top = np.argsort(spec)[-4:][::-1]
    print("averaged frames:", frames.shape[0])
    print("top bins:", sorted(int(b) for b in top),
          "(expect clusters near", int(0.0123 * 4 * 1024),
          "and", int(0.0441 * 4 * 1024), ")")


if __name__ == "__main__":
    main()
