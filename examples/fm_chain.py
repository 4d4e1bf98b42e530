#!/usr/bin/env python3
"""FM-receiver front-end shape (BASELINE configs[2], the reference's
examples/fm-receiver decimation chain): Fir lowpass -> polyphase
resampler 4:1 -> 1024-pt FFT, via the native flowgraph driver."""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import futuresdr_amd as fa  # noqa: E402


def main():
    if fa.device_count() < 1:
        raise SystemExit("needs a HIP device (MI355X)")
    rng = np.random.default_rng(1)
    x = (rng.uniform(-1, 1, (4 * 1024 * 32 + 4096, 2)) @ [1, 1j]).astype(
        np.complex64)
    taps1 = fa.kaiser_lowpass(0.1, 0.02, 1e-4)
    taps2 = rng.uniform(-1, 1, 128).astype(np.float32)  # 128 % interp(1)==0

    fg = fa.Flowgraph()
    src = fg.vector_source(x)
    f1 = fg.filter(fa.Fir(taps1))
    f2 = fg.filter(fa.Resampler(1, 4, taps2))
    f3 = fg.filter(fa.Fft(1024))
    snk = fg.vector_sink()
    fg.connect(src, f1, f2, f3, snk)
    fg.run()
    out = fg.sink_data(snk)
    print("spectra frames:", out.size // 1024)


if __name__ == "__main__":
    main()
