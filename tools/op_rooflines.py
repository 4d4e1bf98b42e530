#!/usr/bin/env python3
"""Standalone rooflines for the off-metric-path kernels (VERDICT weak-3):
XlatingFir (tiled), FirCC (WLAN correlator shape + bulk), resampler
shapes, and the standalone FFT kernel. Prints TF/s or GB/s vs peak."""
import ctypes
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import futuresdr_amd as fa  # noqa: E402

PEAK_TF = 157.3
PEAK_HBM = 8.0  # TB/s


def timeit(fn, reps=10, warm=3):
    st = torch.cuda.current_stream()
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    e0 = torch.cuda.Event(enable_timing=True)
    e1 = torch.cuda.Event(enable_timing=True)
    e0.record(st)
    for _ in range(reps):
        fn()
    e1.record(st)
    torch.cuda.synchronize()
    return e0.elapsed_time(e1) / reps


def main():
    lib = fa.lib()
    fa.set_device(0)
    st = torch.cuda.current_stream()
    S = 1 << 26
    d_in = ctypes.c_void_p()
    d_out = ctypes.c_void_p()
    assert lib.fsdr_dev_alloc(ctypes.byref(d_in), S * 8) == 0
    assert lib.fsdr_dev_alloc(ctypes.byref(d_out), S * 8) == 0
    fa.fill_uniform_dev(d_in.value, S, seed=5, stream=st.cuda_stream)

    # XlatingFir: 127 taps, decim 4 (the FM front-end shape)
    taps = fa.lowpass_kaiser_n(127, fa.kaiser_beta(1e-4), 0.1)
    xl = fa.XlatingFir(taps, 4, 0.25, 1.0)
    ms = timeit(lambda: xl.filter_dev(d_in.value, S, d_out.value, S,
                                      stream=st.cuda_stream))
    prod = (S + 1 - 127) // 4
    # complex taps x complex samples: 8 flops/tap + rotator ~10
    tf = prod * (127 * 8 + 10) / (ms * 1e-3) / 1e12
    print(f"xlating_fir(127,D4) tiled: {ms:7.3f} ms  {tf:6.1f} TF/s "
          f"({tf / PEAK_TF:.3f} of fp32 peak)")

    # FirCC bulk (complex taps, D=1) — WLAN correlator math at scale
    ltf = np.exp(2j * np.pi * np.arange(64) / 7).astype(np.complex64)
    fcc = fa.FirCC(np.conj(ltf[::-1]))
    ms = timeit(lambda: fcc.filter_dev(d_in.value, S, d_out.value, S,
                                       stream=st.cuda_stream))
    prod = S + 1 - 64
    tf = prod * 64 * 8 / (ms * 1e-3) / 1e12
    print(f"fir_ccf32(64) bulk:        {ms:7.3f} ms  {tf:6.1f} TF/s "
          f"({tf / PEAK_TF:.3f} of fp32 peak)")

    # standalone FFT kernel (bulk 1024-pt)
    fft = fa.Fft(1024)
    frames = S // 1024
    ms = timeit(lambda: fft.bulk_dev(d_in.value, d_out.value, frames,
                                     stream=st.cuda_stream))
    gb = frames * 1024 * 16 / 1e9  # in+out
    tbs = gb / (ms * 1e-3) / 1e3
    flops = frames * 5 * 1024 * 10
    tf = flops / (ms * 1e-3) / 1e12
    print(f"fft_stockham(1024) bulk:   {ms:7.3f} ms  {tbs:6.2f} TB/s HBM "
          f"({tbs / PEAK_HBM:.3f} of peak), {tf:5.1f} TF/s")

    lib.fsdr_dev_free(d_in)
    lib.fsdr_dev_free(d_out)


if __name__ == "__main__":
    main()
