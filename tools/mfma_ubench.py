#!/usr/bin/env python3
"""Intrinsic ceiling of the decim kernel's MFMA loop (no staging/FFT):
sweep grid sizes (occupancy) and print TF/s vs the 157.3 fp32 peak."""
import ctypes
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import futuresdr_amd as fa  # noqa: E402


def main():
    lib = fa.lib()
    lib.fsdr_mfma_ubench.restype = ctypes.c_int
    lib.fsdr_mfma_ubench.argtypes = [ctypes.c_int, ctypes.c_int,
                                     ctypes.POINTER(ctypes.c_double),
                                     ctypes.c_void_p]
    for grid in (256, 512, 1024, 1536, 2048, 4096, 8192):
        tf = ctypes.c_double()
        rc = lib.fsdr_mfma_ubench(grid, 2000, ctypes.byref(tf), None)
        assert rc == 0, fa.lib().fsdr_last_error().decode()
        print(f"grid={grid:5d} ({grid // 256:3d} blk/CU): "
              f"{tf.value:7.1f} TF/s  ({tf.value / 157.3:.3f} of peak)")
    lib.fsdr_chain_ubench.restype = ctypes.c_int
    lib.fsdr_chain_ubench.argtypes = [ctypes.c_int,
                                      ctypes.POINTER(ctypes.c_double),
                                      ctypes.c_void_p]
    names = {0: "loop+outwrites only", 1: "+global staging loads",
             2: "+LDS writes+barriers", 3: "+loads+LDS+barriers",
             4: "+fft only", 6: "+LDS+fft", 7: "FULL kernel shape"}
    for mode in (0, 1, 2, 3, 4, 6, 7):
        tf = ctypes.c_double()
        rc = lib.fsdr_chain_ubench(mode, ctypes.byref(tf), None)
        assert rc == 0, fa.lib().fsdr_last_error().decode()
        print(f"chain mode={mode} ({names[mode]:24s}): "
              f"{tf.value:7.1f} TF/s  ({tf.value / 157.3:.3f} of peak)")


if __name__ == "__main__":
    main()
