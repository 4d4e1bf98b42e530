"""oracle — TEST INFRASTRUCTURE ONLY.

ctypes bindings for the CPU restatement of FutureSDR's futuredsp hot-path
cores (see oracle/oracle.h). Only tests/, __graft_entry__.smoke() and
bench.py's cpu_baseline leg may import this package; the product
(futuresdr_amd/) never touches it and fails loudly without its HIP
extension instead.
"""
import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "liboracle.so")

CF32 = np.dtype(np.complex64)

INSUFFICIENT_INPUT = 0
INSUFFICIENT_OUTPUT = 1
BOTH_SUFFICIENT = 2


class _Result(ctypes.Structure):
    _fields_ = [
        ("consumed", ctypes.c_size_t),
        ("produced", ctypes.c_size_t),
        ("status", ctypes.c_int),
    ]


def build():
    """Compile oracle/liboracle.so (gcc, seconds)."""
    subprocess.run(["make", "-s", "-C", _DIR], check=True)


_lib = None


def _load():
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(_SO):
        build()
    lib = ctypes.CDLL(_SO)
    szp = ctypes.POINTER(ctypes.c_size_t)
    f32p = ctypes.POINTER(ctypes.c_float)
    f64p = ctypes.POINTER(ctypes.c_double)
    vp = ctypes.c_void_p
    sz = ctypes.c_size_t

    lib.oracle_fir_f32.restype = _Result
    lib.oracle_fir_f32.argtypes = [f32p, sz, vp, sz, vp, sz]
    lib.oracle_fir_cf32.restype = _Result
    lib.oracle_fir_cf32.argtypes = [f32p, sz, vp, sz, vp, sz]
    lib.oracle_fir_ccf32.restype = _Result
    lib.oracle_fir_ccf32.argtypes = [vp, sz, vp, sz, vp, sz]
    lib.oracle_decim_fir_f32.restype = _Result
    lib.oracle_decim_fir_f32.argtypes = [sz, f32p, sz, vp, sz, vp, sz]
    lib.oracle_decim_fir_cf32.restype = _Result
    lib.oracle_decim_fir_cf32.argtypes = [sz, f32p, sz, vp, sz, vp, sz]
    lib.oracle_decim_fir_ccf32.restype = _Result
    lib.oracle_decim_fir_ccf32.argtypes = [sz, vp, sz, vp, sz, vp, sz]
    lib.oracle_resamp_f32.restype = _Result
    lib.oracle_resamp_f32.argtypes = [sz, sz, f32p, sz, vp, sz, vp, sz]
    lib.oracle_resamp_cf32.restype = _Result
    lib.oracle_resamp_cf32.argtypes = [sz, sz, f32p, sz, vp, sz, vp, sz]
    lib.oracle_dft_cf32.restype = None
    lib.oracle_dft_cf32.argtypes = [ctypes.c_int, ctypes.c_int, vp, vp]
    lib.oracle_fft_block.restype = sz
    lib.oracle_fft_block.argtypes = [sz, ctypes.c_int, ctypes.c_int, f32p,
                                     vp, sz, vp, sz]
    lib.oracle_mag2.restype = sz
    lib.oracle_mag2.argtypes = [vp, sz, vp, sz]
    lib.oracle_cmul.restype = sz
    lib.oracle_cmul.argtypes = [vp, sz, vp, sz, vp, sz]
    lib.oracle_rotator.restype = sz
    lib.oracle_rotator.argtypes = [ctypes.c_float, vp, vp, sz, vp, sz]
    lib.oracle_besseli0.restype = ctypes.c_double
    lib.oracle_besseli0.argtypes = [ctypes.c_double]
    lib.oracle_kaiser_window.restype = None
    lib.oracle_kaiser_window.argtypes = [sz, ctypes.c_double, f64p]
    lib.oracle_firdes_lowpass.restype = None
    lib.oracle_firdes_lowpass.argtypes = [ctypes.c_double, f64p, sz, f64p]
    lib.oracle_kaiser_lowpass_f32.restype = sz
    lib.oracle_kaiser_lowpass_f32.argtypes = [
        ctypes.c_double, ctypes.c_double, ctypes.c_double, f32p, sz]
    lib.oracle_kaiser_multirate_f32.restype = sz
    lib.oracle_kaiser_multirate_f32.argtypes = [
        sz, sz, sz, ctypes.c_double, f32p, sz]
    lib.oracle_moving_avg.restype = None
    lib.oracle_moving_avg.argtypes = [sz, ctypes.c_float, sz, f32p, szp,
                                      vp, sz, vp, sz, szp, szp]
    lib.oracle_pfb_channelizer.restype = sz
    lib.oracle_pfb_channelizer.argtypes = [sz, sz, f32p, sz, vp, sz, vp, sz]
    lib.oracle_wlan_moving_sum.restype = sz
    lib.oracle_wlan_moving_sum.argtypes = [sz, ctypes.c_int, f32p, sz,
                                           f32p, sz]
    lib.oracle_cmul_conj.restype = sz
    lib.oracle_cmul_conj.argtypes = [vp, sz, vp, sz, vp, sz]
    lib.oracle_chain_cf32.restype = sz
    lib.oracle_chain_cf32.argtypes = [f32p, sz, f32p, sz, sz, sz,
                                      vp, sz, vp, sz, ctypes.c_int]
    lib.oracle_chain_cf32_fast.restype = sz
    lib.oracle_chain_cf32_fast.argtypes = [f32p, sz, f32p, sz, sz, sz,
                                           vp, sz, vp, sz, ctypes.c_int]
    _lib = lib
    return lib


def _f32p(a):
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_float))


def _c(a):
    return ctypes.c_void_p(a.ctypes.data)


def fir_f32(taps, inp, n_out):
    lib = _load()
    taps = np.ascontiguousarray(taps, np.float32)
    inp = np.ascontiguousarray(inp, np.float32)
    out = np.zeros(n_out, np.float32)
    r = lib.oracle_fir_f32(_f32p(taps), taps.size, _c(inp), inp.size,
                           _c(out), out.size)
    return out[: r.produced], r.consumed, r.produced, r.status


def fir_cf32(taps, inp, n_out):
    lib = _load()
    taps = np.ascontiguousarray(taps, np.float32)
    inp = np.ascontiguousarray(inp, CF32)
    out = np.zeros(n_out, CF32)
    r = lib.oracle_fir_cf32(_f32p(taps), taps.size, _c(inp), inp.size,
                            _c(out), out.size)
    return out[: r.produced], r.consumed, r.produced, r.status


def fir_ccf32(taps, inp, n_out):
    lib = _load()
    taps = np.ascontiguousarray(taps, CF32)
    inp = np.ascontiguousarray(inp, CF32)
    out = np.zeros(n_out, CF32)
    r = lib.oracle_fir_ccf32(_c(taps), taps.size, _c(inp), inp.size,
                             _c(out), out.size)
    return out[: r.produced], r.consumed, r.produced, r.status


def decim_fir_f32(decim, taps, inp, n_out):
    lib = _load()
    taps = np.ascontiguousarray(taps, np.float32)
    inp = np.ascontiguousarray(inp, np.float32)
    out = np.zeros(n_out, np.float32)
    r = lib.oracle_decim_fir_f32(decim, _f32p(taps), taps.size, _c(inp),
                                 inp.size, _c(out), out.size)
    return out[: r.produced], r.consumed, r.produced, r.status


def decim_fir_cf32(decim, taps, inp, n_out):
    lib = _load()
    taps = np.ascontiguousarray(taps, np.float32)
    inp = np.ascontiguousarray(inp, CF32)
    out = np.zeros(n_out, CF32)
    r = lib.oracle_decim_fir_cf32(decim, _f32p(taps), taps.size, _c(inp),
                                  inp.size, _c(out), out.size)
    return out[: r.produced], r.consumed, r.produced, r.status


def decim_fir_ccf32(decim, taps, inp, n_out):
    lib = _load()
    taps = np.ascontiguousarray(taps, CF32)
    inp = np.ascontiguousarray(inp, CF32)
    out = np.zeros(n_out, CF32)
    r = lib.oracle_decim_fir_ccf32(decim, _c(taps), taps.size, _c(inp),
                                   inp.size, _c(out), out.size)
    return out[: r.produced], r.consumed, r.produced, r.status


def resamp_f32(interp, decim, taps, inp, n_out):
    lib = _load()
    taps = np.ascontiguousarray(taps, np.float32)
    inp = np.ascontiguousarray(inp, np.float32)
    out = np.zeros(n_out, np.float32)
    r = lib.oracle_resamp_f32(interp, decim, _f32p(taps), taps.size,
                              _c(inp), inp.size, _c(out), out.size)
    return out[: r.produced], r.consumed, r.produced, r.status


def resamp_cf32(interp, decim, taps, inp, n_out):
    lib = _load()
    taps = np.ascontiguousarray(taps, np.float32)
    inp = np.ascontiguousarray(inp, CF32)
    out = np.zeros(n_out, CF32)
    r = lib.oracle_resamp_cf32(interp, decim, _f32p(taps), taps.size,
                               _c(inp), inp.size, _c(out), out.size)
    return out[: r.produced], r.consumed, r.produced, r.status


def dft_cf32(inp, inverse=False):
    lib = _load()
    inp = np.ascontiguousarray(inp, CF32)
    out = np.zeros(inp.size, CF32)
    lib.oracle_dft_cf32(inp.size, int(inverse), _c(inp), _c(out))
    return out


def fft_block(length, inp, n_out, inverse=False, fft_shift=False,
              normalize=None):
    lib = _load()
    inp = np.ascontiguousarray(inp, CF32)
    out = np.zeros(n_out, CF32)
    np_ = None
    if normalize is not None:
        np_ = ctypes.pointer(ctypes.c_float(normalize))
    m = lib.oracle_fft_block(length, int(inverse), int(fft_shift), np_,
                             _c(inp), inp.size, _c(out), out.size)
    return out[:m], m


def mag2(inp, n_out=None):
    lib = _load()
    inp = np.ascontiguousarray(inp, CF32)
    n_out = inp.size if n_out is None else n_out
    out = np.zeros(n_out, np.float32)
    m = lib.oracle_mag2(_c(inp), inp.size, _c(out), out.size)
    return out[:m]


def cmul(a, b, n_out=None):
    lib = _load()
    a = np.ascontiguousarray(a, CF32)
    b = np.ascontiguousarray(b, CF32)
    n_out = min(a.size, b.size) if n_out is None else n_out
    out = np.zeros(n_out, CF32)
    m = lib.oracle_cmul(_c(a), a.size, _c(b), b.size, _c(out), out.size)
    return out[:m]


def rotator(phase_incr_angle, inp, phase=None):
    lib = _load()
    inp = np.ascontiguousarray(inp, CF32)
    out = np.zeros(inp.size, CF32)
    st = np.array([1.0 + 0.0j] if phase is None else [phase], CF32)
    lib.oracle_rotator(phase_incr_angle, _c(st), _c(inp), inp.size,
                       _c(out), out.size)
    return out, complex(st[0])


def besseli0(x):
    return _load().oracle_besseli0(float(x))


def kaiser_window(length, beta):
    lib = _load()
    out = np.zeros(length, np.float64)
    lib.oracle_kaiser_window(length, beta,
                             out.ctypes.data_as(ctypes.POINTER(ctypes.c_double)))
    return out


def firdes_lowpass(cutoff, window):
    lib = _load()
    window = np.ascontiguousarray(window, np.float64)
    out = np.zeros(window.size, np.float64)
    lib.oracle_firdes_lowpass(
        cutoff, window.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        window.size, out.ctypes.data_as(ctypes.POINTER(ctypes.c_double)))
    return out


def kaiser_lowpass_f32(cutoff, transition_bw, max_ripple):
    lib = _load()
    n = lib.oracle_kaiser_lowpass_f32(cutoff, transition_bw, max_ripple,
                                      None, 0)
    out = np.zeros(n, np.float32)
    lib.oracle_kaiser_lowpass_f32(cutoff, transition_bw, max_ripple,
                                  _f32p(out), out.size)
    return out


def kaiser_multirate_f32(interp, decim, half_polyphase_len, max_ripple):
    lib = _load()
    n = lib.oracle_kaiser_multirate_f32(interp, decim, half_polyphase_len,
                                        max_ripple, None, 0)
    out = np.zeros(n, np.float32)
    lib.oracle_kaiser_multirate_f32(interp, decim, half_polyphase_len,
                                    max_ripple, _f32p(out), out.size)
    return out


def moving_avg(width, decay, history, inp, n_out, avg=None, i_state=0):
    """MovingAvg work() — moving_avg.rs:79-118. Returns
    (out, consumed, produced, avg_state, i_state)."""
    lib = _load()
    inp = np.ascontiguousarray(inp, np.float32)
    out = np.zeros(n_out, np.float32)
    avg = np.zeros(width, np.float32) if avg is None else         np.ascontiguousarray(avg, np.float32)
    ist = ctypes.c_size_t(i_state)
    cons = ctypes.c_size_t()
    prod = ctypes.c_size_t()
    lib.oracle_moving_avg(width, decay, history, _f32p(avg),
                          ctypes.byref(ist), _c(inp), inp.size, _c(out),
                          out.size, ctypes.byref(cons), ctypes.byref(prod))
    return out[:prod.value], cons.value, prod.value, avg, ist.value


def wlan_moving_sum(inp, length):
    """WLAN MovingAverage (sliding sum) one-shot — moving_average.rs."""
    lib = _load()
    is_c = np.iscomplexobj(inp)
    inp = np.ascontiguousarray(inp, CF32 if is_c else np.float32)
    n_out = inp.size + length - 1
    out = np.zeros(n_out, inp.dtype)
    p = lib.oracle_wlan_moving_sum(
        length, int(is_c),
        inp.view(np.float32).ctypes.data_as(
            ctypes.POINTER(ctypes.c_float)), inp.size,
        out.view(np.float32).ctypes.data_as(
            ctypes.POINTER(ctypes.c_float)), n_out)
    return out[:p]


def cmul_conj(a, b):
    lib = _load()
    a = np.ascontiguousarray(a, CF32)
    b = np.ascontiguousarray(b, CF32)
    m = min(a.size, b.size)
    out = np.zeros(m, CF32)
    mm = lib.oracle_cmul_conj(_c(a), a.size, _c(b), b.size, _c(out), m)
    return out[:mm]


def pfb_channelizer(num_channels, decim, taps, inp, out_cap_per_chan):
    """PfbChannelizer one-shot (channelizer.rs restatement). Returns
    [num_channels, produced] complex array."""
    lib = _load()
    taps = np.ascontiguousarray(taps, np.float32)
    inp = np.ascontiguousarray(inp, CF32)
    out = np.zeros(num_channels * out_cap_per_chan, CF32)
    prod = lib.oracle_pfb_channelizer(num_channels, decim, _f32p(taps),
                                      taps.size, _c(inp), inp.size,
                                      _c(out), out_cap_per_chan)
    return out.reshape(num_channels, out_cap_per_chan)[:, :prod]


def chain_cf32(taps1, taps2, decim, fft_len, inp, capture=True, nthreads=0,
               fast=False):
    """FIR(taps1) -> decim-FIR(taps2, decim) -> fft_len-pt forward DFT.
    fast=True uses the vectorized fused-tap variant (bench baseline leg
    only — reassociated sums, tolerance-equal to the strict chain)."""
    lib = _load()
    taps1 = np.ascontiguousarray(taps1, np.float32)
    taps2 = np.ascontiguousarray(taps2, np.float32)
    inp = np.ascontiguousarray(inp, CF32)
    y1 = inp.size + 1 - taps1.size
    frames = max(0, (y1 + 1 - taps2.size)) // decim // fft_len
    out = np.zeros(frames * fft_len, CF32) if capture else None
    fn = lib.oracle_chain_cf32_fast if fast else lib.oracle_chain_cf32
    consumed = fn(
        _f32p(taps1), taps1.size, _f32p(taps2), taps2.size, decim, fft_len,
        _c(inp), inp.size,
        _c(out) if capture else None, out.size if capture else 0, nthreads)
    return out, consumed
