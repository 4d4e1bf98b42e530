"""futuresdr_amd — MI355X-native FutureSDR streaming-DSP hot path.

Python plumbing over the product C-ABI (include/futuresdr_hip.h,
implemented in csrc/futuresdr_hip.hip). The compute path is the HIP
extension; there is NO CPU fallback here — if the shared library or a HIP
device is missing, calls raise. The CPU restatement used for parity lives
in oracle/ (test infrastructure) and is never imported by this package.
"""
import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "libfutursdr_hip.so")

CF32 = np.dtype(np.complex64)

INSUFFICIENT_INPUT = 0
INSUFFICIENT_OUTPUT = 1
BOTH_SUFFICIENT = 2

OK = 0
ERR_NO_GPU = 1
ERR_HIP = 2
ERR_INVALID = 3
ERR_UNSUPPORTED = 4


class FsdrError(RuntimeError):
    pass


class _Result(ctypes.Structure):
    _fields_ = [
        ("consumed", ctypes.c_size_t),
        ("produced", ctypes.c_size_t),
        ("status", ctypes.c_int),
    ]


def build():
    """Compile the gfx950 HIP extension in-tree (hipcc, ~5 s)."""
    subprocess.run(["make", "-s", "-C", _DIR], check=True)


_lib = None


def _load():
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(_SO):
        raise ImportError(
            f"futuresdr_amd HIP extension missing: {_SO} not built. "
            "Run futuresdr_amd.build() (or make -C futuresdr_amd). "
            "There is no CPU fallback — this is the GPU product path.")
    lib = ctypes.CDLL(_SO)
    sz = ctypes.c_size_t
    vp = ctypes.c_void_p
    f32p = ctypes.POINTER(ctypes.c_float)
    rp = ctypes.POINTER(_Result)

    lib.fsdr_last_error.restype = ctypes.c_char_p
    lib.fsdr_version.restype = ctypes.c_char_p
    lib.fsdr_device_count.restype = ctypes.c_int
    lib.fsdr_set_device.argtypes = [ctypes.c_int]
    lib.fsdr_fir_cf32_create.restype = vp
    lib.fsdr_fir_cf32_create.argtypes = [f32p, sz]
    lib.fsdr_fir_f32_create.restype = vp
    lib.fsdr_fir_f32_create.argtypes = [f32p, sz]
    lib.fsdr_fir_ccf32_create.restype = vp
    lib.fsdr_fir_ccf32_create.argtypes = [vp, sz]
    lib.fsdr_rotator_dev.restype = ctypes.c_int
    lib.fsdr_rotator_dev.argtypes = [vp, vp, sz, ctypes.c_float,
                                     ctypes.c_float, ctypes.c_float, vp,
                                     ctypes.POINTER(ctypes.c_float),
                                     ctypes.POINTER(ctypes.c_float)]
    lib.fsdr_decim_fir_cf32_create.restype = vp
    lib.fsdr_decim_fir_cf32_create.argtypes = [sz, f32p, sz]
    lib.fsdr_resamp_cf32_create.restype = vp
    lib.fsdr_resamp_cf32_create.argtypes = [sz, sz, f32p, sz]
    lib.fsdr_fft_cf32_create.restype = vp
    lib.fsdr_fft_cf32_create.argtypes = [sz, ctypes.c_int, ctypes.c_int, f32p]
    lib.fsdr_mag2_create.restype = vp
    lib.fsdr_moving_avg_create.restype = vp
    lib.fsdr_pfb_channelizer_create.restype = vp
    lib.fsdr_pfb_channelizer_create.argtypes = [sz, f32p, sz, ctypes.c_float]
    lib.fsdr_pfb_channelizer_stream_dev.restype = ctypes.c_int
    lib.fsdr_pfb_channelizer_stream_dev.argtypes = [vp, vp, sz, vp, sz,
                                                    vp, ctypes.POINTER(sz),
                                                    ctypes.POINTER(sz)]
    lib.fsdr_pfb_channelizer_run_dev.restype = ctypes.c_int
    lib.fsdr_pfb_channelizer_run_dev.argtypes = [vp, vp, sz, vp, sz, vp,
                                                 ctypes.POINTER(sz)]
    lib.fsdr_xlating_fir_cf32_create.restype = vp
    lib.fsdr_xlating_fir_cf32_create.argtypes = [f32p, sz, sz,
                                                 ctypes.c_float,
                                                 ctypes.c_float]
    lib.fsdr_moving_avg_create.argtypes = [sz, ctypes.c_float, sz]
    lib.fsdr_filter_length.restype = sz
    lib.fsdr_filter_length.argtypes = [vp]
    lib.fsdr_filter_host.restype = ctypes.c_int
    lib.fsdr_filter_host.argtypes = [vp, vp, sz, vp, sz, rp]
    lib.fsdr_filter_dev.restype = ctypes.c_int
    lib.fsdr_filter_dev.argtypes = [vp, vp, sz, vp, sz, vp, rp]
    lib.fsdr_fft_bulk_dev.restype = ctypes.c_int
    lib.fsdr_fft_bulk_dev.argtypes = [vp, vp, vp, vp, sz, vp]
    lib.fsdr_filter_destroy.argtypes = [vp]
    lib.fsdr_cmul_conj_dev.restype = ctypes.c_int
    lib.fsdr_cmul_conj_dev.argtypes = [vp, sz, vp, sz, vp, sz, vp,
                                       ctypes.POINTER(sz)]
    lib.fsdr_divide_mag_dev.restype = ctypes.c_int
    lib.fsdr_divide_mag_dev.argtypes = [vp, sz, vp, sz, vp, sz, vp,
                                        ctypes.POINTER(sz)]
    lib.fsdr_wlan_rx_create.restype = vp
    lib.fsdr_wlan_rx_create.argtypes = []
    lib.fsdr_wlan_rx_destroy.argtypes = [vp]
    lib.fsdr_wlan_sync_short_run.restype = sz
    lib.fsdr_wlan_sync_short_run.argtypes = [vp, vp, vp, vp, sz, vp, sz,
                                             vp, vp, sz,
                                             ctypes.POINTER(sz),
                                             ctypes.POINTER(sz)]
    lib.fsdr_wlan_sync_long_run.restype = sz
    lib.fsdr_wlan_sync_long_run.argtypes = [vp, vp, sz, vp, vp, sz, vp,
                                            sz, vp, vp, sz,
                                            ctypes.POINTER(sz)]
    lib.fsdr_wlan_moving_sum_dev.restype = ctypes.c_int
    lib.fsdr_wlan_moving_sum_dev.argtypes = [vp, sz, vp, sz, sz,
                                             ctypes.c_int, vp,
                                             ctypes.POINTER(sz)]
    lib.fsdr_cmul_dev.restype = ctypes.c_int
    lib.fsdr_cmul_dev.argtypes = [vp, sz, vp, sz, vp, sz, vp,
                                  ctypes.POINTER(sz)]
    lib.fsdr_cmul_host.restype = ctypes.c_int
    lib.fsdr_cmul_host.argtypes = [vp, sz, vp, sz, vp, sz,
                                   ctypes.POINTER(sz)]
    lib.fsdr_dev_alloc.restype = ctypes.c_int
    lib.fsdr_dev_alloc.argtypes = [ctypes.POINTER(vp), sz]
    lib.fsdr_dev_free.argtypes = [vp]
    lib.fsdr_memcpy_h2d.restype = ctypes.c_int
    lib.fsdr_memcpy_h2d.argtypes = [vp, vp, sz]
    lib.fsdr_memcpy_d2h.restype = ctypes.c_int
    lib.fsdr_memcpy_d2h.argtypes = [vp, vp, sz]
    lib.fsdr_fill_uniform_cf32.restype = ctypes.c_int
    lib.fsdr_fill_uniform_cf32.argtypes = [vp, sz, ctypes.c_uint64,
                                           ctypes.c_uint64, vp]
    lib.fsdr_kaiser_beta.restype = ctypes.c_double
    lib.fsdr_kaiser_beta.argtypes = [ctypes.c_double]
    lib.fsdr_kaiser_window.restype = None
    lib.fsdr_kaiser_window.argtypes = [sz, ctypes.c_double,
                                       ctypes.POINTER(ctypes.c_double)]
    lib.fsdr_firdes_kaiser_lowpass_f32.restype = sz
    lib.fsdr_firdes_kaiser_lowpass_f32.argtypes = [
        ctypes.c_double, ctypes.c_double, ctypes.c_double, f32p, sz]
    lib.fsdr_firdes_lowpass_kaiser_n_f32.restype = ctypes.c_int
    lib.fsdr_firdes_lowpass_kaiser_n_f32.argtypes = [
        sz, ctypes.c_double, ctypes.c_double, f32p]
    lib.fsdr_chain_create.restype = vp
    lib.fsdr_chain_create.argtypes = [f32p, sz, f32p, sz, sz, sz]
    lib.fsdr_chain_run_dev.restype = ctypes.c_int
    lib.fsdr_chain_run_dev.argtypes = [vp, vp, sz, vp, sz, vp, sz, vp,
                                       ctypes.POINTER(sz),
                                       ctypes.POINTER(sz)]
    lib.fsdr_chain_destroy.argtypes = [vp]
    lib.fsdr_synchronize.restype = ctypes.c_int
    lib.fsdr_fg_create.restype = vp
    lib.fsdr_fg_add_null_source_cf32.restype = ctypes.c_int
    lib.fsdr_fg_add_null_source_cf32.argtypes = [vp]
    lib.fsdr_fg_add_vector_source_cf32.restype = ctypes.c_int
    lib.fsdr_fg_add_vector_source_cf32.argtypes = [vp, vp, sz]
    lib.fsdr_fg_add_head.restype = ctypes.c_int
    lib.fsdr_fg_add_head.argtypes = [vp, ctypes.c_uint64]
    lib.fsdr_fg_add_filter.restype = ctypes.c_int
    lib.fsdr_fg_add_filter.argtypes = [vp, vp]
    lib.fsdr_fg_add_null_sink.restype = ctypes.c_int
    lib.fsdr_fg_add_null_sink.argtypes = [vp]
    lib.fsdr_fg_add_vector_sink.restype = ctypes.c_int
    lib.fsdr_fg_add_vector_sink.argtypes = [vp]
    lib.fsdr_fg_stream.restype = ctypes.c_int
    lib.fsdr_fg_stream.argtypes = [vp, ctypes.c_int, ctypes.c_int]
    lib.fsdr_fg_run.restype = ctypes.c_int
    lib.fsdr_fg_run.argtypes = [vp]
    lib.fsdr_fg_n_received.restype = ctypes.c_uint64
    lib.fsdr_fg_n_received.argtypes = [vp, ctypes.c_int]
    lib.fsdr_fg_vector_sink_get.restype = sz
    lib.fsdr_fg_vector_sink_get.argtypes = [vp, ctypes.c_int, vp, sz]
    lib.fsdr_fg_destroy.argtypes = [vp]
    lib.fsdr_filter_item_sizes.restype = sz
    lib.fsdr_filter_item_sizes.argtypes = [vp, ctypes.POINTER(sz)]
    lib.fsdr_ring_create.restype = vp
    lib.fsdr_ring_create.argtypes = [sz, sz, sz, sz]
    lib.fsdr_ring_writer_acquire.restype = ctypes.c_int
    lib.fsdr_ring_writer_acquire.argtypes = [vp, ctypes.POINTER(vp),
                                             ctypes.POINTER(sz)]
    lib.fsdr_ring_writer_commit.restype = ctypes.c_int
    lib.fsdr_ring_writer_commit.argtypes = [vp, sz]
    lib.fsdr_ring_reader_acquire.restype = ctypes.c_int
    lib.fsdr_ring_reader_acquire.argtypes = [vp, ctypes.POINTER(vp),
                                             ctypes.POINTER(sz)]
    lib.fsdr_ring_reader_release.restype = ctypes.c_int
    lib.fsdr_ring_reader_release.argtypes = [vp]
    lib.fsdr_ring_reader_release_consumed.restype = ctypes.c_int
    lib.fsdr_ring_reader_release_consumed.argtypes = [vp, sz, vp]
    lib.fsdr_ring_destroy.argtypes = [vp]
    lib.fsdr_ring_d2h_create.restype = vp
    lib.fsdr_ring_d2h_create.argtypes = [sz, sz, sz]
    lib.fsdr_ring_d2h_writer_acquire.restype = ctypes.c_int
    lib.fsdr_ring_d2h_writer_acquire.argtypes = [vp, ctypes.POINTER(vp),
                                                 ctypes.POINTER(sz), vp]
    lib.fsdr_ring_d2h_writer_commit.restype = ctypes.c_int
    lib.fsdr_ring_d2h_writer_commit.argtypes = [vp, sz, vp]
    lib.fsdr_ring_d2h_reader_acquire.restype = ctypes.c_int
    lib.fsdr_ring_d2h_reader_acquire.argtypes = [vp, ctypes.POINTER(vp),
                                                 ctypes.POINTER(sz)]
    lib.fsdr_ring_d2h_reader_release.restype = ctypes.c_int
    lib.fsdr_ring_d2h_reader_release.argtypes = [vp]
    lib.fsdr_ring_d2h_destroy.argtypes = [vp]
    _lib = lib
    return lib


def lib():
    return _load()


def exported_symbols():
    """Every symbol include/futuresdr_hip.h declares (for the CPU test)."""
    hdr = os.path.join(_DIR, "..", "include", "futuresdr_hip.h")
    import re
    syms = []
    with open(hdr) as f:
        for m in re.finditer(r"\b(fsdr_\w+)\s*\(", f.read()):
            syms.append(m.group(1))
    return sorted(set(syms))


def _check(rc):
    if rc != OK:
        raise FsdrError(f"fsdr error {rc}: "
                        f"{_load().fsdr_last_error().decode()}")


def _f32(a):
    a = np.ascontiguousarray(a, np.float32)
    return a, a.ctypes.data_as(ctypes.POINTER(ctypes.c_float))


def _c(a):
    return ctypes.c_void_p(a.ctypes.data)


def version():
    return _load().fsdr_version().decode()


def device_count():
    return _load().fsdr_device_count()


def set_device(i):
    _check(_load().fsdr_set_device(i))


def synchronize():
    _check(_load().fsdr_synchronize())


class Filter:
    """GPU filter handle mirroring futuredsp::Filter (lib.rs:48-68)."""

    ITEM_IN = CF32
    ITEM_OUT = CF32

    def __init__(self, handle):
        if not handle:
            raise FsdrError("filter create failed: "
                            f"{_load().fsdr_last_error().decode()}")
        self._h = handle

    @property
    def length(self):
        return _load().fsdr_filter_length(self._h)

    def __del__(self):
        if getattr(self, "_h", None):
            _load().fsdr_filter_destroy(self._h)
            self._h = None

    def filter(self, inp, n_out):
        """Host-span path: numpy in -> (out, consumed, produced, status)."""
        lib = _load()
        inp = np.ascontiguousarray(inp, self.ITEM_IN)
        out = np.zeros(n_out, self.ITEM_OUT)
        r = _Result()
        _check(lib.fsdr_filter_host(
            self._h, ctypes.c_void_p(inp.ctypes.data), inp.size,
            ctypes.c_void_p(out.ctypes.data), out.size, ctypes.byref(r)))
        return out[: r.produced], r.consumed, r.produced, r.status

    def filter_dev(self, d_in, n_in, d_out, n_out, stream=None):
        """Device-pointer path (ints = device addresses). Async."""
        lib = _load()
        r = _Result()
        _check(lib.fsdr_filter_dev(self._h, ctypes.c_void_p(d_in), n_in,
                                   ctypes.c_void_p(d_out), n_out,
                                   ctypes.c_void_p(stream or 0),
                                   ctypes.byref(r)))
        return r.consumed, r.produced, r.status


class Fir(Filter):
    """Fir block core — src/blocks/fir.rs + futuredsp fir.rs (cf32 x f32)."""

    def __init__(self, taps):
        self._taps_keep, p = _f32(taps)
        super().__init__(_load().fsdr_fir_cf32_create(p,
                                                      self._taps_keep.size))


class FirCC(Filter):
    """Complex-taps FIR — fir.rs:257-277 (WLAN correlator core)."""

    def __init__(self, taps):
        self._taps_keep = np.ascontiguousarray(taps, CF32)
        super().__init__(_load().fsdr_fir_ccf32_create(
            ctypes.c_void_p(self._taps_keep.ctypes.data),
            self._taps_keep.size))


def rotator_host(inp, phase_incr_angle, phase0=1.0 + 0.0j):
    """Rotator over a host span (stages via dev helpers)."""
    lib = _load()
    inp = np.ascontiguousarray(inp, CF32)
    n = inp.size
    d_in = ctypes.c_void_p()
    d_out = ctypes.c_void_p()
    _check(lib.fsdr_dev_alloc(ctypes.byref(d_in), max(n, 1) * 8))
    _check(lib.fsdr_dev_alloc(ctypes.byref(d_out), max(n, 1) * 8))
    try:
        _check(lib.fsdr_memcpy_h2d(d_in, ctypes.c_void_p(inp.ctypes.data),
                                   n * 8))
        fr = ctypes.c_float()
        fi = ctypes.c_float()
        _check(lib.fsdr_rotator_dev(d_in, d_out, n, phase_incr_angle,
                                    phase0.real, phase0.imag, None,
                                    ctypes.byref(fr), ctypes.byref(fi)))
        _check(lib.fsdr_synchronize())
        out = np.zeros(n, CF32)
        _check(lib.fsdr_memcpy_d2h(ctypes.c_void_p(out.ctypes.data), d_out,
                                   n * 8))
        return out, complex(fr.value, fi.value)
    finally:
        lib.fsdr_dev_free(d_in)
        lib.fsdr_dev_free(d_out)


class FirF32(Filter):
    ITEM_IN = np.dtype(np.float32)
    ITEM_OUT = np.dtype(np.float32)

    def __init__(self, taps):
        self._taps_keep, p = _f32(taps)
        super().__init__(_load().fsdr_fir_f32_create(p, self._taps_keep.size))


class DecimFir(Filter):
    def __init__(self, decimation, taps):
        self._taps_keep, p = _f32(taps)
        super().__init__(_load().fsdr_decim_fir_cf32_create(
            decimation, p, self._taps_keep.size))


class Resampler(Filter):
    def __init__(self, interp, decim, taps):
        self._taps_keep, p = _f32(taps)
        super().__init__(_load().fsdr_resamp_cf32_create(
            interp, decim, p, self._taps_keep.size))


class Fft(Filter):
    def __init__(self, length, inverse=False, fft_shift=False,
                 normalize=None):
        np_ = None
        if normalize is not None:
            np_ = ctypes.pointer(ctypes.c_float(normalize))
        super().__init__(_load().fsdr_fft_cf32_create(
            length, int(inverse), int(fft_shift), np_))

    def bulk_dev(self, d_in, d_out, frames, d_mag=0, stream=None):
        """One launch over `frames` device-resident frames (the batch
        path; fsdr_filter_dev keeps the reference's 32-frame quantum)."""
        _check(_load().fsdr_fft_bulk_dev(
            self._h, ctypes.c_void_p(d_in), ctypes.c_void_p(d_out),
            ctypes.c_void_p(d_mag), frames, ctypes.c_void_p(stream or 0)))


class Mag2(Filter):
    ITEM_OUT = np.dtype(np.float32)

    def __init__(self):
        super().__init__(_load().fsdr_mag2_create())


class XlatingFir(Filter):
    """XlatingFir block — xlating_fir.rs (rotate + filter + decimate)."""

    def __init__(self, taps, decimation, offset, sample_rate):
        self._taps_keep, p = _f32(taps)
        super().__init__(_load().fsdr_xlating_fir_cf32_create(
            p, self._taps_keep.size, decimation, offset, sample_rate))


class PfbChannelizer(Filter):
    """PFB channelizer — pfb/channelizer.rs (any oversample N/i;
    decimation D = N/oversample). Stateful streaming via stream()."""

    def __init__(self, num_channels, taps, oversample_rate=1.0):
        self._taps_keep, p = _f32(taps)
        self.n = num_channels
        self.decim = int(num_channels / oversample_rate)
        super().__init__(_load().fsdr_pfb_channelizer_create(
            num_channels, p, self._taps_keep.size, oversample_rate))

    def stream(self, chunk):
        """Feed one host chunk through the stateful streaming path;
        returns ([num_channels, produced] array, consumed). Unconsumed
        samples must be re-presented by the caller (slab semantics)."""
        lib = _load()
        chunk = np.ascontiguousarray(chunk, CF32)
        cap = max(1, (chunk.size + self.n * self._tpf()) // self.decim)
        d_in = ctypes.c_void_p()
        d_out = ctypes.c_void_p()
        _check(lib.fsdr_dev_alloc(ctypes.byref(d_in), chunk.size * 8))
        _check(lib.fsdr_dev_alloc(ctypes.byref(d_out), self.n * cap * 8))
        try:
            _check(lib.fsdr_memcpy_h2d(
                d_in, ctypes.c_void_p(chunk.ctypes.data), chunk.size * 8))
            prod = ctypes.c_size_t()
            cons = ctypes.c_size_t()
            _check(lib.fsdr_pfb_channelizer_stream_dev(
                self._h, d_in, chunk.size, d_out, cap, None,
                ctypes.byref(prod), ctypes.byref(cons)))
            _check(lib.fsdr_synchronize())
            out = np.zeros(self.n * cap, CF32)
            _check(lib.fsdr_memcpy_d2h(ctypes.c_void_p(out.ctypes.data),
                                       d_out, self.n * cap * 8))
            return (out.reshape(self.n, cap)[:, :prod.value], cons.value)
        finally:
            lib.fsdr_dev_free(d_in)
            lib.fsdr_dev_free(d_out)

    def _tpf(self):
        return -(-self._taps_keep.size // self.n)

    def run(self, inp):
        """Bulk one-shot from zero state over a host span; returns an
        array of shape [num_channels, produced]."""
        lib = _load()
        inp = np.ascontiguousarray(inp, CF32)
        tpf = self._tpf()
        cap = max(1, (inp.size - self.n * tpf) // self.decim)
        d_in = ctypes.c_void_p()
        d_out = ctypes.c_void_p()
        _check(lib.fsdr_dev_alloc(ctypes.byref(d_in), inp.size * 8))
        _check(lib.fsdr_dev_alloc(ctypes.byref(d_out), self.n * cap * 8))
        try:
            _check(lib.fsdr_memcpy_h2d(d_in,
                                       ctypes.c_void_p(inp.ctypes.data),
                                       inp.size * 8))
            prod = ctypes.c_size_t()
            _check(lib.fsdr_pfb_channelizer_run_dev(
                self._h, d_in, inp.size, d_out, cap, None,
                ctypes.byref(prod)))
            _check(lib.fsdr_synchronize())
            out = np.zeros(self.n * cap, CF32)
            _check(lib.fsdr_memcpy_d2h(ctypes.c_void_p(out.ctypes.data),
                                       d_out, self.n * cap * 8))
            return out.reshape(self.n, cap)[:, :prod.value]
        finally:
            lib.fsdr_dev_free(d_in)
            lib.fsdr_dev_free(d_out)


class MovingAvg(Filter):
    """MovingAvg block — moving_avg.rs:79-118 (stateful per-bin EMA)."""

    ITEM_IN = np.dtype(np.float32)
    ITEM_OUT = np.dtype(np.float32)

    def __init__(self, width, decay_factor, history):
        super().__init__(_load().fsdr_moving_avg_create(
            width, decay_factor, history))


class Chain:
    """Fused Fir -> DecimFir -> Fft pipeline (the bench hot path)."""

    def __init__(self, taps1, taps2, decim, fft_len):
        lib = _load()
        self._t1, p1 = _f32(taps1)
        self._t2, p2 = _f32(taps2)
        self.decim, self.fft_len = decim, fft_len
        self._h = lib.fsdr_chain_create(p1, self._t1.size, p2, self._t2.size,
                                        decim, fft_len)
        if not self._h:
            raise FsdrError("chain create failed: "
                            f"{lib.fsdr_last_error().decode()}")

    def __del__(self):
        if getattr(self, "_h", None):
            _load().fsdr_chain_destroy(self._h)
            self._h = None

    def run_dev(self, d_in, n_in, d_out=0, out_cap=0, d_mag=0, mag_cap=0,
                stream=None):
        lib = _load()
        cons = ctypes.c_size_t()
        prod = ctypes.c_size_t()
        _check(lib.fsdr_chain_run_dev(
            self._h, ctypes.c_void_p(d_in), n_in, ctypes.c_void_p(d_out),
            out_cap, ctypes.c_void_p(d_mag), mag_cap,
            ctypes.c_void_p(stream or 0), ctypes.byref(cons),
            ctypes.byref(prod)))
        return cons.value, prod.value


class Flowgraph:
    """Native-driver flowgraph: mirrors Flowgraph::add + connect! + run
    (flowgraph.rs, runtime.rs) for 1-in/1-out chains on the GPU."""

    def __init__(self):
        self._h = _load().fsdr_fg_create()
        if not self._h:
            raise FsdrError("flowgraph create failed (no HIP device?)")
        self._keep = []  # keep filter objects alive

    def __del__(self):
        if getattr(self, "_h", None):
            _load().fsdr_fg_destroy(self._h)
            self._h = None

    def null_source(self):
        return _load().fsdr_fg_add_null_source_cf32(self._h)

    def vector_source(self, data):
        data = np.ascontiguousarray(data, CF32)
        self._keep.append(data)
        return _load().fsdr_fg_add_vector_source_cf32(
            self._h, ctypes.c_void_p(data.ctypes.data), data.size)

    def head(self, n):
        return _load().fsdr_fg_add_head(self._h, n)

    def filter(self, f):
        self._keep.append(f)
        return _load().fsdr_fg_add_filter(self._h, f._h)

    def null_sink(self):
        return _load().fsdr_fg_add_null_sink(self._h)

    def vector_sink(self):
        return _load().fsdr_fg_add_vector_sink(self._h)

    def stream(self, src, dst):
        _check(_load().fsdr_fg_stream(self._h, src, dst))

    def connect(self, *blocks):
        for a, b in zip(blocks, blocks[1:]):
            self.stream(a, b)

    def run(self):
        _check(_load().fsdr_fg_run(self._h))

    def n_received(self, block):
        return _load().fsdr_fg_n_received(self._h, block)

    def sink_data(self, block, dtype=CF32):
        lib = _load()
        nbytes = lib.fsdr_fg_vector_sink_get(self._h, block, None, 0)
        out = np.zeros(nbytes // np.dtype(dtype).itemsize, dtype)
        lib.fsdr_fg_vector_sink_get(self._h, block,
                                    ctypes.c_void_p(out.ctypes.data), nbytes)
        return out


def kaiser_lowpass(cutoff, transition_bw, max_ripple):
    """firdes::kaiser::lowpass<f32> — product host-side designer."""
    lib = _load()
    n = lib.fsdr_firdes_kaiser_lowpass_f32(cutoff, transition_bw,
                                           max_ripple, None, 0)
    out = np.zeros(n, np.float32)
    lib.fsdr_firdes_kaiser_lowpass_f32(
        cutoff, transition_bw, max_ripple,
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_float)), out.size)
    return out


def lowpass_kaiser_n(n_taps, beta, cutoff):
    """firdes::lowpass over a kaiser(n_taps, beta) window (basic.rs:25-42)."""
    lib = _load()
    out = np.zeros(n_taps, np.float32)
    _check(lib.fsdr_firdes_lowpass_kaiser_n_f32(
        n_taps, beta, cutoff,
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_float))))
    return out


def kaiser_beta(max_ripple):
    return _load().fsdr_kaiser_beta(max_ripple)


def _dev_roundtrip(fn, arrays_in, out_dtype, out_items, *extra):
    """Helper: upload arrays, call fn(d_ptrs..., d_out, ...), download."""
    lib = _load()
    ptrs = []
    try:
        for a in arrays_in:
            p = ctypes.c_void_p()
            _check(lib.fsdr_dev_alloc(ctypes.byref(p), max(1, a.nbytes)))
            _check(lib.fsdr_memcpy_h2d(p, ctypes.c_void_p(a.ctypes.data),
                                       a.nbytes))
            ptrs.append(p)
        out = np.zeros(out_items, out_dtype)
        d_out = ctypes.c_void_p()
        _check(lib.fsdr_dev_alloc(ctypes.byref(d_out), max(1, out.nbytes)))
        ptrs.append(d_out)
        produced = fn(ptrs[:-1], d_out, *extra)
        _check(lib.fsdr_synchronize())
        _check(lib.fsdr_memcpy_d2h(ctypes.c_void_p(out.ctypes.data), d_out,
                                   out.nbytes))
        return out, produced
    finally:
        for p in ptrs:
            lib.fsdr_dev_free(p)


def divide_mag_host(a, b):
    lib = _load()
    a = np.ascontiguousarray(a, CF32)
    b = np.ascontiguousarray(b, np.float32)
    n = min(a.size, b.size)

    def call(d_ins, d_out):
        m = ctypes.c_size_t()
        _check(lib.fsdr_divide_mag_dev(d_ins[0], a.size, d_ins[1], b.size,
                                       d_out, n, None, ctypes.byref(m)))
        return m.value

    out, m = _dev_roundtrip(call, [a, b], np.float32, n)
    return out[:m]


def cmul_conj_host(a, b):
    lib = _load()
    a = np.ascontiguousarray(a, CF32)
    b = np.ascontiguousarray(b, CF32)
    n = min(a.size, b.size)

    def call(d_ins, d_out):
        m = ctypes.c_size_t()
        _check(lib.fsdr_cmul_conj_dev(d_ins[0], a.size, d_ins[1], b.size,
                                      d_out, n, None, ctypes.byref(m)))
        return m.value

    out, m = _dev_roundtrip(call, [a, b], CF32, n)
    return out[:m]


def wlan_moving_sum_host(inp, length):
    lib = _load()
    is_c = np.iscomplexobj(inp)
    inp = np.ascontiguousarray(inp, CF32 if is_c else np.float32)
    n_out = inp.size + length - 1  # pad + sums (fresh state)

    def call(d_ins, d_out):
        p = ctypes.c_size_t()
        _check(lib.fsdr_wlan_moving_sum_dev(d_ins[0], inp.size, d_out,
                                            n_out, length, int(is_c), None,
                                            ctypes.byref(p)))
        return p.value

    out, p = _dev_roundtrip(call, [inp], inp.dtype, n_out)
    return out[:p]


class WlanRx:
    """WLAN rx front end host state machines (config 5): SyncShort
    (sync_short.rs:92-150) + SyncLong (sync_long.rs:96-185, GPU 64-tap
    correlator). Stateful like the reference blocks."""

    def __init__(self):
        self._h = _load().fsdr_wlan_rx_create()
        if not self._h:
            raise FsdrError("wlan rx create failed: "
                            + _load().fsdr_last_error().decode())

    def __del__(self):
        if getattr(self, "_h", None):
            _load().fsdr_wlan_rx_destroy(self._h)
            self._h = None

    def sync_short(self, sig, abs48, cor, max_tags=64):
        """Returns (frame_samples, tags) where tags is a list of
        (output_index, coarse_freq_offset)."""
        lib = _load()
        sig = np.ascontiguousarray(sig, CF32)
        abs48 = np.ascontiguousarray(abs48, CF32)
        cor = np.ascontiguousarray(cor, np.float32)
        n = min(sig.size, abs48.size, cor.size)
        out = np.zeros(n, CF32)
        tag_idx = np.zeros(max_tags, np.uintp)
        tag_freq = np.zeros(max_tags, np.float32)
        n_tags = ctypes.c_size_t()
        consumed = ctypes.c_size_t()
        prod = lib.fsdr_wlan_sync_short_run(
            self._h, _c(sig), _c(abs48), _c(cor), n, _c(out), out.size,
            _c(tag_idx), _c(tag_freq), max_tags, ctypes.byref(n_tags),
            ctypes.byref(consumed))
        tags = [(int(tag_idx[i]), float(tag_freq[i]))
                for i in range(n_tags.value)]
        return out[:prod], tags

    def sync_long(self, frame_samples, tags, max_frames=16):
        """Returns (symbol_stream, frames) where frames is a list of
        (correlator_offset, fine_freq_offset)."""
        lib = _load()
        x = np.ascontiguousarray(frame_samples, CF32)
        tag_idx = np.array([t[0] for t in tags], np.uintp)
        tag_freq = np.array([t[1] for t in tags], np.float32)
        out = np.zeros(x.size + 128 * max(1, len(tags)), CF32)
        f_off = np.zeros(max_frames, np.uintp)
        f_freq = np.zeros(max_frames, np.float32)
        nf = ctypes.c_size_t()
        prod = lib.fsdr_wlan_sync_long_run(
            self._h, _c(x), x.size, _c(tag_idx), _c(tag_freq),
            len(tags), _c(out), out.size, _c(f_off), _c(f_freq),
            max_frames, ctypes.byref(nf))
        frames = [(int(f_off[i]), float(f_freq[i]))
                  for i in range(nf.value)]
        return out[:prod], frames


def cmul_host(a, b):
    lib = _load()
    a = np.ascontiguousarray(a, CF32)
    b = np.ascontiguousarray(b, CF32)
    m = min(a.size, b.size)
    out = np.zeros(m, CF32)
    mm = ctypes.c_size_t()
    _check(lib.fsdr_cmul_host(ctypes.c_void_p(a.ctypes.data), a.size,
                              ctypes.c_void_p(b.ctypes.data), b.size,
                              ctypes.c_void_p(out.ctypes.data), out.size,
                              ctypes.byref(mm)))
    return out[: mm.value]


def fill_uniform_dev(d_ptr, n, seed=0, offset=0, stream=None):
    _check(_load().fsdr_fill_uniform_cf32(ctypes.c_void_p(d_ptr), n, seed,
                                          offset,
                                          ctypes.c_void_p(stream or 0)))
