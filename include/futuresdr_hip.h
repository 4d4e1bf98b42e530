/* futuresdr_hip.h — C-ABI of the MI355X-native FutureSDR streaming-DSP hot
 * path. This is the drop-in boundary: plain pointers and
 * sizes, no framework types. A FutureSDR maintainer binds these symbols from
 * Rust with a plain `extern "C"` block.
 *
 * Contract mirrored, per entry point:
 *  - fsdr_filter_host/_dev mirror `futuredsp::Filter::filter(&self, &[I],
 *    &mut [O]) -> (usize, usize, ComputationStatus)` —
 *    /root/reference/crates/futuredsp/src/lib.rs:48-68. Same
 *    consumed/produced/status math as the cores they replace:
 *      fir:    crates/futuredsp/src/fir.rs:52-91
 *      decim:  crates/futuredsp/src/decimating_fir.rs:53-95
 *      resamp: crates/futuredsp/src/polyphase_resampling_fir.rs:70-124
 *      fft:    src/blocks/fft.rs:160-221 (m = min(in,out) rounded to len,
 *              capped at 32*len; consumed == produced == m)
 *      mag2:   src/blocks/apply.rs:100-131 (m = min(in,out))
 *      cmul:   src/blocks/combine.rs:92-135 (m = min(in0,in1,out))
 *  - fsdr_ring_* mirror the Slab stream-buffer circulation
 *    (src/runtime/buffer/slab.rs:110-152,369-399) and the accelerator
 *    buffer pattern (src/runtime/buffer/vulkan/{h2d,d2h}.rs,
 *    src/runtime/buffer/wgpu): N pinned-host buffers cycling empty<->full
 *    with a reserved-items history prefix.
 *  - fsdr_chain_* is the fused pipeline runner (the `fsdr_chain_run` of
 *    SURVEY.md §8b).
 *
 * Failure convention: int status + thread-local error string
 * (fsdr_last_error). The Rust Filter trait has no error path (lib.rs:60-64);
 * the ABI's failures are device/allocation errors only. All compute entry
 * points fail with FSDR_ERR_NO_GPU when no HIP device is present — there is
 * NO CPU fallback in the product (the CPU restatement lives in oracle/,
 * which is test infrastructure).
 *
 * Threading: a handle may be used from one thread at a time (same as the
 * reference, where one block owns its kernel). Distinct handles are
 * independent. `stream` parameters take a hipStream_t (or NULL for the
 * default stream) so callers (e.g. torch) can pass their own streams.
 */
#ifndef FUTURESDR_HIP_H
#define FUTURESDR_HIP_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* Complex32 — layout-identical to num_complex::Complex<f32> (re, im). */
typedef struct { float re, im; } fsdr_cf32;

/* ComputationStatus — crates/futuredsp/src/lib.rs:31-45 */
typedef enum {
    FSDR_INSUFFICIENT_INPUT  = 0,
    FSDR_INSUFFICIENT_OUTPUT = 1,
    FSDR_BOTH_SUFFICIENT     = 2,
} fsdr_status;

typedef struct {
    size_t      consumed;
    size_t      produced;
    fsdr_status status;
} fsdr_filter_result;

/* Error codes (returned by every int-returning entry point). */
enum {
    FSDR_OK              = 0,
    FSDR_ERR_NO_GPU      = 1,
    FSDR_ERR_HIP         = 2,
    FSDR_ERR_INVALID     = 3,
    FSDR_ERR_UNSUPPORTED = 4,
};

/* ---- device management ---------------------------------------------- */
int         fsdr_device_count(void);
int         fsdr_set_device(int device);
int         fsdr_synchronize(void);
const char* fsdr_last_error(void);
const char* fsdr_version(void);

/* ---- Filter level ---------------------------------------------------- *
 * Opaque stateless-filter handle (GPU-backed). Replaces the futuredsp
 * cores behind the same span semantics. item types are fixed per
 * constructor; *_host paths stage through persistent device buffers
 * (PCIe-inclusive), *_dev paths take caller device pointers. */
typedef struct fsdr_filter fsdr_filter;

/* FirFilter<Complex32,Complex32,f32> — fir.rs:228-255 (f32 taps). */
fsdr_filter* fsdr_fir_cf32_create(const float* taps, size_t n_taps);
/* FirFilter<f32,f32,f32> — fir.rs:206-215. */
fsdr_filter* fsdr_fir_f32_create(const float* taps, size_t n_taps);
/* FirFilter<Complex32,Complex32,Complex32> — fir.rs:257-277 (complex
 * taps; the WLAN SyncLong correlator core, sync_long.rs:18-50). */
fsdr_filter* fsdr_fir_ccf32_create(const fsdr_cf32* taps, size_t n_taps);
/* DecimatingFirFilter<Complex32,Complex32,f32> — decimating_fir.rs. */
fsdr_filter* fsdr_decim_fir_cf32_create(size_t decimation,
                                        const float* taps, size_t n_taps);
/* PolyphaseResamplingFir<Complex32,Complex32,f32> — n_taps must be a
 * multiple of interp (polyphase_resampling_fir.rs:54-56 assert). */
fsdr_filter* fsdr_resamp_cf32_create(size_t interp, size_t decim,
                                     const float* taps, size_t n_taps);
/* Fft block — pow2 lengths in [4,4096] run the radix-4 Stockham kernel;
 * ANY other length in [2,2048] runs via Bluestein (two pow2 M>=2len-1
 * passes; the reference block is generic over rustfft plan lengths,
 * fft.rs:98-103). inverse/fft_shift flags and optional normalize factor
 * as src/blocks/fft.rs:92-121. */
fsdr_filter* fsdr_fft_cf32_create(size_t len, int inverse, int fft_shift,
                                  const float* normalize);
/* Apply |x|^2 (Complex32 -> f32) — the spectrum mag^2 map. */
fsdr_filter* fsdr_mag2_create(void);
/* XlatingFir (src/blocks/xlating_fir.rs): rotate-by-offset + filter +
 * decimate, fused: DecimatingFir with complex band-pass taps
 * (bpf[i] = e^{i*TAU*offset/fs*i}*taps[i]) and the output Rotator
 * (phase_incr = -TAU*offset*decimation/fs); rotator phase is carried
 * state. decimation must be >= 2 (xlating_fir.rs:44). */
fsdr_filter* fsdr_xlating_fir_cf32_create(const float* taps, size_t n_taps,
                                          size_t decimation, float offset,
                                          float sample_rate);
/* MovingAvg block (src/blocks/moving_avg.rs:79-118): stateful per-bin
 * EMA over width-sized f32 frames, emitting every `history` frames. */
fsdr_filter* fsdr_moving_avg_create(size_t width, float decay_factor,
                                    size_t history);

/* Filter::length() — lib.rs:65-67. */
size_t fsdr_filter_length(const fsdr_filter* f);

/* filter() over host spans. Stages H2D/D2H internally. */
int fsdr_filter_host(fsdr_filter* f, const void* in, size_t n_in,
                     void* out, size_t n_out, fsdr_filter_result* r);
/* filter() over device pointers, async on `stream` (no sync performed;
 * the result math is computed on the host and returns immediately). */
int fsdr_filter_dev(fsdr_filter* f, const void* d_in, size_t n_in,
                    void* d_out, size_t n_out, void* stream,
                    fsdr_filter_result* r);
/* Bulk batch FFT over `frames` fft_len-sized device-resident frames in
 * one launch (the 32-frame cap in fsdr_filter_dev mirrors the reference
 * work() quantum, fft.rs:56 — this is the GPU-native batch path the
 * chain uses). d_mag: optional fused |X|^2 output (nullable). */
int fsdr_fft_bulk_dev(fsdr_filter* f, const void* d_in, void* d_out,
                      void* d_mag, size_t frames, void* stream);
void fsdr_filter_destroy(fsdr_filter* f);

/* Combine (2-input zip-map), complex-multiply variant. m = produced. */
int fsdr_cmul_dev(const void* d_a, size_t n_a, const void* d_b, size_t n_b,
                  void* d_out, size_t n_out, void* stream, size_t* m);
int fsdr_cmul_host(const void* a, size_t n_a, const void* b, size_t n_b,
                   void* out, size_t n_out, size_t* m);

/* Rotator (futuredsp rotator.rs:23-49): out[i] = in[i] * phase0 *
 * e^{i*angle*(i+1)}, closed-form phase (the reference iterates — see the
 * parity note in tests/test_gpu_parity.py). Returns the final phase. */
int fsdr_rotator_dev(const void* d_in, void* d_out, size_t n,
                     float phase_incr_angle, float phase0_re,
                     float phase0_im, void* stream, float* final_re,
                     float* final_im);

/* PfbChannelizer (src/blocks/pfb/channelizer.rs, liquid-dsp scheme):
 * splits a Complex32 stream into num_channels frequency channels.
 * Supports any oversample_rate = N/i (decimation D = N/oversample,
 * channelizer.rs:100-105); num_channels must be a power of two in
 * [4,4096] (FFT kernel). Output is channel-major:
 * out[c*out_cap_per_chan + k]. run_dev = bulk from zero state;
 * stream_dev carries the round-robin window state across calls and
 * consumes in D quanta after the N*tpf prefill (pair with the ring's
 * release_consumed carry for exact arbitrary-chunk streaming). */
fsdr_filter* fsdr_pfb_channelizer_create(size_t num_channels,
                                         const float* taps, size_t n_taps,
                                         float oversample_rate);
int fsdr_pfb_channelizer_run_dev(fsdr_filter* f, const void* d_in,
                                 size_t n_in, void* d_out,
                                 size_t out_cap_per_chan, void* stream,
                                 size_t* produced_per_chan);
int fsdr_pfb_channelizer_stream_dev(fsdr_filter* f, const void* d_chunk,
                                    size_t n, void* d_out,
                                    size_t out_cap_per_chan, void* stream,
                                    size_t* produced_per_chan,
                                    size_t* consumed);

/* WLAN sync-short autocorrelation helpers (examples/wlan/src/bin/
 * rx.rs:73-96): a*conj(b) Combine and the sliding-SUM MovingAverage
 * (moving_average.rs:65-105; one-shot: len-1 zero items then sums). */
int fsdr_cmul_conj_dev(const void* d_a, size_t n_a, const void* d_b,
                       size_t n_b, void* d_out, size_t n_out, void* stream,
                       size_t* m);
/* divide_mag Combine (rx.rs:97): out = |a| / b. */
int fsdr_divide_mag_dev(const void* d_a, size_t n_a, const void* d_b,
                        size_t n_b, void* d_out, size_t n_out, void* stream,
                        size_t* m);
int fsdr_wlan_moving_sum_dev(const void* d_in, size_t n_in, void* d_out,
                             size_t n_out, size_t len, int is_complex,
                             void* stream, size_t* produced);

/* WLAN rx front end (config 5): the host-side SyncShort state machine
 * (examples/wlan/src/sync_short.rs:92-150; THRESHOLD 0.56, MIN_GAP 480,
 * MAX_SAMPLES 540*80) and SyncLong (sync_long.rs:96-185; SEARCH_WINDOW
 * 320, 64-tap LONG correlator run on the GPU, top-2 peak sync, CP
 * strip). sync_short consumes aligned spans of (delayed signal, 48-avg
 * autocorrelation, correlation metric) and emits frame samples +
 * "wifi_start" tags; sync_long consumes that tagged stream and emits
 * 128 preamble samples + 64-sample OFDM symbols per frame, ready for
 * the 64-pt Fft block (rx.rs:84-103). */
typedef struct fsdr_wlan_rx fsdr_wlan_rx;
fsdr_wlan_rx* fsdr_wlan_rx_create(void);
void fsdr_wlan_rx_destroy(fsdr_wlan_rx* rx);
size_t fsdr_wlan_sync_short_run(fsdr_wlan_rx* rx, const fsdr_cf32* sig,
                                const fsdr_cf32* abs48, const float* cor,
                                size_t n, fsdr_cf32* out, size_t out_cap,
                                size_t* tag_idx, float* tag_freq,
                                size_t tag_cap, size_t* n_tags,
                                size_t* consumed_out);
size_t fsdr_wlan_sync_long_run(fsdr_wlan_rx* rx, const fsdr_cf32* in,
                               size_t n, const size_t* tag_idx,
                               const float* tag_freq, size_t num_tags,
                               fsdr_cf32* out, size_t out_cap,
                               size_t* frame_off, float* frame_freq,
                               size_t frame_cap, size_t* num_frames);

/* MFMA-loop microbenchmark (diagnostics; tools/mfma_ubench.py): times
 * the chain kernel's inner MFMA loop on LDS staged once. */
int fsdr_mfma_ubench(int grid, int iters, double* tflops, void* stream);
/* Incremental chain ubench: real tile loop with components gated by
 * mode bits (1 staging loads, 2 LDS writes+barriers, 4 deposit+FFT);
 * reports executed-MFMA TF/s for pipe-utilization comparison. */
int fsdr_chain_ubench(int mode, double* tflops, void* stream);

/* ---- device memory helpers (for harnesses driving the _dev paths) ---- */
int fsdr_dev_alloc(void** d_ptr, size_t bytes);
int fsdr_dev_free(void* d_ptr);
int fsdr_memcpy_h2d(void* d_dst, const void* src, size_t bytes);
int fsdr_memcpy_d2h(void* dst, const void* d_src, size_t bytes);
/* Fill a device buffer with n Complex32 samples, re/im iid uniform[-1,1)
 * from a counter-based generator seeded by `seed` (deterministic,
 * device-generated — the NullSource/bench synthetic source). */
int fsdr_fill_uniform_cf32(void* d_ptr, size_t n, uint64_t seed,
                           uint64_t offset, void* stream);

/* ---- firdes (host-side tap designers; no GPU required) ---------------- *
 * Mirror crates/futuredsp/src/firdes/basic.rs + windows.rs:144 +
 * math/special_funs.rs:22-45 — the tap-generation row of SURVEY.md §8a. */
double fsdr_kaiser_beta(double max_ripple); /* basic.rs:444-452 */
void   fsdr_kaiser_window(size_t len, double beta, double* out);
/* firdes::kaiser::lowpass<f32> (basic.rs:310-321). Returns tap count;
 * fills out up to cap (query with out=NULL, cap=0). */
size_t fsdr_firdes_kaiser_lowpass_f32(double cutoff, double transition_bw,
                                      double max_ripple, float* out,
                                      size_t cap);
/* firdes::lowpass<f32> over a kaiser window of explicit length
 * (basic.rs:25-42 with windows::kaiser) — fixed-length designer used by
 * the bench (127 taps). */
int fsdr_firdes_lowpass_kaiser_n_f32(size_t n_taps, double beta,
                                     double cutoff, float* out);

/* ---- Chain level ------------------------------------------------------ *
 * Fused hot path: Fir(taps1) -> DecimatingFir(decim, taps2) -> Fft(len).
 * Mirrors BASELINE configs[2]/[3]; per-stage math identical to the three
 * filters composed (intermediates stay resident in HBM). */
typedef struct fsdr_chain fsdr_chain;
fsdr_chain* fsdr_chain_create(const float* taps1, size_t n_taps1,
                              const float* taps2, size_t n_taps2,
                              size_t decim, size_t fft_len);
/* Run over device input; writes `frames*fft_len` Complex32 spectra to
 * d_out (may be NULL with out_cap 0 -> spectra discarded into an internal
 * buffer, NullSink-style). If d_mag is non-NULL also writes f32 |X|^2 of
 * each bin (the config-4 spectrum join payload). Async on stream. */
int fsdr_chain_run_dev(fsdr_chain* c, const void* d_in, size_t n_in,
                       void* d_out, size_t out_cap,
                       void* d_mag, size_t mag_cap,
                       void* stream, size_t* consumed, size_t* produced);
void fsdr_chain_destroy(fsdr_chain* c);

/* ---- Ring (Slab-style stream buffer) ---------------------------------- *
 * N pinned-host buffers + N device mirrors circulating between an empty
 * queue (writer side) and a full queue (reader side), carrying the
 * reader's UNCONSUMED tail in front of the next buffer exactly like
 * slab.rs:369-399 (device-side D2D on the ring's copy stream), so
 * arbitrary chunk sizes stream exactly. reserved_items = carry capacity
 * (>= the consumer's worst-case leftover; for a FIR chain:
 * taps-1 + decim*fft_len + decim). Writer: acquire -> fill host slice ->
 * commit(n) (enqueues async H2D on the copy stream). Reader: acquire
 * (waits for the copies; yields the device pointer at the carry start
 * and items = carry + payload) -> ... launch kernels consuming
 * `consumed` items ... -> release_consumed(consumed, compute_stream).
 * release() = consumed everything. Single-producer single-consumer. */
typedef struct fsdr_ring fsdr_ring;
fsdr_ring* fsdr_ring_create(size_t n_buffers, size_t items_per_buffer,
                            size_t item_bytes, size_t reserved_items);
int  fsdr_ring_writer_acquire(fsdr_ring* r, void** host_ptr, size_t* items);
int  fsdr_ring_writer_commit(fsdr_ring* r, size_t items);
int  fsdr_ring_reader_acquire(fsdr_ring* r, void** dev_ptr, size_t* items);
int  fsdr_ring_reader_release_consumed(fsdr_ring* r, size_t consumed,
                                       void* stream);
int  fsdr_ring_reader_release(fsdr_ring* r);
void fsdr_ring_destroy(fsdr_ring* r);

/* D2H return ring (vulkan d2h.rs Writer::submit / host Reader + the
 * circuit's empty recirculation): device producer fills ring buffers on
 * its stream, async D2H on the ring's copy stream hands pinned host
 * spans to the consumer. Writer acquire takes the producer's stream so
 * buffer reuse is ordered after the previous D2H without host syncs. */
typedef struct fsdr_ring_d2h fsdr_ring_d2h;
fsdr_ring_d2h* fsdr_ring_d2h_create(size_t n_buffers,
                                    size_t items_per_buffer,
                                    size_t item_bytes);
int  fsdr_ring_d2h_writer_acquire(fsdr_ring_d2h* r, void** dev_ptr,
                                  size_t* items, void* stream);
int  fsdr_ring_d2h_writer_commit(fsdr_ring_d2h* r, size_t items,
                                 void* stream);
int  fsdr_ring_d2h_reader_acquire(fsdr_ring_d2h* r, void** host_ptr,
                                  size_t* items);
int  fsdr_ring_d2h_reader_release(fsdr_ring_d2h* r);
void fsdr_ring_d2h_destroy(fsdr_ring_d2h* r);

/* ---- Flowgraph driver (native C++ harness) ---------------------------- *
 * Minimal mirror of Flowgraph::add / Flowgraph::stream / Runtime::run for
 * 1-in/1-out chains (src/runtime/flowgraph.rs:227-241,364-423,
 * runtime.rs:169-215, wrapped_kernel.rs:106-229) with
 * NullSource/VectorSource/Head/NullSink/VectorSink endpoints and
 * GPU-filter blocks. Stream data stays resident in HBM between blocks.
 * add_* return a block id (>= 0); connect defaults to the single
 * output->input ports, like connect!'s defaults
 * (crates/macros/src/lib.rs:56-60). */
typedef struct fsdr_fg fsdr_fg;
fsdr_fg* fsdr_fg_create(void);
int  fsdr_fg_add_null_source_cf32(fsdr_fg* fg);
int  fsdr_fg_add_vector_source_cf32(fsdr_fg* fg, const fsdr_cf32* data,
                                    size_t n);
int  fsdr_fg_add_head(fsdr_fg* fg, unsigned long long n);
int  fsdr_fg_add_filter(fsdr_fg* fg, fsdr_filter* f);
int  fsdr_fg_add_null_sink(fsdr_fg* fg);
int  fsdr_fg_add_vector_sink(fsdr_fg* fg);
int  fsdr_fg_stream(fsdr_fg* fg, int src_block, int dst_block);
int  fsdr_fg_run(fsdr_fg* fg);
unsigned long long fsdr_fg_n_received(fsdr_fg* fg, int block);
size_t fsdr_fg_vector_sink_get(fsdr_fg* fg, int block, void* out,
                               size_t cap_bytes);
void fsdr_fg_destroy(fsdr_fg* fg);
size_t fsdr_filter_item_sizes(const fsdr_filter* f, size_t* out_bytes);

#ifdef __cplusplus
}
#endif
#endif /* FUTURESDR_HIP_H */
