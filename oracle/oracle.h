/* oracle/oracle.h — TEST INFRASTRUCTURE ONLY.
 *
 * CPU restatement of the FutureSDR `futuredsp` hot-path inner loops, used
 * exclusively as the parity checker (and as bench.py's `cpu_baseline` leg).
 * It is NOT the product: nothing under futuresdr_amd/ may import, link or
 * call anything in oracle/. Only tests/, __graft_entry__.smoke() and
 * bench.py's cpu_baseline leg may use it.
 *
 * Each function restates the cited reference routine line-for-line in
 * semantics (same iteration order, same f32 arithmetic where the reference
 * is f32). The reference (Rust) cannot be compiled in this image (no
 * cargo/rustc, no network), so parity is pinned by the reference's own
 * in-file known-answer tests, hard-coded in tests/test_oracle_kats.py.
 * The FFT is pinned against numpy golden vectors (tests/golden/) because
 * the reference's FFT math lives in the third-party rustfft 6.4 crate
 * (Cargo.toml:82) and the reference has no FFT-output KAT — see DESIGN.md
 * §(c) ("parity unpinned" at the rustfft boundary, pinned to the
 * unnormalized-DFT definition instead).
 */
#ifndef FSDR_ORACLE_H
#define FSDR_ORACLE_H

#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct { float re, im; } ocf32;

/* ComputationStatus — crates/futuredsp/src/lib.rs:31-45 */
enum {
    ORACLE_INSUFFICIENT_INPUT  = 0,
    ORACLE_INSUFFICIENT_OUTPUT = 1,
    ORACLE_BOTH_SUFFICIENT     = 2,
};

typedef struct {
    size_t consumed;
    size_t produced;
    int    status;
} oracle_result;

/* FirFilter<f32,f32,f32>::filter — crates/futuredsp/src/fir.rs:52-91,206-215 */
oracle_result oracle_fir_f32(const float* taps, size_t n_taps,
                             const float* in, size_t n_in,
                             float* out, size_t n_out);

/* FirFilter<Complex32,Complex32,f32>::filter — fir.rs:52-91,228-255 */
oracle_result oracle_fir_cf32(const float* taps, size_t n_taps,
                              const ocf32* in, size_t n_in,
                              ocf32* out, size_t n_out);

/* FirFilter<Complex32,Complex32,Complex32>::filter — fir.rs:257-277 */
oracle_result oracle_fir_ccf32(const ocf32* taps, size_t n_taps,
                               const ocf32* in, size_t n_in,
                               ocf32* out, size_t n_out);

/* DecimatingFirFilter<f32,f32,f32>::filter — decimating_fir.rs:53-95,223-237 */
oracle_result oracle_decim_fir_f32(size_t decimation,
                                   const float* taps, size_t n_taps,
                                   const float* in, size_t n_in,
                                   float* out, size_t n_out);

/* DecimatingFirFilter<Complex32,Complex32,f32>::filter — decimating_fir.rs:53-95,255-283 */
oracle_result oracle_decim_fir_cf32(size_t decimation,
                                    const float* taps, size_t n_taps,
                                    const ocf32* in, size_t n_in,
                                    ocf32* out, size_t n_out);

/* DecimatingFirFilter<Complex32,Complex32,Complex32> (complex taps, the
 * XlatingFir core — src/blocks/xlating_fir.rs:79-96,110-127) */
oracle_result oracle_decim_fir_ccf32(size_t decimation,
                                     const ocf32* taps, size_t n_taps,
                                     const ocf32* in, size_t n_in,
                                     ocf32* out, size_t n_out);

/* PolyphaseResamplingFir<f32,f32,f32>::filter — polyphase_resampling_fir.rs:70-141 */
oracle_result oracle_resamp_f32(size_t interp, size_t decim,
                                const float* taps, size_t n_taps,
                                const float* in, size_t n_in,
                                float* out, size_t n_out);

/* PolyphaseResamplingFir<Complex32,Complex32,f32>::filter — polyphase_resampling_fir.rs:70-124,143-167 */
oracle_result oracle_resamp_cf32(size_t interp, size_t decim,
                                 const float* taps, size_t n_taps,
                                 const ocf32* in, size_t n_in,
                                 ocf32* out, size_t n_out);

/* Unnormalized C32 DFT of one frame, rustfft forward convention
 * X[k] = sum_n x[n]·e^{-2πi·kn/N} (inverse: e^{+…}, still unnormalized).
 * Computed in f64 internally (reference-grade accuracy), cast to f32.
 * Any n ≥ 1. Reference call site: src/blocks/fft.rs:190-194. */
void oracle_dft_cf32(int n, int inverse, const ocf32* in, ocf32* out);

/* Fft block work() semantics — src/blocks/fft.rs:160-214:
 * m = min(n_in, n_out) rounded down to a multiple of len, capped at 32·len;
 * per-frame unnormalized (I)DFT; optional fft_shift (forward: shift output,
 * inverse: shift input); optional scale. Returns consumed = produced = m. */
size_t oracle_fft_block(size_t len, int inverse, int fft_shift,
                        const float* normalize /* NULL = none */,
                        const ocf32* in, size_t n_in,
                        ocf32* out, size_t n_out);

/* Apply |x|² map — norm_sqr per examples/spectrum/src/bin/cpu.rs:21-28,
 * block semantics src/blocks/apply.rs:100-131 (m = min(in,out)). */
size_t oracle_mag2(const ocf32* in, size_t n_in, float* out, size_t n_out);

/* Combine zip-map, complex multiply variant — src/blocks/combine.rs:92-135. */
size_t oracle_cmul(const ocf32* a, size_t n_a, const ocf32* b, size_t n_b,
                   ocf32* out, size_t n_out);

/* Rotator::rotate — crates/futuredsp/src/rotator.rs:23-49. phase is in/out
 * state (starts at 1+0i); phase_incr given as angle in radians. */
size_t oracle_rotator(float phase_incr_angle, ocf32* phase,
                      const ocf32* in, size_t n_in, ocf32* out, size_t n_out);

/* besseli0 — crates/futuredsp/src/math/special_funs.rs:22-45 */
double oracle_besseli0(double x);

/* windows::kaiser — crates/futuredsp/src/windows.rs:144-152 */
void oracle_kaiser_window(size_t len, double beta, double* out);

/* firdes::lowpass<f64> — crates/futuredsp/src/firdes/basic.rs:25-42 */
void oracle_firdes_lowpass(double cutoff, const double* window, size_t len,
                           double* out);

/* firdes::kaiser::lowpass<f32> — firdes/basic.rs:310-321,444-459.
 * Returns the number of taps; writes up to cap taps into out (out may be
 * NULL with cap 0 to query the size). */
size_t oracle_kaiser_lowpass_f32(double cutoff, double transition_bw,
                                 double max_ripple, float* out, size_t cap);

/* firdes::kaiser::multirate<f32> — firdes/basic.rs:412-442. */
size_t oracle_kaiser_multirate_f32(size_t interp, size_t decim,
                                   size_t half_polyphase_len,
                                   double max_ripple, float* out, size_t cap);

/* MovingAvg block work() — src/blocks/moving_avg.rs:79-118: per-bin EMA
 * over WIDTH-sized frames, emitting one averaged frame every
 * history_size inputs; avg[] and *i_state are carried state. Consumes
 * while input has a full frame AND output has room for one more frame
 * (the reference's while condition). */
void oracle_moving_avg(size_t width, float decay_factor, size_t history,
                       float* avg, size_t* i_state,
                       const float* in, size_t n_in,
                       float* out, size_t n_out,
                       size_t* consumed, size_t* produced);

/* PfbChannelizer — src/blocks/pfb/channelizer.rs:125-223 (liquid-dsp
 * scheme): round-robin window buffers (window_buffer.rs), per-channel
 * sub-filters from partition_filter_taps (utilities.rs:5-25, taps NOT
 * reversed in partitioning; FirFilter then applies them reversed), one
 * unnormalized inverse FFT of size num_channels per output step.
 * One-shot from zero state: prefills, then produces
 * min(out_cap_per_chan, remaining/decimation) samples per channel.
 * out is channel-major: out[c*out_cap_per_chan + k]. decimation_factor =
 * num_channels / oversample_rate (:104-106). */
size_t oracle_pfb_channelizer(size_t num_channels, size_t decimation_factor,
                              const float* taps, size_t n_taps,
                              const ocf32* in, size_t n_in,
                              ocf32* out, size_t out_cap_per_chan);

/* WLAN MovingAverage (sliding SUM with len-1 zero prologue) —
 * examples/wlan/src/moving_average.rs:65-105; KAT :117-127 (len 2,
 * [1,2] -> [0,3]). One-shot from fresh state; item = f32 (width 1) or
 * cf32 (width 2 floats). Returns produced items. */
size_t oracle_wlan_moving_sum(size_t len, int is_complex,
                              const float* in, size_t n_in_items,
                              float* out, size_t n_out_items);

/* Combine a * conj(b) — examples/wlan/src/bin/rx.rs:81. */
size_t oracle_cmul_conj(const ocf32* a, size_t n_a, const ocf32* b,
                        size_t n_b, ocf32* out, size_t n_out);

/* CPU-baseline chain: FIR(taps1) → decim-by-D(taps2) → per-frame
 * unnormalized fft_len-pt forward DFT (f32 radix-2, same work as the GPU
 * chain), OpenMP-sharded over contiguous chunks with (n_taps-1)-sample
 * halos (legal: the cores are stateless — lib.rs:48-60).
 * Returns the number of chain-input samples fully processed (trailing
 * partial frame dropped). out_spectra may be NULL (sink discards, like
 * NullSink — null_sink.rs:63-71). nthreads ≤ 0 → omp default. */
size_t oracle_chain_cf32(const float* taps1, size_t n_taps1,
                         const float* taps2, size_t n_taps2, size_t decim,
                         size_t fft_len,
                         const ocf32* in, size_t n_in,
                         ocf32* out_spectra, size_t n_out_cap,
                         int nthreads);

/* Vectorized (AVX2 FMA via omp simd) fused-tap variant of the chain,
 * used ONLY as bench.py's cpu_baseline leg — the defensible tuned-CPU
 * number. Same fused algorithm as the GPU chain (taps convolved in f64);
 * reassociated f32 sums, so tolerance-compare against the strict chain. */
size_t oracle_chain_cf32_fast(const float* taps1, size_t n_taps1,
                              const float* taps2, size_t n_taps2,
                              size_t decim, size_t fft_len,
                              const ocf32* in, size_t n_in,
                              ocf32* out_spectra, size_t n_out_cap,
                              int nthreads);

#ifdef __cplusplus
}
#endif
#endif
