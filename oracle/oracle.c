/* oracle/oracle.c — TEST INFRASTRUCTURE ONLY (see oracle.h header).
 *
 * CPU restatement of FutureSDR's futuredsp hot-path cores. Iteration order
 * and f32 arithmetic match the reference's stable-Rust implementations
 * (plain mul/add in source order), so on identical inputs this is
 * bit-identical to the reference's non-nightly build; the nightly
 * `algebraic_*` variants (fir.rs:93-200) permit reassociation, which is why
 * product-vs-oracle parity is tolerance-based.
 */
#include "oracle.h"

#include <math.h>
#include <stdlib.h>
#include <string.h>
#ifdef _OPENMP
#include <omp.h>
#endif

/* ---------- status math ---------------------------------------------- */

static size_t sat_sub(size_t a, size_t b) { return a > b ? a - b : 0; }

/* fir_kernel_core status selection — crates/futuredsp/src/fir.rs:69-74 */
static oracle_result fir_status(size_t n_in, size_t n_taps, size_t n_out) {
    size_t producable = sat_sub(n_in + 1, n_taps);
    oracle_result r;
    if (producable > n_out) {
        r.consumed = r.produced = n_out;
        r.status = ORACLE_INSUFFICIENT_OUTPUT;
    } else if (producable == n_out) {
        r.consumed = r.produced = producable;
        r.status = ORACLE_BOTH_SUFFICIENT;
    } else {
        r.consumed = r.produced = producable;
        r.status = ORACLE_INSUFFICIENT_INPUT;
    }
    return r;
}

/* decimating fir_kernel_core status — decimating_fir.rs:71-78,94 */
static oracle_result decim_status(size_t decim, size_t n_in, size_t n_taps,
                                  size_t n_out) {
    size_t filterable = sat_sub(n_in + 1, n_taps);
    size_t consumable = filterable / decim;
    oracle_result r;
    if (consumable > n_out) {
        r.produced = n_out;
        r.status = ORACLE_INSUFFICIENT_OUTPUT;
    } else if (consumable == n_out) {
        r.produced = n_out;
        r.status = ORACLE_BOTH_SUFFICIENT;
    } else {
        r.produced = consumable;
        r.status = ORACLE_INSUFFICIENT_INPUT;
    }
    r.consumed = r.produced * decim;
    return r;
}

/* ---------- FIR ------------------------------------------------------- */

oracle_result oracle_fir_f32(const float* taps, size_t n_taps,
                             const float* in, size_t n_in,
                             float* out, size_t n_out) {
    oracle_result r = fir_status(n_in, n_taps, n_out);
    for (size_t k = 0; k < r.produced; k++) {
        float sum = 0.0f;                       /* fir.rs:78 init */
        for (size_t t = 0; t < n_taps; t++)     /* fir.rs:79-85 */
            sum = sum + in[k + t] * taps[n_taps - 1 - t];
        out[k] = sum;
    }
    return r;
}

oracle_result oracle_fir_cf32(const float* taps, size_t n_taps,
                              const ocf32* in, size_t n_in,
                              ocf32* out, size_t n_out) {
    oracle_result r = fir_status(n_in, n_taps, n_out);
    for (size_t k = 0; k < r.produced; k++) {
        float sre = 0.0f, sim = 0.0f;           /* fir.rs:242-249 MAC */
        for (size_t t = 0; t < n_taps; t++) {
            float tap = taps[n_taps - 1 - t];
            sre = sre + in[k + t].re * tap;
            sim = sim + in[k + t].im * tap;
        }
        out[k].re = sre;
        out[k].im = sim;
    }
    return r;
}

oracle_result oracle_fir_ccf32(const ocf32* taps, size_t n_taps,
                               const ocf32* in, size_t n_in,
                               ocf32* out, size_t n_out) {
    oracle_result r = fir_status(n_in, n_taps, n_out);
    for (size_t k = 0; k < r.produced; k++) {
        /* fir.rs:265-271: accum + sample * tap (num_complex mul:
         * re = a.re*b.re - a.im*b.im, im = a.re*b.im + a.im*b.re) */
        float sre = 0.0f, sim = 0.0f;
        for (size_t t = 0; t < n_taps; t++) {
            ocf32 s = in[k + t], h = taps[n_taps - 1 - t];
            sre = sre + (s.re * h.re - s.im * h.im);
            sim = sim + (s.re * h.im + s.im * h.re);
        }
        out[k].re = sre;
        out[k].im = sim;
    }
    return r;
}

/* ---------- Decimating FIR -------------------------------------------- */

oracle_result oracle_decim_fir_f32(size_t decimation,
                                   const float* taps, size_t n_taps,
                                   const float* in, size_t n_in,
                                   float* out, size_t n_out) {
    oracle_result r = decim_status(decimation, n_in, n_taps, n_out);
    for (size_t k = 0; k < r.produced; k++) {
        float sum = 0.0f;
        for (size_t t = 0; t < n_taps; t++)     /* decimating_fir.rs:81-89 */
            sum = sum + in[decimation - 1 + k * decimation + t]
                          * taps[n_taps - 1 - t];
        out[k] = sum;
    }
    return r;
}

oracle_result oracle_decim_fir_cf32(size_t decimation,
                                    const float* taps, size_t n_taps,
                                    const ocf32* in, size_t n_in,
                                    ocf32* out, size_t n_out) {
    oracle_result r = decim_status(decimation, n_in, n_taps, n_out);
    for (size_t k = 0; k < r.produced; k++) {
        float sre = 0.0f, sim = 0.0f;
        for (size_t t = 0; t < n_taps; t++) {
            ocf32 s = in[decimation - 1 + k * decimation + t];
            float tap = taps[n_taps - 1 - t];
            sre = sre + s.re * tap;
            sim = sim + s.im * tap;
        }
        out[k].re = sre;
        out[k].im = sim;
    }
    return r;
}

oracle_result oracle_decim_fir_ccf32(size_t decimation,
                                     const ocf32* taps, size_t n_taps,
                                     const ocf32* in, size_t n_in,
                                     ocf32* out, size_t n_out) {
    /* decimating_fir.rs:183-216 (complex taps, num_complex mul) */
    oracle_result r = decim_status(decimation, n_in, n_taps, n_out);
    for (size_t k = 0; k < r.produced; k++) {
        float sre = 0.0f, sim = 0.0f;
        for (size_t t = 0; t < n_taps; t++) {
            ocf32 s = in[decimation - 1 + k * decimation + t];
            ocf32 hh = taps[n_taps - 1 - t];
            sre = sre + (s.re * hh.re - s.im * hh.im);
            sim = sim + (s.re * hh.im + s.im * hh.re);
        }
        out[k].re = sre;
        out[k].im = sim;
    }
    return r;
}

/* ---------- Polyphase resampler --------------------------------------- */

/* resampling_fir_kernel_core — polyphase_resampling_fir.rs:70-124 */
static oracle_result resamp_status(size_t interp, size_t decim,
                                   size_t n_taps_total, size_t n_in,
                                   size_t n_out) {
    size_t num_taps = n_taps_total / interp;    /* :91 */
    size_t producable =
        sat_sub(sat_sub(n_in + 1, num_taps) * interp, 1) / decim; /* :92-93 */
    producable = (producable / interp) * interp;                  /* :95 */
    oracle_result r;
    if (producable > n_out) {                                     /* :96-103 */
        r.produced = (n_out / interp) * interp;
        r.status = ORACLE_INSUFFICIENT_OUTPUT;
    } else if (producable == n_out) {
        r.produced = producable;
        r.status = ORACLE_BOTH_SUFFICIENT;
    } else {
        r.produced = producable;
        r.status = ORACLE_INSUFFICIENT_INPUT;
    }
    r.consumed = (r.produced / interp) * decim;                   /* :106 */
    return r;
}

oracle_result oracle_resamp_f32(size_t interp, size_t decim,
                                const float* taps, size_t n_taps,
                                const float* in, size_t n_in,
                                float* out, size_t n_out) {
    oracle_result r = resamp_status(interp, decim, n_taps, n_in, n_out);
    size_t num_taps = n_taps / interp;
    for (size_t k = 0; k < r.produced; k++) {   /* :109-118 */
        size_t bank_idx = (k * decim) % interp;
        size_t input_idx = k * decim / interp;
        float sum = 0.0f;
        for (size_t t = 0; t < num_taps; t++) {
            size_t tap_idx = interp * (num_taps - t - 1) + bank_idx;
            sum = sum + in[input_idx + t] * taps[tap_idx];
        }
        out[k] = sum;
    }
    return r;
}

oracle_result oracle_resamp_cf32(size_t interp, size_t decim,
                                 const float* taps, size_t n_taps,
                                 const ocf32* in, size_t n_in,
                                 ocf32* out, size_t n_out) {
    oracle_result r = resamp_status(interp, decim, n_taps, n_in, n_out);
    size_t num_taps = n_taps / interp;
    for (size_t k = 0; k < r.produced; k++) {
        size_t bank_idx = (k * decim) % interp;
        size_t input_idx = k * decim / interp;
        float sre = 0.0f, sim = 0.0f;
        for (size_t t = 0; t < num_taps; t++) {
            size_t tap_idx = interp * (num_taps - t - 1) + bank_idx;
            ocf32 s = in[input_idx + t];
            float tap = taps[tap_idx];
            sre = sre + s.re * tap;             /* :158-161 */
            sim = sim + s.im * tap;
        }
        out[k].re = sre;
        out[k].im = sim;
    }
    return r;
}

/* ---------- DFT / Fft block ------------------------------------------- */

/* Iterative radix-2 DIT FFT in f64 for power-of-2 n; naive DFT otherwise.
 * Forward = unnormalized e^{-2πikn/N}, matching rustfft's convention at
 * the reference call site src/blocks/fft.rs:190-194. */
static void dft_f64(int n, int inverse, double* re, double* im) {
    if (n <= 1) return;
    if ((n & (n - 1)) == 0) {
        /* bit-reverse permutation */
        for (int i = 1, j = 0; i < n; i++) {
            int bit = n >> 1;
            for (; j & bit; bit >>= 1) j ^= bit;
            j ^= bit;
            if (i < j) {
                double tr = re[i]; re[i] = re[j]; re[j] = tr;
                double ti = im[i]; im[i] = im[j]; im[j] = ti;
            }
        }
        double sgn = inverse ? 1.0 : -1.0;
        for (int len = 2; len <= n; len <<= 1) {
            double ang = sgn * 2.0 * M_PI / (double)len;
            for (int i = 0; i < n; i += len) {
                for (int k = 0; k < len / 2; k++) {
                    double wr = cos(ang * (double)k);
                    double wi = sin(ang * (double)k);
                    int a = i + k, b = i + k + len / 2;
                    double xr = re[b] * wr - im[b] * wi;
                    double xi = re[b] * wi + im[b] * wr;
                    re[b] = re[a] - xr; im[b] = im[a] - xi;
                    re[a] = re[a] + xr; im[a] = im[a] + xi;
                }
            }
        }
    } else {
        double* tr = (double*)malloc(sizeof(double) * (size_t)n * 2);
        double* ti = tr + n;
        double sgn = inverse ? 1.0 : -1.0;
        for (int k = 0; k < n; k++) {
            double sr = 0.0, si = 0.0;
            for (int j = 0; j < n; j++) {
                double ang = sgn * 2.0 * M_PI * (double)k * (double)j / n;
                double wr = cos(ang), wi = sin(ang);
                sr += re[j] * wr - im[j] * wi;
                si += re[j] * wi + im[j] * wr;
            }
            tr[k] = sr; ti[k] = si;
        }
        memcpy(re, tr, sizeof(double) * (size_t)n);
        memcpy(im, ti, sizeof(double) * (size_t)n);
        free(tr);
    }
}

void oracle_dft_cf32(int n, int inverse, const ocf32* in, ocf32* out) {
    double* re = (double*)malloc(sizeof(double) * (size_t)n * 2);
    double* im = re + n;
    for (int i = 0; i < n; i++) { re[i] = in[i].re; im[i] = in[i].im; }
    dft_f64(n, inverse, re, im);
    for (int i = 0; i < n; i++) {
        out[i].re = (float)re[i];
        out[i].im = (float)im[i];
    }
    free(re);
}

size_t oracle_fft_block(size_t len, int inverse, int fft_shift,
                        const float* normalize,
                        const ocf32* in, size_t n_in,
                        ocf32* out, size_t n_out) {
    /* src/blocks/fft.rs:169-171: m = min(in,out) rounded to len, cap 32·len */
    size_t m = n_in < n_out ? n_in : n_out;
    m = (m / len) * len;
    size_t cap = len * 32;
    if (m > cap) m = cap;
    if (m == 0) return 0;

    ocf32* buff = (ocf32*)malloc(sizeof(ocf32) * m);
    if (inverse && fft_shift) {                 /* fft.rs:179-185 */
        for (size_t f = 0; f < m / len; f++)
            for (size_t k = 0; k < len; k++)
                buff[f * len + k] = in[f * len + (k + len / 2) % len];
    } else {
        memcpy(buff, in, sizeof(ocf32) * m);    /* fft.rs:187 */
    }
    for (size_t f = 0; f < m / len; f++)        /* fft.rs:190-194 */
        oracle_dft_cf32((int)len, inverse, buff + f * len, out + f * len);
    if (!inverse && fft_shift) {                /* fft.rs:196-204 */
        ocf32* sym = (ocf32*)malloc(sizeof(ocf32) * len);
        for (size_t f = 0; f < m / len; f++) {
            memcpy(sym, out + f * len, sizeof(ocf32) * len);
            for (size_t k = 0; k < len; k++)
                out[f * len + k] = sym[(k + len / 2) % len];
        }
        free(sym);
    }
    if (normalize) {                            /* fft.rs:206-210 */
        float fac = *normalize;
        for (size_t i = 0; i < m; i++) {
            out[i].re *= fac;
            out[i].im *= fac;
        }
    }
    free(buff);
    return m;
}

/* ---------- element-wise blocks --------------------------------------- */

size_t oracle_mag2(const ocf32* in, size_t n_in, float* out, size_t n_out) {
    size_t m = n_in < n_out ? n_in : n_out;     /* apply.rs:108 */
    for (size_t i = 0; i < m; i++)
        out[i] = in[i].re * in[i].re + in[i].im * in[i].im;
    return m;
}

size_t oracle_cmul(const ocf32* a, size_t n_a, const ocf32* b, size_t n_b,
                   ocf32* out, size_t n_out) {
    size_t m = n_a < n_b ? n_a : n_b;           /* combine.rs:115-116 */
    if (n_out < m) m = n_out;
    for (size_t i = 0; i < m; i++) {
        out[i].re = a[i].re * b[i].re - a[i].im * b[i].im;
        out[i].im = a[i].re * b[i].im + a[i].im * b[i].re;
    }
    return m;
}

size_t oracle_rotator(float phase_incr_angle, ocf32* phase,
                      const ocf32* in, size_t n_in, ocf32* out,
                      size_t n_out) {
    /* rotator.rs:15-20 from_polar; :31-49 rotate */
    float ir = cosf(phase_incr_angle), ii = sinf(phase_incr_angle);
    size_t m = n_in < n_out ? n_in : n_out;
    float pr = phase->re, pi = phase->im;
    for (size_t i = 0; i < m; i++) {
        float nr = pr * ir - pi * ii;
        float ni = pr * ii + pi * ir;
        pr = nr; pi = ni;
        out[i].re = in[i].re * pr - in[i].im * pi;
        out[i].im = in[i].re * pi + in[i].im * pr;
    }
    phase->re = pr; phase->im = pi;
    return m;
}

/* ---------- firdes ----------------------------------------------------- */

double oracle_besseli0(double x) {
    /* special_funs.rs:22-45 (Abramowitz & Stegun 9.8.1/9.8.2) */
    double t = x / 3.75;
    if (fabs(x) <= 3.75) {
        double t2 = t * t;
        return 1.0 + 3.5156229 * t2 + 3.0899424 * t2 * t2
             + 1.2067492 * pow(t, 6.0) + 0.2659732 * pow(t, 8.0)
             + 0.0360768 * pow(t, 10.0) + 0.0045813 * pow(t, 12.0);
    }
    return 1.0 / (sqrt(fabs(x)) * exp(-x))
         * (0.39894228 + 0.01328592 * pow(t, -1.0) + 0.00225319 * pow(t, -2.0)
            - 0.00157565 * pow(t, -3.0) + 0.00916281 * pow(t, -4.0)
            - 0.02057706 * pow(t, -5.0) + 0.02635537 * pow(t, -6.0)
            - 0.01647633 * pow(t, -7.0) + 0.00392377 * pow(t, -8.0));
}

void oracle_kaiser_window(size_t len, double beta, double* out) {
    /* windows.rs:144-152 */
    double alpha = (double)(len - 1) / 2.0;
    double denom = oracle_besseli0(beta);
    for (size_t n = 0; n < len; n++) {
        double q = ((double)n - alpha) / alpha;
        double x = beta * sqrt(1.0 - q * q);
        out[n] = oracle_besseli0(x) / denom;
    }
}

void oracle_firdes_lowpass(double cutoff, const double* window, size_t len,
                           double* out) {
    /* firdes/basic.rs:25-42 */
    double omega_c = 2.0 * M_PI * cutoff;
    double alpha = (double)(len - 1) / 2.0;
    for (size_t n = 0; n < len; n++) {
        double x = (double)n - alpha;
        double filter_tap =
            (x == 0.0) ? omega_c / M_PI : sin(omega_c * x) / (M_PI * x);
        out[n] = window[n] * filter_tap;
    }
}

/* firdes/basic.rs:444-452 */
static double kaiser_beta(double max_ripple) {
    double ripple_db = -20.0 * log10(max_ripple);
    if (ripple_db > 50.0) return 0.1102 * (ripple_db - 8.7);
    if (ripple_db >= 21.0)
        return 0.5842 * pow(ripple_db - 21.0, 0.4)
             + 0.07886 * (ripple_db - 21.0);
    return 0.0;
}

size_t oracle_kaiser_lowpass_f32(double cutoff, double transition_bw,
                                 double max_ripple, float* out, size_t cap) {
    /* firdes/basic.rs:310-321 + design_kaiser_window :454-459 */
    double beta = kaiser_beta(max_ripple);
    double ripple_db = -20.0 * log10(max_ripple);
    size_t num_taps =
        (size_t)(ceil((ripple_db - 7.95) / (14.36 * transition_bw)) + 1.0);
    if (!out || cap < num_taps) return num_taps;
    double* win = (double*)malloc(sizeof(double) * num_taps * 2);
    double* tapsd = win + num_taps;
    oracle_kaiser_window(num_taps, beta, win);
    double omega_c = (2.0 * cutoff + transition_bw) / 2.0; /* :319 */
    oracle_firdes_lowpass(omega_c, win, num_taps, tapsd);
    for (size_t i = 0; i < num_taps; i++) out[i] = (float)tapsd[i];
    free(win);
    return num_taps;
}

size_t oracle_kaiser_multirate_f32(size_t interp, size_t decim,
                                   size_t half_polyphase_len,
                                   double max_ripple, float* out, size_t cap) {
    /* firdes/basic.rs:412-442 */
    if (interp == 1 && decim == 1) {
        if (out && cap >= 1) out[0] = 1.0f;
        return 1;
    }
    size_t band = (interp == 1) ? decim : interp;
    size_t num_taps = 2 * half_polyphase_len * band;
    if (!out || cap < num_taps) return num_taps;
    double beta = kaiser_beta(max_ripple);
    size_t wlen = num_taps + 1;
    double* win = (double*)malloc(sizeof(double) * wlen * 2);
    double* tapsd = win + wlen;
    oracle_kaiser_window(wlen, beta, win);
    for (size_t i = 0; i < wlen; i++) win[i] *= (double)interp; /* :434-437 */
    size_t mx = interp > decim ? interp : decim;
    double omega_c = 1.0 / (2.0 * (double)mx);                  /* :438 */
    oracle_firdes_lowpass(omega_c, win, wlen, tapsd);
    for (size_t i = 0; i < num_taps; i++) out[i] = (float)tapsd[i]; /* :441 truncate */
    free(win);
    return num_taps;
}

/* moving_avg.rs:79-118 */
void oracle_moving_avg(size_t width, float decay_factor, size_t history,
                       float* avg, size_t* i_state,
                       const float* in, size_t n_in,
                       float* out, size_t n_out,
                       size_t* consumed, size_t* produced) {
    size_t cons = 0, prod = 0, i = *i_state;
    while ((cons + 1) * width <= n_in && (prod + 1) * width <= n_out) {
        for (size_t b = 0; b < width; b++) {
            float t = in[cons * width + b];
            if (isfinite(t))
                avg[b] = (1.0f - decay_factor) * avg[b] + decay_factor * t;
            else
                avg[b] *= 1.0f - decay_factor;
        }
        i++;
        if (i == history) {
            memcpy(out + prod * width, avg, width * sizeof(float));
            i = 0;
            prod++;
        }
        cons++;
    }
    *i_state = i;
    *consumed = cons * width;
    *produced = prod * width;
}

/* pfb/channelizer.rs one-shot restatement */
size_t oracle_pfb_channelizer(size_t N, size_t D,
                              const float* taps, size_t n_taps,
                              const ocf32* in, size_t n_in,
                              ocf32* out, size_t out_cap_per_chan) {
    /* partition_filter_taps (utilities.rs:5-25) */
    size_t tpf = (n_taps + N - 1) / N;
    float* part = (float*)calloc(N * tpf, sizeof(float));
    for (size_t i = 0; i < N; i++) {
        size_t cnt = 0;
        for (size_t t = i; t < n_taps; t += N) part[i * tpf + cnt++] = taps[t];
        /* remaining entries stay 0 (the pad) */
    }
    /* window buffers: w[b][0..tpf) oldest->newest (window_buffer.rs) */
    ocf32* win = (ocf32*)calloc(N * tpf, sizeof(ocf32));
    size_t* missing = (size_t*)malloc(N * sizeof(size_t));
    for (size_t b = 0; b < N; b++) missing[b] = tpf;
    size_t base = N - 1;
    size_t consumed = 0;
    /* prefill (:156-180) */
    int all_filled = 0;
    while (!all_filled && consumed < n_in) {
        ocf32* w = win + base * tpf;
        memmove(w, w + 1, (tpf - 1) * sizeof(ocf32));
        w[tpf - 1] = in[consumed];
        if (missing[base]) missing[base]--;
        base = (base == 0) ? N - 1 : base - 1;
        consumed++;
        all_filled = 1;
        for (size_t b = 0; b < N; b++)
            if (missing[b]) all_filled = 0;
    }
    size_t produced = 0;
    if (all_filled) {
        size_t steps = (n_in - consumed) / D;
        if (steps > out_cap_per_chan) steps = out_cap_per_chan;
        double* fr = (double*)malloc(N * 2 * sizeof(double));
        double* fi = fr + N;
        ocf32* fb = (ocf32*)malloc(N * sizeof(ocf32));
        for (size_t k = 0; k < steps; k++) {
            for (size_t j = 0; j < D; j++) { /* :185-191 */
                ocf32* w = win + base * tpf;
                memmove(w, w + 1, (tpf - 1) * sizeof(ocf32));
                w[tpf - 1] = in[consumed + k * D + j];
                base = (base == 0) ? N - 1 : base - 1;
            }
            for (size_t i = 0; i < N; i++) { /* :193-202 */
                size_t b = (base + i + 1) % N;
                const ocf32* w = win + b * tpf;
                float sre = 0.f, sim = 0.f;
                for (size_t t = 0; t < tpf; t++) {
                    float tap = part[i * tpf + (tpf - 1 - t)];
                    sre = sre + w[t].re * tap;
                    sim = sim + w[t].im * tap;
                }
                fb[b].re = sre;
                fb[b].im = sim;
            }
            for (size_t b = 0; b < N; b++) { fr[b] = fb[b].re; fi[b] = fb[b].im; }
            dft_f64((int)N, 1, fr, fi); /* rustfft inverse, unnormalized */
            for (size_t cidx = 0; cidx < N; cidx++)
                out[cidx * out_cap_per_chan + k] =
                    (ocf32){(float)fr[cidx], (float)fi[cidx]};
        }
        produced = steps;
        free(fr);
        free(fb);
    }
    free(part);
    free(win);
    free(missing);
    return produced;
}

size_t oracle_wlan_moving_sum(size_t len, int is_complex,
                              const float* in, size_t n_in_items,
                              float* out, size_t n_out_items) {
    size_t w = is_complex ? 2 : 1;
    size_t pad = len - 1;
    size_t prod = 0;
    while (pad > 0 && prod < n_out_items) { /* :77-84 zero prologue */
        for (size_t q = 0; q < w; q++) out[prod * w + q] = 0.f;
        pad--;
        prod++;
    }
    size_t m = n_in_items + 1 > len ? n_in_items + 1 - len : 0;
    if (m > n_out_items - prod) m = n_out_items - prod;
    for (size_t q = 0; q < w; q++) { /* :92-99 running sum per lane */
        float sum = 0.f;
        for (size_t t = 0; t + 1 < len; t++) sum += in[t * w + q];
        for (size_t i = 0; i < m; i++) {
            sum += in[(i + len - 1) * w + q];
            out[(prod + i) * w + q] = sum;
            sum -= in[i * w + q];
        }
    }
    return prod + m;
}

size_t oracle_cmul_conj(const ocf32* a, size_t n_a, const ocf32* b,
                        size_t n_b, ocf32* out, size_t n_out) {
    size_t m = n_a < n_b ? n_a : n_b;
    if (n_out < m) m = n_out;
    for (size_t i = 0; i < m; i++) {
        out[i].re = a[i].re * b[i].re + a[i].im * b[i].im;
        out[i].im = a[i].im * b[i].re - a[i].re * b[i].im;
    }
    return m;
}

/* ---------- CPU-baseline chain ---------------------------------------- */

/* f32 iterative radix-2 FFT (forward, unnormalized) for the baseline leg:
 * same algorithmic work class as the GPU chain's FFT stage. */
static void fft_f32_pow2(int n, float* re, float* im) {
    for (int i = 1, j = 0; i < n; i++) {
        int bit = n >> 1;
        for (; j & bit; bit >>= 1) j ^= bit;
        j ^= bit;
        if (i < j) {
            float tr = re[i]; re[i] = re[j]; re[j] = tr;
            float ti = im[i]; im[i] = im[j]; im[j] = ti;
        }
    }
    for (int len = 2; len <= n; len <<= 1) {
        float ang = -2.0f * (float)M_PI / (float)len;
        for (int i = 0; i < n; i += len) {
            for (int k = 0; k < len / 2; k++) {
                float wr = cosf(ang * (float)k);
                float wi = sinf(ang * (float)k);
                int a = i + k, b = i + k + len / 2;
                float xr = re[b] * wr - im[b] * wi;
                float xi = re[b] * wi + im[b] * wr;
                re[b] = re[a] - xr; im[b] = im[a] - xi;
                re[a] = re[a] + xr; im[a] = im[a] + xi;
            }
        }
    }
}

size_t oracle_chain_cf32(const float* taps1, size_t n_taps1,
                         const float* taps2, size_t n_taps2, size_t decim,
                         size_t fft_len,
                         const ocf32* in, size_t n_in,
                         ocf32* out_spectra, size_t n_out_cap,
                         int nthreads) {
    size_t y1_total = sat_sub(n_in + 1, n_taps1);           /* FIR1 outputs */
    size_t y2_total = sat_sub(y1_total + 1, n_taps2) / decim;
    size_t frames = y2_total / fft_len;
    if (out_spectra && n_out_cap < frames * fft_len)
        frames = n_out_cap / fft_len;
    if (frames == 0) return 0;

    long f;
#ifdef _OPENMP
    if (nthreads > 0) omp_set_num_threads(nthreads);
#else
    (void)nthreads;
#endif
#pragma omp parallel
    {
        size_t span1 = (fft_len - 1) * decim + n_taps2; /* y1 span / frame */
        float* y1re = (float*)malloc(sizeof(float) * span1 * 2);
        float* y1im = y1re + span1;
        float* fre = (float*)malloc(sizeof(float) * fft_len * 2);
        float* fim = fre + fft_len;
#pragma omp for schedule(static)
        for (f = 0; f < (long)frames; f++) {
            size_t y2_base = (size_t)f * fft_len;
            /* y2[k] = sum_t y1[decim-1 + k*decim + t] * taps2rev — so this
             * frame needs y1[decim-1 + y2_base*decim .. +span1-1], and
             * y1[j] = sum_t in[j+t]*taps1rev. */
            size_t y1_base = decim - 1 + y2_base * decim;
            for (size_t j = 0; j < span1; j++) {
                float sre = 0.0f, sim = 0.0f;
                const ocf32* xp = in + y1_base + j;
                for (size_t t = 0; t < n_taps1; t++) {
                    float tap = taps1[n_taps1 - 1 - t];
                    sre = sre + xp[t].re * tap;
                    sim = sim + xp[t].im * tap;
                }
                y1re[j] = sre; y1im[j] = sim;
            }
            for (size_t k = 0; k < fft_len; k++) {
                float sre = 0.0f, sim = 0.0f;
                size_t base = k * decim;        /* decim-1 folded into y1_base */
                for (size_t t = 0; t < n_taps2; t++) {
                    float tap = taps2[n_taps2 - 1 - t];
                    sre = sre + y1re[base + t] * tap;
                    sim = sim + y1im[base + t] * tap;
                }
                fre[k] = sre; fim[k] = sim;
            }
            if ((fft_len & (fft_len - 1)) == 0) {
                fft_f32_pow2((int)fft_len, fre, fim);
            } else { /* non-pow2 frames: exact f64 DFT (test infra) */
                double* dr = (double*)malloc(2 * fft_len * sizeof(double));
                double* di = dr + fft_len;
                for (size_t q = 0; q < fft_len; q++) {
                    dr[q] = fre[q];
                    di[q] = fim[q];
                }
                dft_f64((int)fft_len, 0, dr, di);
                for (size_t q = 0; q < fft_len; q++) {
                    fre[q] = (float)dr[q];
                    fim[q] = (float)di[q];
                }
                free(dr);
            }
            if (out_spectra) {
                for (size_t k = 0; k < fft_len; k++) {
                    out_spectra[y2_base + k].re = fre[k];
                    out_spectra[y2_base + k].im = fim[k];
                }
            } else {
                /* keep the work observable (NullSink discards) */
                volatile float sink = fre[0] + fim[fft_len - 1];
                (void)sink;
            }
        }
        free(y1re);
        free(fre);
    }
    return frames * fft_len * decim;            /* chain-input samples */
}

/* Vectorized CPU-baseline chain (bench leg only; the parity oracle is
 * oracle_chain_cf32 above, strict two-stage order). Executes the SAME
 * fused algorithm as the GPU chain (taps1 (*) taps2 convolved in f64 into
 * one decimating filter), on deinterleaved re/im planes so the 253-tap
 * dot product is a stride-1 loop the compiler turns into AVX2 FMAs
 * (omp simd reduction). This is the defensible all-cores number a tuned
 * futuredsp-style CPU implementation would post. */
size_t oracle_chain_cf32_fast(const float* taps1, size_t n_taps1,
                              const float* taps2, size_t n_taps2,
                              size_t decim, size_t fft_len,
                              const ocf32* in, size_t n_in,
                              ocf32* out_spectra, size_t n_out_cap,
                              int nthreads) {
    size_t g_len = n_taps1 + n_taps2 - 1;
    double* gd = (double*)calloc(g_len, sizeof(double));
    float* grev = (float*)malloc(g_len * sizeof(float));
    if (!gd || !grev) { free(gd); free(grev); return 0; }
    for (size_t a = 0; a < n_taps1; a++)
        for (size_t b = 0; b < n_taps2; b++)
            gd[a + b] += (double)taps1[a] * (double)taps2[b];
    for (size_t i = 0; i < g_len; i++)
        grev[i] = (float)gd[g_len - 1 - i];
    free(gd);

    size_t y1_total = sat_sub(n_in + 1, n_taps1);
    size_t y2_total = sat_sub(y1_total + 1, n_taps2) / decim;
    size_t frames = y2_total / fft_len;
    if (out_spectra && n_out_cap < frames * fft_len)
        frames = n_out_cap / fft_len;
    if (frames == 0) { free(grev); return 0; }

    size_t need = decim - 1 + (frames * fft_len - 1) * decim + g_len;
    float* xre = (float*)malloc(need * sizeof(float));
    float* xim = (float*)malloc(need * sizeof(float));
    if (!xre || !xim) { free(xre); free(xim); free(grev); return 0; }

    long i, f;
#ifdef _OPENMP
    if (nthreads > 0) omp_set_num_threads(nthreads);
#else
    (void)nthreads;
#endif
#pragma omp parallel
    {
        float* fre = (float*)malloc(sizeof(float) * fft_len * 2);
        float* fim = fre + fft_len;
#pragma omp for schedule(static)
        for (i = 0; i < (long)need; i++) {
            xre[i] = in[i].re;
            xim[i] = in[i].im;
        }
#pragma omp for schedule(static)
        for (f = 0; f < (long)frames; f++) {
            size_t y2_base = (size_t)f * fft_len;
            for (size_t k = 0; k < fft_len; k++) {
                size_t base = decim - 1 + (y2_base + k) * decim;
                float sre = 0.0f, sim = 0.0f;
                const float* xr = xre + base;
                const float* xi = xim + base;
#pragma omp simd reduction(+ : sre, sim)
                for (size_t t = 0; t < g_len; t++) {
                    sre += xr[t] * grev[t];
                    sim += xi[t] * grev[t];
                }
                fre[k] = sre;
                fim[k] = sim;
            }
            if ((fft_len & (fft_len - 1)) == 0) {
                fft_f32_pow2((int)fft_len, fre, fim);
            } else { /* non-pow2 frames: exact f64 DFT (test infra) */
                double* dr = (double*)malloc(2 * fft_len * sizeof(double));
                double* di = dr + fft_len;
                for (size_t q = 0; q < fft_len; q++) {
                    dr[q] = fre[q];
                    di[q] = fim[q];
                }
                dft_f64((int)fft_len, 0, dr, di);
                for (size_t q = 0; q < fft_len; q++) {
                    fre[q] = (float)dr[q];
                    fim[q] = (float)di[q];
                }
                free(dr);
            }
            if (out_spectra) {
                for (size_t k = 0; k < fft_len; k++) {
                    out_spectra[y2_base + k].re = fre[k];
                    out_spectra[y2_base + k].im = fim[k];
                }
            } else {
                volatile float sink = fre[0] + fim[fft_len - 1];
                (void)sink;
            }
        }
        free(fre);
    }
    free(xre);
    free(xim);
    free(grev);
    return frames * fft_len * decim;
}
