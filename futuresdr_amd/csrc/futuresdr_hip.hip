/* futuresdr_hip.hip — MI355X-native (gfx950/CDNA4) implementation of the
 * FutureSDR streaming-DSP hot path, behind the C-ABI in
 * include/futuresdr_hip.h.
 *
 * Written for CDNA4 from scratch: 64-wide wavefronts, LDS-staged FIR tiles
 * with register sliding windows, SoA re/im LDS planes with a 2-dwords-per-16
 * pad (breaks the stride-16B bank pattern of the pair reads), radix-2
 * Stockham FFT in LDS. No CUDA shims, no hipify output.
 *
 * Numerical semantics follow the reference cores (cited per function); the
 * status/consumed/produced math is bit-exact, the float results are
 * tolerance-compared (the reference itself permits reassociation on
 * nightly — crates/futuredsp/src/fir.rs:93-200 — and uses 5*eps for GPU
 * parity, examples/vulkan/src/main.rs:103).
 */
#include <hip/hip_runtime.h>

#include <atomic>
#include <condition_variable>
#include <cstdio>
#include <cstring>
#include <cstdlib>
#include <deque>
#include <mutex>
#include <vector>
#include <algorithm>
#include <cmath>

#include "../../include/futuresdr_hip.h"

/* ================= error plumbing ==================================== */

static thread_local char g_err[512];
static thread_local const char* g_err_ptr = "";

static void set_err(const char* msg) {
    snprintf(g_err, sizeof(g_err), "%s", msg);
    g_err_ptr = g_err;
}

extern "C" const char* fsdr_last_error(void) { return g_err_ptr; }
extern "C" const char* fsdr_version(void) { return "futuresdr-hip 0.1 gfx950"; }

#define HIP_TRY(call)                                                        \
    do {                                                                     \
        hipError_t e_ = (call);                                              \
        if (e_ != hipSuccess) {                                              \
            snprintf(g_err, sizeof(g_err), "%s failed: %s (%s:%d)", #call,   \
                     hipGetErrorString(e_), __FILE__, __LINE__);             \
            g_err_ptr = g_err;                                               \
            return (e_ == hipErrorNoDevice || e_ == hipErrorInvalidDevice)   \
                       ? FSDR_ERR_NO_GPU                                     \
                       : FSDR_ERR_HIP;                                       \
        }                                                                    \
    } while (0)

static bool have_gpu() {
    static int cached = -1;
    if (cached < 0) {
        int n = 0;
        cached = (hipGetDeviceCount(&n) == hipSuccess && n > 0) ? 1 : 0;
    }
    return cached == 1;
}

#define REQUIRE_GPU()                                                        \
    do {                                                                     \
        if (!have_gpu()) {                                                   \
            set_err("no HIP device present — the futuresdr_hip product "     \
                    "path has no CPU fallback (oracle/ is test infra)");     \
            return FSDR_ERR_NO_GPU;                                          \
        }                                                                    \
    } while (0)

extern "C" int fsdr_device_count(void) {
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) return 0;
    return n;
}
extern "C" int fsdr_set_device(int device) {
    REQUIRE_GPU();
    HIP_TRY(hipSetDevice(device));
    return FSDR_OK;
}
extern "C" int fsdr_synchronize(void) {
    REQUIRE_GPU();
    HIP_TRY(hipDeviceSynchronize());
    return FSDR_OK;
}

/* ================= shared device helpers ============================== */

/* LDS SoA plane padding: +2 dwords per 16 — breaks the 16 B lane-stride
 * bank pattern of the f32-pair (ds_read_b64) window loads while keeping
 * even-indexed pairs contiguous and 8 B-aligned (DESIGN.md kernel notes). */
__device__ __host__ __forceinline__ unsigned lds_pad(unsigned e) {
    return e + ((e >> 4) << 1);
}

/* floats per SoA plane for a tile of `elems` elements, rounded to a
 * 16 B multiple so the second plane stays aligned for pair reads. */
__device__ __host__ __forceinline__ unsigned plane_floats(unsigned elems) {
    return (lds_pad(elems - 1) + 4u) & ~3u;
}

/* LDS index swizzle for the ping/pong buffers: float2 element i sits at
 * bank (2i) mod 64, so accesses whose lane stride is a multiple of 32
 * elements (the strided Stockham stages) are up-to-8-way conflicted;
 * XORing bits 5..7 of the element index into bits 2..4 makes them
 * conflict-free while keeping contiguous stages conflict-free. */
__device__ __forceinline__ unsigned fft_swz(unsigned i) {
    /* full 5-bit XOR: two elements collide on a bank only when their
     * indices differ by a multiple of 1024 — conflict-free for every
     * Stockham stage stride at n <= 1024 (and 2-way max at 2048/4096).
     * FFT accesses are single float2 elements, so remapping any bit of
     * the element index is layout-legal. */
    return i ^ ((i >> 5) & 31u);
}

__device__ __forceinline__ float2 f2_add(float2 a, float2 b) {
    return make_float2(a.x + b.x, a.y + b.y);
}
__device__ __forceinline__ float2 f2_sub(float2 a, float2 b) {
    return make_float2(a.x - b.x, a.y - b.y);
}
__device__ __forceinline__ float2 cmulf(float2 a, float2 b) {
    return make_float2(a.x * b.x - a.y * b.y, a.x * b.y + a.y * b.x);
}

/* ================= FIR (decimation 1), Complex32 x f32 taps =========== *
 * Restates fir_kernel_core + the Complex<f32>/f32 MAC
 * (crates/futuredsp/src/fir.rs:76-88, 242-249): y[k] = sum_t x[k+t] *
 * h[T-1-t], accumulating re/im separately in fp32.
 *
 * Tiling: 256 lanes/block, R=4 consecutive outputs per lane, input tile in
 * SoA LDS planes (pad-2-per-16 keeps the 16 B lane-stride pair reads
 * conflict-free). Inner loop: 2 taps per step, 16 FMAs per step; the
 * window pair AND the (LDS-staged, reversed) tap pair for step s+1 are
 * prefetched during step s so no FMA waits on a just-issued LDS read
 * (the v1 kernel was SQ_WAIT_INST-bound at 28% of fp32 peak for exactly
 * that reason — profiles/rocprof_r01*). Requires n_taps_padded % 8 == 1
 * (host pads taps with leading zeros; padded taps multiply staged zeros
 * only). */

#define FIR_BLOCK 256
#define FIR_R 4
#define FIR_TILE_OUT (FIR_BLOCK * FIR_R) /* 1024 outputs per tile */

__global__ __launch_bounds__(FIR_BLOCK) void k_fir_cf32(
    const float2* __restrict__ in, float2* __restrict__ out,
    const float* __restrict__ taps, int n_taps_padded, long long n_out,
    long long n_in_valid) {
    const int tp = n_taps_padded;                     /* tp % 8 == 1 */
    const unsigned elems = FIR_TILE_OUT + tp - 1 + 6; /* + prefetch slack */
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* s_re = (float*)smem;
    float* s_im = s_re + plane_floats(elems);
    float* s_rt = s_im + plane_floats(elems); /* reversed taps, tp+1 slots */

    const int tid = threadIdx.x;
    for (long long tile = blockIdx.x;
         tile * (long long)FIR_TILE_OUT < n_out; tile += gridDim.x) {
        const long long out_base = tile * FIR_TILE_OUT;
        for (unsigned i = tid; i < elems; i += FIR_BLOCK) {
            long long g = out_base + i;
            float2 v = (g < n_in_valid) ? in[g] : make_float2(0.f, 0.f);
            s_re[lds_pad(i)] = v.x;
            s_im[lds_pad(i)] = v.y;
        }
        for (int i = tid; i <= tp; i += FIR_BLOCK)
            s_rt[i] = (i < tp) ? taps[tp - 1 - i] : 0.f;
        __syncthreads();

        const unsigned eb = (unsigned)tid * FIR_R; /* tile-relative base */
        float ar[FIR_R] = {0.f, 0.f, 0.f, 0.f};
        float ai[FIR_R] = {0.f, 0.f, 0.f, 0.f};
        /* 4-pair rotating window: slot q holds pair q mod 4 (elements
         * 2q, 2q+1 relative to eb), flat layout w[2*slot + parity]. */
        float wre[8], wim[8];
#pragma unroll
        for (int p = 0; p < 3; p++) {
            float2 pr = *(const float2*)&s_re[lds_pad(eb + 2 * p)];
            float2 pi = *(const float2*)&s_im[lds_pad(eb + 2 * p)];
            wre[2 * p] = pr.x; wre[2 * p + 1] = pr.y;
            wim[2 * p] = pi.x; wim[2 * p + 1] = pi.y;
        }
        float2 ht0 = *(const float2*)&s_rt[0];
        const int pair_steps = (tp - 1) / 2; /* multiple of 4 */
        int s = 0;
        /* flat window index of tile-relative element e = 2s + d at phase
         * U = s%4: 2*((U + d/2) % 4) + d%2, d in [0,5] */
#define FIR_W(U, d) (2 * (((U) + (d) / 2) % 4) + (d) % 2)
#define FIR_STEP(U)                                                          \
    do {                                                                     \
        float2 nr = *(const float2*)&s_re[lds_pad(eb + 2 * s + 6)];          \
        float2 ni = *(const float2*)&s_im[lds_pad(eb + 2 * s + 6)];          \
        float2 ht1 = *(const float2*)&s_rt[2 * s + 2];                       \
        _Pragma("unroll") for (int j = 0; j < FIR_R; j++) {                  \
            ar[j] = fmaf(wre[FIR_W(U, j)], ht0.x, ar[j]);                    \
            ai[j] = fmaf(wim[FIR_W(U, j)], ht0.x, ai[j]);                    \
            ar[j] = fmaf(wre[FIR_W(U, j + 1)], ht0.y, ar[j]);                \
            ai[j] = fmaf(wim[FIR_W(U, j + 1)], ht0.y, ai[j]);                \
        }                                                                    \
        wre[2 * (((U) + 3) % 4)] = nr.x;                                     \
        wre[2 * (((U) + 3) % 4) + 1] = nr.y;                                 \
        wim[2 * (((U) + 3) % 4)] = ni.x;                                     \
        wim[2 * (((U) + 3) % 4) + 1] = ni.y;                                 \
        ht0 = ht1;                                                           \
        s++;                                                                 \
    } while (0)
        for (; s < pair_steps;) {
            FIR_STEP(0);
            FIR_STEP(1);
            FIR_STEP(2);
            FIR_STEP(3);
        }
#undef FIR_STEP
        /* final (unpaired) tap t = tp-1: element e = tp-1+j is in pair
         * ps + j/2 -> slot j/2 (ps % 4 == 0), parity j%2 */
        {
            const float h0 = s_rt[tp - 1];
#pragma unroll
            for (int j = 0; j < FIR_R; j++) {
                ar[j] = fmaf(wre[2 * (j / 2) + (j % 2)], h0, ar[j]);
                ai[j] = fmaf(wim[2 * (j / 2) + (j % 2)], h0, ai[j]);
            }
        }
#pragma unroll
        for (int j = 0; j < FIR_R; j++) {
            long long o = out_base + eb + j;
            if (o < n_out) out[o] = make_float2(ar[j], ai[j]);
        }
        __syncthreads();
    }
}

/* ---- Fully-unrolled compile-time-tap-count FIR variant --------------- *
 * Same math as k_fir_cf32; TP is the padded tap count (TP % 4 == 1) and
 * the device tap array is REVERSED (rt[i] = h[n_taps-1-i], zero-filled to
 * TP). Linear SoA LDS planes (no pad): the window is read as float4
 * groups at 16 B lane stride (the conflict-free ds_read_b128 pattern) and
 * addresses are affine (fold to ds_read immediate offsets). Reversed taps
 * are staged in LDS too (keeping SMEM out of the loop: s_load shares
 * lgkmcnt with ds_read and forces lgkmcnt(0) drains). Accumulators are
 * float2 OUTPUT pairs so every v_pk_fma reads two adjacent float4
 * components — no register-pairing movs (the v3 variant spent ~40% of
 * VALU on such movs). */
template <int TP>
__global__ __launch_bounds__(FIR_BLOCK) void k_fir_cf32_tpl(
    const float2* __restrict__ in, float2* __restrict__ out,
    const float* __restrict__ rtaps, long long n_out, long long n_in_valid) {
    static_assert(TP % 4 == 1, "TP must be 1 mod 4");
    const unsigned elems = FIR_TILE_OUT + TP + 8;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* s_re = (float*)smem;
    float* s_im = s_re + ((elems + 7u) & ~7u);
    float* s_rt = s_im + ((elems + 7u) & ~7u); /* TP+3 floats */

    const int tid = threadIdx.x;
    for (long long tile = blockIdx.x;
         tile * (long long)FIR_TILE_OUT < n_out; tile += gridDim.x) {
        const long long out_base = tile * FIR_TILE_OUT;
        for (unsigned i = tid; i < elems; i += FIR_BLOCK) {
            long long g = out_base + i;
            float2 v = (g < n_in_valid) ? in[g] : make_float2(0.f, 0.f);
            s_re[i] = v.x;
            s_im[i] = v.y;
        }
        for (int i = tid; i < TP + 3; i += FIR_BLOCK)
            s_rt[i] = (i < TP) ? rtaps[i] : 0.f;
        __syncthreads();

        const unsigned eb = (unsigned)tid * FIR_R;
        float2 a01r = make_float2(0.f, 0.f), a23r = a01r;
        float2 a01i = a01r, a23i = a01r;
        float4 r0 = *(const float4*)&s_re[eb];
        float4 r1 = *(const float4*)&s_re[eb + 4];
        float4 i0 = *(const float4*)&s_im[eb];
        float4 i1 = *(const float4*)&s_im[eb + 4];
        float4 hc = *(const float4*)&s_rt[0];
        constexpr int NG = (TP - 1) / 4; /* 4 taps per group */
        /* software pipeline: every load issued one group ahead of its use,
         * so in steady state no FMA waits on a just-issued ds_read */
#pragma unroll 8
        for (int m = 0; m < NG; m++) {
            const float4 rn = *(const float4*)&s_re[eb + 4 * m + 8];
            const float4 in_ = *(const float4*)&s_im[eb + 4 * m + 8];
            const float4 h4 = *(const float4*)&s_rt[4 * m + 4];
            const float wr[8] = {r0.x, r0.y, r0.z, r0.w,
                                 r1.x, r1.y, r1.z, r1.w};
            const float wi[8] = {i0.x, i0.y, i0.z, i0.w,
                                 i1.x, i1.y, i1.z, i1.w};
            const float ht[4] = {hc.x, hc.y, hc.z, hc.w};
#pragma unroll
            for (int tl = 0; tl < 4; tl++) {
                const float h = ht[tl];
                a01r.x = fmaf(wr[tl], h, a01r.x);
                a01r.y = fmaf(wr[tl + 1], h, a01r.y);
                a23r.x = fmaf(wr[tl + 2], h, a23r.x);
                a23r.y = fmaf(wr[tl + 3], h, a23r.y);
                a01i.x = fmaf(wi[tl], h, a01i.x);
                a01i.y = fmaf(wi[tl + 1], h, a01i.y);
                a23i.x = fmaf(wi[tl + 2], h, a23i.x);
                a23i.y = fmaf(wi[tl + 3], h, a23i.y);
            }
            r0 = r1; r1 = rn;
            i0 = i1; i1 = in_;
            hc = h4;
        }
        { /* final tap TP-1: r0 = elements TP-1..TP+2, hc = rt[TP-1..] */
            const float h = hc.x;
            a01r.x = fmaf(r0.x, h, a01r.x);
            a01r.y = fmaf(r0.y, h, a01r.y);
            a23r.x = fmaf(r0.z, h, a23r.x);
            a23r.y = fmaf(r0.w, h, a23r.y);
            a01i.x = fmaf(i0.x, h, a01i.x);
            a01i.y = fmaf(i0.y, h, a01i.y);
            a23i.x = fmaf(i0.z, h, a23i.x);
            a23i.y = fmaf(i0.w, h, a23i.y);
        }
        const float ar[4] = {a01r.x, a01r.y, a23r.x, a23r.y};
        const float ai[4] = {a01i.x, a01i.y, a23i.x, a23i.y};
#pragma unroll
        for (int j = 0; j < FIR_R; j++) {
            long long o = out_base + eb + j;
            if (o < n_out) out[o] = make_float2(ar[j], ai[j]);
        }
        __syncthreads();
    }
}

/* instantiation dispatch for the templated FIR (host side below) */
typedef void (*fir_tpl_fn)(const float2*, float2*, const float*, long long,
                           long long);
template <int TP>
static fir_tpl_fn fir_tpl_ptr() { return k_fir_cf32_tpl<TP>; }

/* ================= Decimating FIR (D=4 fast path), cf32 x f32 ========= *
 * Restates decimating_fir.rs:80-95 for Complex<f32>/f32: y[k] =
 * sum_t x[D-1 + k*D + t] * h[T-1-t].
 *
 * 256 lanes, R2=4 consecutive decimated outputs per lane; 8-pair rotating
 * window with one-step-ahead prefetch of both the window pair and the
 * LDS-staged reversed tap pair (same latency-hiding structure as
 * k_fir_cf32). Taps processed as tap 0 (prologue) + pairs (1,2),(3,4),...
 * Requires n_taps_padded % 16 == 1. */

#define DFIR_BLOCK 256
#define DFIR_R 4
#define DFIR_TILE_OUT (DFIR_BLOCK * DFIR_R) /* 1024 decimated outputs */

__global__ __launch_bounds__(DFIR_BLOCK) void k_fir_decim4_cf32(
    const float2* __restrict__ in, float2* __restrict__ out,
    const float* __restrict__ taps, int n_taps_padded, long long n_out,
    long long n_in_valid) {
    constexpr int D = 4;
    const int tp = n_taps_padded; /* tp % 16 == 1 */
    const unsigned elems = DFIR_TILE_OUT * D + tp - 1 + 20;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* s_re = (float*)smem;
    float* s_im = s_re + plane_floats(elems);
    float* s_rt = s_im + plane_floats(elems); /* reversed taps MINUS tap0 */

    const int tid = threadIdx.x;
    for (long long tile = blockIdx.x;
         tile * (long long)DFIR_TILE_OUT < n_out; tile += gridDim.x) {
        const long long out_base = tile * DFIR_TILE_OUT;
        const long long in_base = out_base * D;
        for (unsigned i = tid; i < elems; i += DFIR_BLOCK) {
            long long g = in_base + i;
            float2 v = (g < n_in_valid) ? in[g] : make_float2(0.f, 0.f);
            s_re[lds_pad(i)] = v.x;
            s_im[lds_pad(i)] = v.y;
        }
        /* s_rt[i] = coefficient of tap t=i+1, i.e. taps[tp-2-i] */
        for (int i = tid; i <= tp; i += DFIR_BLOCK)
            s_rt[i] = (i < tp - 1) ? taps[tp - 2 - i] : 0.f;
        __syncthreads();

        /* lane-relative input element for (output j, tap t):
         * e = D-1 + 4j + t, with lane base eb = tid*R2*D */
        const unsigned eb = (unsigned)tid * DFIR_R * D;
        float ar[DFIR_R] = {0.f, 0.f, 0.f, 0.f};
        float ai[DFIR_R] = {0.f, 0.f, 0.f, 0.f};
        /* tap 0 prologue: e = 3 + 4j (odd -> single b32 reads) */
        {
            const float h = taps[tp - 1];
#pragma unroll
            for (int j = 0; j < DFIR_R; j++) {
                ar[j] = fmaf(s_re[lds_pad(eb + 3 + 4 * j)], h, ar[j]);
                ai[j] = fmaf(s_im[lds_pad(eb + 3 + 4 * j)], h, ai[j]);
            }
        }
        /* step s covers taps (1+2s, 2+2s); output j reads BOTH elements of
         * pair s+2+2j; window = 8 rotating pair slots (p % 8), prefetch
         * pair s+9 during step s. */
        float wre[16], wim[16];
#pragma unroll
        for (int p = 2; p <= 8; p++) {
            float2 pr = *(const float2*)&s_re[lds_pad(eb + 2 * p)];
            float2 pi = *(const float2*)&s_im[lds_pad(eb + 2 * p)];
            wre[(p % 8) * 2] = pr.x; wre[(p % 8) * 2 + 1] = pr.y;
            wim[(p % 8) * 2] = pi.x; wim[(p % 8) * 2 + 1] = pi.y;
        }
        float2 ht0 = *(const float2*)&s_rt[0];
        const int pair_steps = (tp - 1) / 2; /* multiple of 8 */
        int s = 0;
#define DFIR_STEP(U)                                                         \
    do {                                                                     \
        float2 nr = *(const float2*)&s_re[lds_pad(eb + 2 * (s + 9))];        \
        float2 ni = *(const float2*)&s_im[lds_pad(eb + 2 * (s + 9))];        \
        float2 ht1 = *(const float2*)&s_rt[2 * s + 2];                       \
        _Pragma("unroll") for (int j = 0; j < DFIR_R; j++) {                 \
            const int sl = (((U) + 2 + 2 * j) % 8) * 2;                      \
            ar[j] = fmaf(wre[sl], ht0.x, ar[j]);                             \
            ai[j] = fmaf(wim[sl], ht0.x, ai[j]);                             \
            ar[j] = fmaf(wre[sl + 1], ht0.y, ar[j]);                         \
            ai[j] = fmaf(wim[sl + 1], ht0.y, ai[j]);                         \
        }                                                                    \
        wre[(((U) + 1) % 8) * 2] = nr.x;                                     \
        wre[(((U) + 1) % 8) * 2 + 1] = nr.y;                                 \
        wim[(((U) + 1) % 8) * 2] = ni.x;                                     \
        wim[(((U) + 1) % 8) * 2 + 1] = ni.y;                                 \
        ht0 = ht1;                                                           \
        s++;                                                                 \
    } while (0)
        for (; s < pair_steps;) {
            DFIR_STEP(0); DFIR_STEP(1); DFIR_STEP(2); DFIR_STEP(3);
            DFIR_STEP(4); DFIR_STEP(5); DFIR_STEP(6); DFIR_STEP(7);
        }
#undef DFIR_STEP
#pragma unroll
        for (int j = 0; j < DFIR_R; j++) {
            long long o = out_base + (unsigned)tid * DFIR_R + j;
            if (o < n_out) out[o] = make_float2(ar[j], ai[j]);
        }
        __syncthreads();
    }
}

/* ---- MFMA FIR variant (f32-input matrix cores) ----------------------- *
 * Same math as k_fir_cf32_tpl, reformulated as an implicit GEMM for the
 * exact-f32 matrix cores (v_mfma_f32_16x16x4_f32 — same 157.3 TF peak as
 * the f32 VALU but ~30x fewer instructions, so the VALU overhead and
 * issue stalls of the FMA kernel disappear):
 *   C[i][j] = sum_k X[i][k] * H[k][j],  X[i][k] = x[base + 16 i + k],
 *   H[k][j] = rt[k - j]  (0 <= k-j < T else 0),  K = KK >= T+15, KK%4==0
 * => C[i][j] = y[base + 16 i + j]; one 16x16 C tile (re + im, two
 * accumulator chains = full MFMA issue rate) per wave per 256 outputs.
 * The H fragments depend only on the taps: computed once per block into
 * KK/4 VGPRs. A fragments are single ds_read_b32 per MFMA from linear
 * SoA planes with an XOR bank swizzle (rows stride 16 dwords; lanes r,
 * r+2m collide mod 32, so XOR bits 5..7 of the dword index into bits
 * 2..4 — conflict-free, verified in the bank math of DESIGN.md). */
typedef float v4f __attribute__((ext_vector_type(4)));

__device__ __forceinline__ unsigned mfma_swz(unsigned idx) {
    return idx ^ (((idx >> 5) & 7u) << 2);
}

#define MFIR_BLOCK 256
#define MFIR_TILE 1024 /* 4 waves x 256 outputs */

template <int KK>
__global__ __launch_bounds__(MFIR_BLOCK) void k_fir_mfma_tpl(
    const float2* __restrict__ in, float2* __restrict__ out,
    const float* __restrict__ rtaps /* reversed, zero-filled to KK */,
    long long n_out, long long n_in_valid) {
    static_assert(KK % 4 == 0, "KK must be a multiple of 4");
    const unsigned elems = MFIR_TILE + KK + 8;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* s_re = (float*)smem;
    float* s_im = s_re + ((elems + 31u) & ~31u);
    float* s_rtx = s_im + ((elems + 31u) & ~31u); /* 15 zeros + rt[KK] + 1 */

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int r16 = lane & 15;   /* A row / C col */
    const int k4 = lane >> 4;    /* k sub-slice 0..3 */

    /* B fragments: lane holds H[k0 + k4][j = r16] = rtx[k0+k4-r16],
     * identical for every tile -> compute once (s_rtx has a 15-zero
     * prologue so the index is never negative). */
    for (int i = tid; i < KK + 16; i += MFIR_BLOCK)
        s_rtx[i] = (i >= 15 && i < 15 + KK) ? rtaps[i - 15] : 0.f;
    __syncthreads();
    float bfrag[KK / 4];
#pragma unroll
    for (int s = 0; s < KK / 4; s++)
        bfrag[s] = s_rtx[15 + 4 * s + k4 - r16];
    __syncthreads();

    /* global staging is software-pipelined across the tile loop: tile
     * t+1's loads are issued right after the staging barrier of tile t,
     * so their HBM latency hides under t's 72 MFMAs. */
    constexpr int NL = (MFIR_TILE + KK + 8 + 2 * MFIR_BLOCK - 1) /
                       (2 * MFIR_BLOCK); /* float4 (2 complex) per slot */
    float4 stg[NL];
    auto load_tile = [&](long long tl) {
        const long long ob = tl * MFIR_TILE;
#pragma unroll
        for (int j = 0; j < NL; j++) {
            unsigned i = 2 * (tid + j * MFIR_BLOCK);
            long long g = ob + i;
            float2 v0 = (i < elems && g < n_in_valid)
                            ? in[g] : make_float2(0.f, 0.f);
            float2 v1 = (i + 1 < elems && g + 1 < n_in_valid)
                            ? in[g + 1] : make_float2(0.f, 0.f);
            stg[j] = make_float4(v0.x, v0.y, v1.x, v1.y);
        }
    };
    load_tile(blockIdx.x);
    for (long long tile = blockIdx.x;
         tile * (long long)MFIR_TILE < n_out; tile += gridDim.x) {
        const long long out_base = tile * MFIR_TILE;
#pragma unroll
        for (int j = 0; j < NL; j++) {
            unsigned i = 2 * (tid + j * MFIR_BLOCK);
            if (i < elems) {
                unsigned d = mfma_swz(i);
                *(float2*)&s_re[d] = make_float2(stg[j].x, stg[j].z);
                *(float2*)&s_im[d] = make_float2(stg[j].y, stg[j].w);
            }
        }
        __syncthreads();
        if ((tile + gridDim.x) * (long long)MFIR_TILE < n_out)
            load_tile(tile + gridDim.x);

        const unsigned ab = (unsigned)wave * 256 + 16u * r16 + k4;
        /* one accumulator pair, re/im alternating: same-C spacing = 2
         * MFMA issues = 64 cyc/SIMD >= the 40-cyc dependent latency
         * (extra pairs measured slower — register pressure only) */
        v4f cre = {0.f, 0.f, 0.f, 0.f};
        v4f cim = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int s = 0; s < KK / 4; s++) {
            float a_re = s_re[mfma_swz(ab + 4 * s)];
            float a_im = s_im[mfma_swz(ab + 4 * s)];
            cre = __builtin_amdgcn_mfma_f32_16x16x4f32(a_re, bfrag[s], cre,
                                                       0, 0, 0);
            cim = __builtin_amdgcn_mfma_f32_16x16x4f32(a_im, bfrag[s], cim,
                                                       0, 0, 0);
        }
        /* C layout: col = lane&15, row = (lane>>4)*4 + q (cdna4 16x16) */
#pragma unroll
        for (int q = 0; q < 4; q++) {
            int row = k4 * 4 + q;
            long long o = out_base + (long long)wave * 256 + 16 * row + r16;
            if (o < n_out) out[o] = make_float2(cre[q], cim[q]);
        }
        __syncthreads();
    }
}

/* ---- MFMA FIR, 32x32x2 variant --------------------------------------- *
 * Same implicit GEMM as k_fir_mfma_tpl but on v_mfma_f32_32x32x2_f32:
 * one 32x32 C tile pair (re+im) = 1024 outputs per WAVE, single-wave
 * blocks — 4x less staging and barrier overhead per output than the
 * 16x16 shape (at the cost of K = T+31 shift inflation). B fragments are
 * re-read from LDS per tile (80 broadcast reads) to keep registers for
 * occupancy. A-plane XOR swizzle: idx ^ (((idx>>5)&15)<<1) makes the
 * 32-dword row stride conflict-free up to a residual 2-way (rows r,
 * r+16) while preserving even-pair adjacency for the staging writes. */
__device__ __forceinline__ unsigned mfma32_swz(unsigned idx) {
    return idx ^ (((idx >> 5) & 15u) << 1);
}

#define MFIR32_BLOCK 64
#define MFIR32_TILE 1024 /* one wave, one 32x32 C pair */

typedef float v16f __attribute__((ext_vector_type(16)));

template <int KK2>
__global__ __launch_bounds__(MFIR32_BLOCK) void k_fir_mfma32_tpl(
    const float2* __restrict__ in, float2* __restrict__ out,
    const float* __restrict__ rtaps /* reversed, zero-filled to KK2 */,
    long long n_out, long long n_in_valid) {
    static_assert(KK2 % 2 == 0, "KK2 must be even");
    const unsigned elems = MFIR32_TILE + KK2 + 8;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* s_re = (float*)smem;
    float* s_im = s_re + ((elems + 31u) & ~31u);
    float* s_rtx = s_im + ((elems + 31u) & ~31u); /* 31 zeros + rt + 1 */

    const int tid = threadIdx.x;
    const int r32 = tid & 31; /* A row / C col */
    const int k2 = tid >> 5;  /* k sub-slice 0..1 */

    for (int i = tid; i < KK2 + 32; i += MFIR32_BLOCK)
        s_rtx[i] = (i >= 31 && i < 31 + KK2) ? rtaps[i - 31] : 0.f;
    __syncthreads();

    constexpr int NL =
        (MFIR32_TILE + KK2 + 8 + 2 * MFIR32_BLOCK - 1) / (2 * MFIR32_BLOCK);
    float4 stg[NL];
    auto load_tile = [&](long long tl) {
        const long long ob = tl * MFIR32_TILE;
#pragma unroll
        for (int j = 0; j < NL; j++) {
            unsigned i = 2 * (tid + j * MFIR32_BLOCK);
            long long g = ob + i;
            float2 v0 = (i < elems && g < n_in_valid)
                            ? in[g] : make_float2(0.f, 0.f);
            float2 v1 = (i + 1 < elems && g + 1 < n_in_valid)
                            ? in[g + 1] : make_float2(0.f, 0.f);
            stg[j] = make_float4(v0.x, v0.y, v1.x, v1.y);
        }
    };
    load_tile(blockIdx.x);
    for (long long tile = blockIdx.x;
         tile * (long long)MFIR32_TILE < n_out; tile += gridDim.x) {
        const long long out_base = tile * MFIR32_TILE;
#pragma unroll
        for (int j = 0; j < NL; j++) {
            unsigned i = 2 * (tid + j * MFIR32_BLOCK);
            if (i < elems) {
                unsigned d = mfma32_swz(i);
                *(float2*)&s_re[d] = make_float2(stg[j].x, stg[j].z);
                *(float2*)&s_im[d] = make_float2(stg[j].y, stg[j].w);
            }
        }
        __syncthreads();
        if ((tile + gridDim.x) * (long long)MFIR32_TILE < n_out)
            load_tile(tile + gridDim.x);

        const unsigned ab = 32u * r32 + k2;
        v16f cre = {};
        v16f cim = {};
#pragma unroll
        for (int s = 0; s < KK2 / 2; s++) {
            float b = s_rtx[31 + 2 * s + k2 - r32];
            float a_re = s_re[mfma32_swz(ab + 2 * s)];
            float a_im = s_im[mfma32_swz(ab + 2 * s)];
            cre = __builtin_amdgcn_mfma_f32_32x32x2f32(a_re, b, cre, 0, 0, 0);
            cim = __builtin_amdgcn_mfma_f32_32x32x2f32(a_im, b, cim, 0, 0, 0);
        }
        /* C layout (cdna4 32x32): col = lane&31,
         * row = (reg&3) + 8*(reg>>2) + 4*(lane>>5) */
#pragma unroll
        for (int q = 0; q < 16; q++) {
            int row = (q & 3) + 8 * (q >> 2) + 4 * k2;
            long long o = out_base + 32 * row + r32;
            if (o < n_out) out[o] = make_float2(cre[q], cim[q]);
        }
        __syncthreads();
    }
}

/* ---- MFMA phase-split decimating FIR (D=4) --------------------------- *
 * decimating_fir.rs:80-95 semantics via the phase decomposition
 * (t = 4u+v => y[k] = sum_v sum_u P_v[k+u]*rtv[v][u], P_v[i] = x[3+v+4i])
 * with each phase dot-product run as the same implicit GEMM as
 * k_fir_mfma_tpl: C accumulates over all 4 phases. KKD >= ceil(T/4)+15,
 * KKD%4==0. Staging is software-pipelined across the tile loop. */
#define MDFIR_BLOCK 256
#define MDFIR_TILE 1024 /* decimated outputs per block */

/* forward declarations (defined with the FFT kernel below): in-block
 * forward Stockham FFT over swizzled LDS ping/pong buffers. fft_pow2_fwd
 * runs any pow2 n (radix-4 + radix-2 tail) with tpf threads striding the
 * butterflies and returns the buffer holding the result; ALL threads of
 * the block must call it with the same n (block-wide barriers inside). */
__device__ float2* fft_pow2_fwd(float2* a, float2* b,
                                const float2* __restrict__ twid, int n,
                                int tf, int tpf);
__device__ void fft1024_block(float2* ping, float2* pong,
                              const float2* __restrict__ twid, int tf);
/* fft1024_block result lands in `pong` (5 swaps); read pong[fft_swz(i)] */

template <int KKD>
__global__ __launch_bounds__(MDFIR_BLOCK) void k_decim4_mfma_tpl(
    const float2* __restrict__ in, float2* __restrict__ out,
    const float* __restrict__ rtv /* [4][KKD], rtv[v][u] = rt[4u+v] */,
    long long n_out, long long n_in_valid) {
    static_assert(KKD % 4 == 0, "KKD must be a multiple of 4");
    const unsigned elemsP = MDFIR_TILE + KKD + 8;     /* per phase plane */
    const unsigned SPm = (elemsP + 31u) & ~31u;
    /* each phase plane is split into 4 sub-planes by element%4 so a
     * lane's MFMA K-walk (idx = ab+4s) becomes stride-1: one
     * ds_read_b128 feeds 4 K-steps (the b32-per-MFMA A-reads were the
     * round-1 issue-stall, SQ_WAIT_INST_ANY 46% — profiles/pmc_r02) */
    const unsigned SUB = SPm / 4;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    /* two phases resident at a time ([re_v0, re_v1, im_v0, im_v1]) —
     * halves LDS vs all-phase planes, doubling resident blocks/CU;
     * accumulators carry across the two halves, and each half's global
     * loads are issued under the other half's MFMAs. */
    float* planes = (float*)smem;        /* [4][SPm] */
    float* s_rtx = planes + 4u * SPm;    /* [4][KKD+16], 15-zero prologue */

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int r16 = lane & 15;
    const int k4 = lane >> 4;

    for (int i = tid; i < 4 * (KKD + 16); i += MDFIR_BLOCK) {
        int v = i / (KKD + 16), t = i % (KKD + 16);
        s_rtx[i] = (t >= 15 && t < 15 + KKD) ? rtv[v * KKD + (t - 15)] : 0.f;
    }
    __syncthreads();

    const unsigned span = 3 + 4 * elemsP; /* input elements per tile */
    constexpr int NL2 =
        (2 * (MDFIR_TILE + KKD + 8) + 3 + MDFIR_BLOCK - 1) / MDFIR_BLOCK;
    float2 stgA[NL2], stgB[NL2];
    /* elements of phases {2h, 2h+1}: rel = 3 + 4i + 2h + vloc */
    auto load_half = [&](long long tl, int h, float2 (&stg)[NL2]) {
        const long long ib = tl * MDFIR_TILE * 4;
#pragma unroll
        for (int j = 0; j < NL2; j++) {
            unsigned idx = (unsigned)(tid + j * MDFIR_BLOCK);
            unsigned rel = 3 + 4 * (idx >> 1) + 2 * h + (idx & 1u);
            long long g = ib + rel;
            stg[j] = (rel < span && g < n_in_valid)
                         ? in[g] : make_float2(0.f, 0.f);
        }
    };
    auto write_half = [&](const float2 (&stg)[NL2]) {
#pragma unroll
        for (int j = 0; j < NL2; j++) {
            unsigned idx = (unsigned)(tid + j * MDFIR_BLOCK);
            unsigned i = idx >> 1, vloc = idx & 1u;
            if (i < elemsP) {
                unsigned d = (i & 3u) * SUB + (i >> 2);
                planes[vloc * SPm + d] = stg[j].x;
                planes[(2 + vloc) * SPm + d] = stg[j].y;
            }
        }
    };
    /* lane (r16,k4)'s K-walk in a sub-plane: sub = k4, dword offset
     * wave*64 + 4*r16 + s — stride-1 in s, 16 B aligned at s%4==0 */
    const unsigned abase = (unsigned)wave * 64 + 4u * r16;
    const unsigned asub = (unsigned)k4 * SUB;
    /* b128 A-reads: one ds_read_b128 pair feeds 8 MFMAs (4 K-steps x
     * re/im), cutting the per-MFMA issue overhead (round-1 b32 reads:
     * SQ_WAIT_INST_ANY-heavy, profiles/pmc_sq_r02.txt). One accumulator
     * pair, re/im alternating: same-C spacing = 2 MFMA issues = 64
     * cyc/SIMD >= the 40-cyc dependent latency, so extra accumulator
     * pairs only cost registers (measured: dual pairs were ~3% slower,
     * forcing 8 waves/SIMD via VGPR caps 2x slower from spills). */
    auto mfma_half = [&](int h, v4f& cre, v4f& cim) {
#pragma unroll
        for (int vloc = 0; vloc < 2; vloc++) {
            const float* pre = planes + (unsigned)vloc * SPm + asub;
            const float* pim = planes + (unsigned)(2 + vloc) * SPm + asub;
            const int v = 2 * h + vloc;
            float bfrag[KKD / 4];
#pragma unroll
            for (int s = 0; s < KKD / 4; s++)
                bfrag[s] = s_rtx[v * (KKD + 16) + 15 + 4 * s + k4 - r16];
            __builtin_amdgcn_s_setprio(1); /* boost MFMA waves over the
                                              FFT/staging phases of
                                              co-resident blocks (T5) */
#pragma unroll
            for (int t = 0; t < (KKD / 4) / 4; t++) {
                float4 ar = *(const float4*)&pre[abase + 4 * t];
                float4 ai = *(const float4*)&pim[abase + 4 * t];
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.x, bfrag[4 * t], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.x, bfrag[4 * t], cim, 0, 0, 0);
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.y, bfrag[4 * t + 1], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.y, bfrag[4 * t + 1], cim, 0, 0, 0);
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.z, bfrag[4 * t + 2], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.z, bfrag[4 * t + 2], cim, 0, 0, 0);
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.w, bfrag[4 * t + 3], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.w, bfrag[4 * t + 3], cim, 0, 0, 0);
            }
#pragma unroll
            for (int s = (KKD / 4) & ~3; s < KKD / 4; s++) {
                float a_re = pre[abase + s];
                float a_im = pim[abase + s];
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    a_re, bfrag[s], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    a_im, bfrag[s], cim, 0, 0, 0);
            }
            __builtin_amdgcn_s_setprio(0);
        }
    };

    load_half(blockIdx.x, 0, stgA);
    for (long long tile = blockIdx.x;
         tile * (long long)MDFIR_TILE < n_out; tile += gridDim.x) {
        const long long out_base = tile * MDFIR_TILE;
        v4f cre = {0.f, 0.f, 0.f, 0.f};
        v4f cim = {0.f, 0.f, 0.f, 0.f};
        write_half(stgA);
        __syncthreads();
        load_half(tile, 1, stgB);     /* in flight under half-0 MFMAs */
        mfma_half(0, cre, cim);
        __syncthreads();
        write_half(stgB);
        __syncthreads();
        if ((tile + gridDim.x) * (long long)MDFIR_TILE < n_out)
            load_half(tile + gridDim.x, 0, stgA); /* under half-1 MFMAs */
        mfma_half(1, cre, cim);
#pragma unroll
        for (int q = 0; q < 4; q++) {
            int row = k4 * 4 + q;
            long long o = out_base + (long long)wave * 256 + 16 * row + r16;
            if (o < n_out)
                out[o] = make_float2(cre[q], cim[q]);
        }
        __syncthreads();
    }
}

template <int KKD, int MINWG = 1>
__global__ __launch_bounds__(MDFIR_BLOCK, MINWG) void k_decim4_fft_mfma_tpl(
    const float2* __restrict__ in, float2* __restrict__ out,
    const float* __restrict__ rtv /* [4][KKD], rtv[v][u] = rt[4u+v] */,
    long long n_out, long long n_in_valid,
    const float2* __restrict__ twid /* fft_len-entry forward table */,
    float* __restrict__ mag_out /* nullable |X|^2 */,
    int fft_len /* pow2 64..1024; tile = 1024/fft_len frames */,
    int stagger /* s_sleep(7) loops per co-residency slot: desyncs the
                   per-tile phases of co-resident blocks so one block's
                   staging/FFT interval overlaps the others' MFMA runs
                   (the bare MFMA loop reaches 97% of peak —
                   tools/mfma_ubench.py — so pipe idle = phase lock) */) {
    static_assert(MDFIR_TILE == 1024, "tile == 1024 decimated outputs");
    static_assert(KKD % 4 == 0, "KKD must be a multiple of 4");
    if (stagger > 0) {
        int loops = (int)(blockIdx.x & 7u) * stagger;
        for (int i = 0; i < loops; i++) __builtin_amdgcn_s_sleep(7);
    }
    const unsigned elemsP = MDFIR_TILE + KKD + 8;     /* per phase plane */
    const unsigned SPm = (elemsP + 31u) & ~31u;
    /* each phase plane is split into 4 sub-planes by element%4 so a
     * lane's MFMA K-walk (idx = ab+4s) becomes stride-1: one
     * ds_read_b128 feeds 4 K-steps (the b32-per-MFMA A-reads were the
     * round-1 issue-stall, SQ_WAIT_INST_ANY 46% — profiles/pmc_r02) */
    const unsigned SUB = SPm / 4;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    /* two phases resident at a time ([re_v0, re_v1, im_v0, im_v1]) —
     * halves LDS vs all-phase planes, doubling resident blocks/CU;
     * accumulators carry across the two halves, and each half's global
     * loads are issued under the other half's MFMAs. */
    float* planes = (float*)smem;        /* [4][SPm] */
    float* s_rtx = planes + 4u * SPm;    /* [4][KKD+16], 15-zero prologue */

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int r16 = lane & 15;
    const int k4 = lane >> 4;

    for (int i = tid; i < 4 * (KKD + 16); i += MDFIR_BLOCK) {
        int v = i / (KKD + 16), t = i % (KKD + 16);
        s_rtx[i] = (t >= 15 && t < 15 + KKD) ? rtv[v * KKD + (t - 15)] : 0.f;
    }
    __syncthreads();

    const unsigned span = 3 + 4 * elemsP; /* input elements per tile */
    constexpr int NL2 =
        (2 * (MDFIR_TILE + KKD + 8) + 3 + MDFIR_BLOCK - 1) / MDFIR_BLOCK;
    float2 stgA[NL2], stgB[NL2];
    /* elements of phases {2h, 2h+1}: rel = 3 + 4i + 2h + vloc */
    auto load_half = [&](long long tl, int h, float2 (&stg)[NL2]) {
        const long long ib = tl * MDFIR_TILE * 4;
#pragma unroll
        for (int j = 0; j < NL2; j++) {
            unsigned idx = (unsigned)(tid + j * MDFIR_BLOCK);
            unsigned rel = 3 + 4 * (idx >> 1) + 2 * h + (idx & 1u);
            long long g = ib + rel;
            stg[j] = (rel < span && g < n_in_valid)
                         ? in[g] : make_float2(0.f, 0.f);
        }
    };
    auto write_half = [&](const float2 (&stg)[NL2]) {
#pragma unroll
        for (int j = 0; j < NL2; j++) {
            unsigned idx = (unsigned)(tid + j * MDFIR_BLOCK);
            unsigned i = idx >> 1, vloc = idx & 1u;
            if (i < elemsP) {
                unsigned d = (i & 3u) * SUB + (i >> 2);
                planes[vloc * SPm + d] = stg[j].x;
                planes[(2 + vloc) * SPm + d] = stg[j].y;
            }
        }
    };
    /* lane (r16,k4)'s K-walk in a sub-plane: sub = k4, dword offset
     * wave*64 + 4*r16 + s — stride-1 in s, 16 B aligned at s%4==0 */
    const unsigned abase = (unsigned)wave * 64 + 4u * r16;
    const unsigned asub = (unsigned)k4 * SUB;
    /* b128 A-reads: one ds_read_b128 pair feeds 8 MFMAs (4 K-steps x
     * re/im), cutting the per-MFMA issue overhead (round-1 b32 reads:
     * SQ_WAIT_INST_ANY-heavy, profiles/pmc_sq_r02.txt). One accumulator
     * pair, re/im alternating: same-C spacing = 2 MFMA issues = 64
     * cyc/SIMD >= the 40-cyc dependent latency, so extra accumulator
     * pairs only cost registers (measured: dual pairs were ~3% slower,
     * forcing 8 waves/SIMD via VGPR caps 2x slower from spills). */
    auto mfma_half = [&](int h, v4f& cre, v4f& cim) {
#pragma unroll
        for (int vloc = 0; vloc < 2; vloc++) {
            const float* pre = planes + (unsigned)vloc * SPm + asub;
            const float* pim = planes + (unsigned)(2 + vloc) * SPm + asub;
            const int v = 2 * h + vloc;
            float bfrag[KKD / 4];
#pragma unroll
            for (int s = 0; s < KKD / 4; s++)
                bfrag[s] = s_rtx[v * (KKD + 16) + 15 + 4 * s + k4 - r16];
            __builtin_amdgcn_s_setprio(1); /* boost MFMA waves over the
                                              FFT/staging phases of
                                              co-resident blocks (T5) */
#pragma unroll
            for (int t = 0; t < (KKD / 4) / 4; t++) {
                float4 ar = *(const float4*)&pre[abase + 4 * t];
                float4 ai = *(const float4*)&pim[abase + 4 * t];
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.x, bfrag[4 * t], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.x, bfrag[4 * t], cim, 0, 0, 0);
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.y, bfrag[4 * t + 1], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.y, bfrag[4 * t + 1], cim, 0, 0, 0);
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.z, bfrag[4 * t + 2], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.z, bfrag[4 * t + 2], cim, 0, 0, 0);
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.w, bfrag[4 * t + 3], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.w, bfrag[4 * t + 3], cim, 0, 0, 0);
            }
#pragma unroll
            for (int s = (KKD / 4) & ~3; s < KKD / 4; s++) {
                float a_re = pre[abase + s];
                float a_im = pim[abase + s];
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    a_re, bfrag[s], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    a_im, bfrag[s], cim, 0, 0, 0);
            }
            __builtin_amdgcn_s_setprio(0);
        }
    };

    load_half(blockIdx.x, 0, stgA);
    for (long long tile = blockIdx.x;
         tile * (long long)MDFIR_TILE < n_out; tile += gridDim.x) {
        const long long out_base = tile * MDFIR_TILE;
        v4f cre = {0.f, 0.f, 0.f, 0.f};
        v4f cim = {0.f, 0.f, 0.f, 0.f};
        write_half(stgA);
        __syncthreads();
        load_half(tile, 1, stgB);     /* in flight under half-0 MFMAs */
        mfma_half(0, cre, cim);
        __syncthreads();
        write_half(stgB);
        __syncthreads();
        if ((tile + gridDim.x) * (long long)MDFIR_TILE < n_out)
            load_half(tile + gridDim.x, 0, stgA); /* under half-1 MFMAs */
        mfma_half(1, cre, cim);
        __syncthreads(); /* phase planes are dead; reuse them as FFT LDS */
        float2* ping = (float2*)planes;       /* 1024 float2 = 8 KB */
        float2* pong = ping + 1024;           /* fits in 4*SPm floats */
        const unsigned Lm = (unsigned)fft_len - 1u;
#pragma unroll
        for (int q = 0; q < 4; q++) {
            int row = k4 * 4 + q;
            unsigned pos = wave * 256 + 16 * row + r16; /* y2 idx in tile */
            ping[(pos & ~Lm) | fft_swz(pos & Lm)] =
                make_float2(cre[q], cim[q]);
        }
        __syncthreads();
        { /* one FFT per frame; frames share the block in lockstep */
            const int F = 1024 / fft_len;
            const int tpf = MDFIR_BLOCK / F;
            const int fl = tid / tpf, tfr = tid - fl * tpf;
            float2* res = fft_pow2_fwd(ping + (size_t)fl * fft_len,
                                       pong + (size_t)fl * fft_len, twid,
                                       fft_len, tfr, tpf) -
                          (size_t)fl * fft_len; /* uniform base */
            for (int i = tid; i < 1024; i += MDFIR_BLOCK) {
                long long o = out_base + i;
                if (o < n_out) {
                    float2 v =
                        res[((unsigned)i & ~Lm) | fft_swz((unsigned)i & Lm)];
                    if (out) out[o] = v; /* null = NullSink'd spectra */
                    if (mag_out) mag_out[o] = v.x * v.x + v.y * v.y;
                }
            }
        }
        __syncthreads();
    }
}

/* All-phase staging variant (FSDR_CHAIN_ALLPHASE=1): all 4 phase planes
 * resident at once (8 planes, ~36 KB LDS -> 4 blocks/CU instead of 6),
 * ONE staging write + barrier per tile and a contiguous 160-MFMA run —
 * trades occupancy for fewer barrier convoys. Experiment for the
 * phase-structure bound documented in profiles/pmc_sq_r02.txt. */
template <int KKD>
__global__ __launch_bounds__(MDFIR_BLOCK) void k_decim4_fft_mfma_ap_tpl(
    const float2* __restrict__ in, float2* __restrict__ out,
    const float* __restrict__ rtv /* [4][KKD], rtv[v][u] = rt[4u+v] */,
    long long n_out, long long n_in_valid,
    const float2* __restrict__ twid /* fft_len-entry forward table */,
    float* __restrict__ mag_out /* nullable |X|^2 */,
    int fft_len) {
    static_assert(KKD % 4 == 0, "KKD must be a multiple of 4");
    const unsigned elemsP = MDFIR_TILE + KKD + 8;
    const unsigned SPm = (elemsP + 31u) & ~31u;
    const unsigned SUB = SPm / 4;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* planes = (float*)smem;        /* [8][SPm]: re v0..3, im v0..3 */
    float* s_rtx = planes + 8u * SPm;    /* [4][KKD+16] */

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int r16 = lane & 15;
    const int k4 = lane >> 4;

    for (int i = tid; i < 4 * (KKD + 16); i += MDFIR_BLOCK) {
        int v = i / (KKD + 16), t = i % (KKD + 16);
        s_rtx[i] = (t >= 15 && t < 15 + KKD) ? rtv[v * KKD + (t - 15)] : 0.f;
    }
    __syncthreads();

    /* Staging by ALIGNED 4-sample groups: group q = x[4q..4q+3] = two
     * 16B float4 loads (x is 32B-aligned at 4q when the base pointer is
     * 16B-aligned — host guards). Sample x[4q+w] belongs to phase
     * v=(w+1)&3 at position q (w==3) or q-1 (w<3), since
     * P_v[i] = x[3+v+4i]. Halves the VMEM instruction count vs float2
     * loads — the incremental ubench showed the staging loads cost ~25
     * points of MFMA-pipe utilization (issue/latency interference). */
    constexpr unsigned GROUPS = (MDFIR_TILE + KKD + 8) + 1;
    constexpr int NG = (GROUPS + MDFIR_BLOCK - 1) / MDFIR_BLOCK;
    float4 stqa[NG], stqb[NG];
    auto load_all = [&](long long tl) {
        const long long qb = tl * MDFIR_TILE;
#pragma unroll
        for (int j = 0; j < NG; j++) {
            unsigned q = (unsigned)(tid + j * MDFIR_BLOCK);
            float4 a = make_float4(0.f, 0.f, 0.f, 0.f);
            float4 b = make_float4(0.f, 0.f, 0.f, 0.f);
            if (q < GROUPS) {
                long long xq = (qb + q) * 4;
                if (xq + 3 < n_in_valid) {
                    a = *(const float4*)&in[xq];
                    b = *(const float4*)&in[xq + 2];
                } else {
                    float2 e0 = (xq < n_in_valid) ? in[xq]
                                                  : make_float2(0.f, 0.f);
                    float2 e1 = (xq + 1 < n_in_valid)
                                    ? in[xq + 1] : make_float2(0.f, 0.f);
                    float2 e2 = (xq + 2 < n_in_valid)
                                    ? in[xq + 2] : make_float2(0.f, 0.f);
                    float2 e3 = (xq + 3 < n_in_valid)
                                    ? in[xq + 3] : make_float2(0.f, 0.f);
                    a = make_float4(e0.x, e0.y, e1.x, e1.y);
                    b = make_float4(e2.x, e2.y, e3.x, e3.y);
                }
            }
            stqa[j] = a;
            stqb[j] = b;
        }
    };
    auto write_all = [&]() {
#pragma unroll
        for (int j = 0; j < NG; j++) {
            unsigned q = (unsigned)(tid + j * MDFIR_BLOCK);
            if (q >= GROUPS) continue;
            float4 a = stqa[j], b = stqb[j];
            if (q >= 1) { /* w=0,1,2 -> phases 1,2,3 at position q-1 */
                unsigned d1 = ((q - 1) & 3u) * SUB + ((q - 1) >> 2);
                planes[1u * SPm + d1] = a.x;
                planes[5u * SPm + d1] = a.y;
                planes[2u * SPm + d1] = a.z;
                planes[6u * SPm + d1] = a.w;
                planes[3u * SPm + d1] = b.x;
                planes[7u * SPm + d1] = b.y;
            }
            if (q < elemsP) { /* w=3 -> phase 0 at position q */
                unsigned d0 = (q & 3u) * SUB + (q >> 2);
                planes[0u * SPm + d0] = b.z;
                planes[4u * SPm + d0] = b.w;
            }
        }
    };
    const unsigned abase = (unsigned)wave * 64 + 4u * r16;
    const unsigned asub = (unsigned)k4 * SUB;

    load_all(blockIdx.x);
    for (long long tile = blockIdx.x;
         tile * (long long)MDFIR_TILE < n_out; tile += gridDim.x) {
        const long long out_base = tile * MDFIR_TILE;
        v4f cre = {0.f, 0.f, 0.f, 0.f};
        v4f cim = {0.f, 0.f, 0.f, 0.f};
        write_all();
        __syncthreads();
        if ((tile + gridDim.x) * (long long)MDFIR_TILE < n_out)
            load_all(tile + gridDim.x); /* under this tile's MFMAs */
#pragma unroll
        for (int v = 0; v < 4; v++) {
            const float* pre = planes + (unsigned)v * SPm + asub;
            const float* pim = planes + (4u + v) * SPm + asub;
            float bfrag[KKD / 4];
#pragma unroll
            for (int s = 0; s < KKD / 4; s++)
                bfrag[s] = s_rtx[v * (KKD + 16) + 15 + 4 * s + k4 - r16];
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int t = 0; t < (KKD / 4) / 4; t++) {
                float4 ar = *(const float4*)&pre[abase + 4 * t];
                float4 ai = *(const float4*)&pim[abase + 4 * t];
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.x, bfrag[4 * t], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.x, bfrag[4 * t], cim, 0, 0, 0);
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.y, bfrag[4 * t + 1], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.y, bfrag[4 * t + 1], cim, 0, 0, 0);
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.z, bfrag[4 * t + 2], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.z, bfrag[4 * t + 2], cim, 0, 0, 0);
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.w, bfrag[4 * t + 3], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.w, bfrag[4 * t + 3], cim, 0, 0, 0);
            }
#pragma unroll
            for (int s = (KKD / 4) & ~3; s < KKD / 4; s++) {
                float a_re = pre[abase + s];
                float a_im = pim[abase + s];
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    a_re, bfrag[s], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    a_im, bfrag[s], cim, 0, 0, 0);
            }
            __builtin_amdgcn_s_setprio(0);
        }
        __syncthreads(); /* planes dead; reuse as FFT LDS */
        float2* ping = (float2*)planes;
        float2* pong = ping + 1024;
        const unsigned Lm = (unsigned)fft_len - 1u;
#pragma unroll
        for (int q = 0; q < 4; q++) {
            int row = k4 * 4 + q;
            unsigned pos = wave * 256 + 16 * row + r16;
            ping[(pos & ~Lm) | fft_swz(pos & Lm)] =
                make_float2(cre[q], cim[q]);
        }
        __syncthreads();
        {
            const int F = 1024 / fft_len;
            const int tpf = MDFIR_BLOCK / F;
            const int fl = tid / tpf, tfr = tid - fl * tpf;
            float2* res = fft_pow2_fwd(ping + (size_t)fl * fft_len,
                                       pong + (size_t)fl * fft_len, twid,
                                       fft_len, tfr, tpf) -
                          (size_t)fl * fft_len;
            for (int i = tid; i < 1024; i += MDFIR_BLOCK) {
                long long o = out_base + i;
                if (o < n_out) {
                    float2 v =
                        res[((unsigned)i & ~Lm) | fft_swz((unsigned)i & Lm)];
                    if (out) out[o] = v;
                    if (mag_out) mag_out[o] = v.x * v.x + v.y * v.y;
                }
            }
        }
        __syncthreads();
    }
}

/* Software-pipelined variant (FSDR_CHAIN_WS=1, fft_len==1024 only):
 * tile t's FFT stages are interleaved between tile t+1's MFMA vloc
 * groups, so the FFT's VALU/LDS work issues inside the MFMA pipe's
 * 32-cyc/SIMD instruction shadow instead of serializing after it (the
 * incremental ubench priced the serial FFT at ~1800 CU-cyc/tile). The
 * FFT runs IN-PLACE (DIF radix-4, digit-reversed output read) in a
 * dedicated 8 KB LDS strip so staging planes stay resident; ~45.6 KB
 * LDS -> 3 blocks/CU (vs the ap kernel's 4). */
__device__ __forceinline__ unsigned rev4_10(unsigned i) {
    /* reverse 5 base-4 digits of a 10-bit index */
    unsigned r = 0;
#pragma unroll
    for (int d = 0; d < 5; d++) {
        r = (r << 2) | (i & 3u);
        i >>= 2;
    }
    return r;
}

template <int KKD>
__global__ __launch_bounds__(MDFIR_BLOCK) void k_decim4_fft_mfma_ws_tpl(
    const float2* __restrict__ in, float2* __restrict__ out,
    const float* __restrict__ rtv, long long n_out, long long n_in_valid,
    const float2* __restrict__ twid /* 1024-entry forward table */,
    float* __restrict__ mag_out) {
    static_assert(KKD % 4 == 0, "KKD must be a multiple of 4");
    const unsigned elemsP = MDFIR_TILE + KKD + 8;
    const unsigned SPm = (elemsP + 31u) & ~31u;
    const unsigned SUB = SPm / 4;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* planes = (float*)smem;            /* [8][SPm] */
    float* s_rtx = planes + 8u * SPm;        /* [4][KKD+16] */
    float2* fbuf = (float2*)(s_rtx + 4 * (KKD + 16)); /* 1024 f2, in-place */

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int r16 = lane & 15;
    const int k4 = lane >> 4;

    for (int i = tid; i < 4 * (KKD + 16); i += MDFIR_BLOCK) {
        int v = i / (KKD + 16), t = i % (KKD + 16);
        s_rtx[i] = (t >= 15 && t < 15 + KKD) ? rtv[v * KKD + (t - 15)] : 0.f;
    }
    __syncthreads();

    constexpr unsigned GROUPS = (MDFIR_TILE + KKD + 8) + 1;
    constexpr int NG = (GROUPS + MDFIR_BLOCK - 1) / MDFIR_BLOCK;
    float4 stqa[NG], stqb[NG];
    auto load_all = [&](long long tl) {
        const long long qb = tl * MDFIR_TILE;
#pragma unroll
        for (int j = 0; j < NG; j++) {
            unsigned q = (unsigned)(tid + j * MDFIR_BLOCK);
            float4 a = make_float4(0.f, 0.f, 0.f, 0.f);
            float4 b = make_float4(0.f, 0.f, 0.f, 0.f);
            if (q < GROUPS) {
                long long xq = (qb + q) * 4;
                if (xq + 3 < n_in_valid) {
                    a = *(const float4*)&in[xq];
                    b = *(const float4*)&in[xq + 2];
                } else {
                    float2 e0 = (xq < n_in_valid) ? in[xq]
                                                  : make_float2(0.f, 0.f);
                    float2 e1 = (xq + 1 < n_in_valid)
                                    ? in[xq + 1] : make_float2(0.f, 0.f);
                    float2 e2 = (xq + 2 < n_in_valid)
                                    ? in[xq + 2] : make_float2(0.f, 0.f);
                    float2 e3 = (xq + 3 < n_in_valid)
                                    ? in[xq + 3] : make_float2(0.f, 0.f);
                    a = make_float4(e0.x, e0.y, e1.x, e1.y);
                    b = make_float4(e2.x, e2.y, e3.x, e3.y);
                }
            }
            stqa[j] = a;
            stqb[j] = b;
        }
    };
    auto write_all = [&]() {
#pragma unroll
        for (int j = 0; j < NG; j++) {
            unsigned q = (unsigned)(tid + j * MDFIR_BLOCK);
            if (q >= GROUPS) continue;
            float4 a = stqa[j], b = stqb[j];
            if (q >= 1) {
                unsigned d1 = ((q - 1) & 3u) * SUB + ((q - 1) >> 2);
                planes[1u * SPm + d1] = a.x;
                planes[5u * SPm + d1] = a.y;
                planes[2u * SPm + d1] = a.z;
                planes[6u * SPm + d1] = a.w;
                planes[3u * SPm + d1] = b.x;
                planes[7u * SPm + d1] = b.y;
            }
            if (q < elemsP) {
                unsigned d0 = (q & 3u) * SUB + (q >> 2);
                planes[0u * SPm + d0] = b.z;
                planes[4u * SPm + d0] = b.w;
            }
        }
    };
    const unsigned abase = (unsigned)wave * 64 + 4u * r16;
    const unsigned asub = (unsigned)k4 * SUB;
    /* g0/g1 select a sub-range of the vloc's b128 K-groups so the last
     * vloc can be split across two FFT-stage windows (tail = run the
     * non-multiple-of-4 K remainder too) */
    auto mfma_vloc = [&](int v, v4f& cre, v4f& cim, int g0, int g1,
                         bool tail) {
        const float* pre = planes + (unsigned)v * SPm + asub;
        const float* pim = planes + (4u + v) * SPm + asub;
        float bfrag[KKD / 4];
#pragma unroll
        for (int s = 0; s < KKD / 4; s++)
            bfrag[s] = s_rtx[v * (KKD + 16) + 15 + 4 * s + k4 - r16];
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int t = 0; t < (KKD / 4) / 4; t++) {
            if (t < g0 || t >= g1) continue;
            float4 ar = *(const float4*)&pre[abase + 4 * t];
            float4 ai = *(const float4*)&pim[abase + 4 * t];
            cre = __builtin_amdgcn_mfma_f32_16x16x4f32(ar.x, bfrag[4 * t],
                                                       cre, 0, 0, 0);
            cim = __builtin_amdgcn_mfma_f32_16x16x4f32(ai.x, bfrag[4 * t],
                                                       cim, 0, 0, 0);
            cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                ar.y, bfrag[4 * t + 1], cre, 0, 0, 0);
            cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                ai.y, bfrag[4 * t + 1], cim, 0, 0, 0);
            cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                ar.z, bfrag[4 * t + 2], cre, 0, 0, 0);
            cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                ai.z, bfrag[4 * t + 2], cim, 0, 0, 0);
            cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                ar.w, bfrag[4 * t + 3], cre, 0, 0, 0);
            cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                ai.w, bfrag[4 * t + 3], cim, 0, 0, 0);
        }
        if (tail) {
#pragma unroll
            for (int s = (KKD / 4) & ~3; s < KKD / 4; s++) {
                float a_re = pre[abase + s];
                float a_im = pim[abase + s];
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    a_re, bfrag[s], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    a_im, bfrag[s], cim, 0, 0, 0);
            }
        }
        __builtin_amdgcn_s_setprio(0);
    };
    /* one in-place DIF radix-4 stage: 256 butterflies, 1 per thread.
     * Stage s: sub-length L = 1024>>(2s), quarter m = L/4; y_r stored
     * back at q + r*m with twiddles W_L^{rq} (derived from one load). */
    auto fft_stage = [&](int s) {
        const unsigned L = 1024u >> (2 * s);
        const unsigned m = L >> 2;
        unsigned bf = (unsigned)tid;
        unsigned blk = bf / m, q = bf - blk * m;
        unsigned base = blk * L + q;
        float2 x0 = fbuf[fft_swz(base)];
        float2 x1 = fbuf[fft_swz(base + m)];
        float2 x2 = fbuf[fft_swz(base + 2 * m)];
        float2 x3 = fbuf[fft_swz(base + 3 * m)];
        float2 t0 = f2_add(x0, x2), t1 = f2_sub(x0, x2);
        float2 t2 = f2_add(x1, x3), t3 = f2_sub(x1, x3);
        float2 t3r = make_float2(t3.y, -t3.x); /* -i * t3 (forward) */
        float2 w1 = twid[(size_t)q << (2 * s)];
        float2 w2 = cmulf(w1, w1);
        float2 w3 = cmulf(w2, w1);
        fbuf[fft_swz(base)] = f2_add(t0, t2);
        fbuf[fft_swz(base + m)] = cmulf(f2_add(t1, t3r), w1);
        fbuf[fft_swz(base + 2 * m)] = cmulf(f2_sub(t0, t2), w2);
        fbuf[fft_swz(base + 3 * m)] = cmulf(f2_sub(t1, t3r), w3);
    };

    v4f cre = {0.f, 0.f, 0.f, 0.f};
    v4f cim = {0.f, 0.f, 0.f, 0.f};
    bool have_prev = false;
    long long prev_base = 0;
    load_all(blockIdx.x);
    for (long long tile = blockIdx.x;
         tile * (long long)MDFIR_TILE < n_out; tile += gridDim.x) {
        write_all();
        if (have_prev) { /* deposit C(prev) into the FFT strip */
#pragma unroll
            for (int q = 0; q < 4; q++) {
                int row = k4 * 4 + q;
                unsigned pos = wave * 256 + 16 * row + r16;
                fbuf[fft_swz(pos)] = make_float2(cre[q], cim[q]);
            }
        }
        __syncthreads();
        if ((tile + gridDim.x) * (long long)MDFIR_TILE < n_out)
            load_all(tile + gridDim.x);
        cre = (v4f){0.f, 0.f, 0.f, 0.f};
        cim = (v4f){0.f, 0.f, 0.f, 0.f};
        constexpr int G = (KKD / 4) / 4;
        constexpr int GH = G / 2;
        mfma_vloc(0, cre, cim, 0, G, true);
        if (have_prev) fft_stage(0);
        __syncthreads();
        mfma_vloc(1, cre, cim, 0, G, true);
        if (have_prev) fft_stage(1);
        __syncthreads();
        mfma_vloc(2, cre, cim, 0, G, true);
        if (have_prev) fft_stage(2);
        __syncthreads();
        mfma_vloc(3, cre, cim, 0, GH, false);
        if (have_prev) fft_stage(3);
        __syncthreads();
        mfma_vloc(3, cre, cim, GH, G, true);
        if (have_prev) fft_stage(4);
        __syncthreads();
        if (have_prev) {
            for (int i = tid; i < 1024; i += MDFIR_BLOCK) {
                long long o = prev_base + i;
                if (o < n_out) {
                    float2 v = fbuf[fft_swz(rev4_10((unsigned)i))];
                    if (out) out[o] = v;
                    if (mag_out) mag_out[o] = v.x * v.x + v.y * v.y;
                }
            }
        }
        prev_base = tile * MDFIR_TILE;
        have_prev = true;
        __syncthreads();
    }
    /* epilogue: FFT + output of the final tile */
    if (have_prev) {
#pragma unroll
        for (int q = 0; q < 4; q++) {
            int row = k4 * 4 + q;
            unsigned pos = wave * 256 + 16 * row + r16;
            fbuf[fft_swz(pos)] = make_float2(cre[q], cim[q]);
        }
        __syncthreads();
        for (int s = 0; s < 5; s++) {
            fft_stage(s);
            __syncthreads();
        }
        for (int i = tid; i < 1024; i += MDFIR_BLOCK) {
            long long o = prev_base + i;
            if (o < n_out) {
                float2 v = fbuf[fft_swz(rev4_10((unsigned)i))];
                if (out) out[o] = v;
                if (mag_out) mag_out[o] = v.x * v.x + v.y * v.y;
            }
        }
    }
}

template <int KKD>
__global__ __launch_bounds__(512) void k_decim4_fft_mfma2_tpl(
    const float2* __restrict__ in, float2* __restrict__ out,
    const float* __restrict__ rtv /* [4][KKD], rtv[v][u] = rt[4u+v] */,
    long long n_out, long long n_in_valid,
    const float2* __restrict__ twid /* 1024-entry forward table */,
    float* __restrict__ mag_out /* nullable |X|^2 */) {
    static_assert(true, "two 1024-pt FFT frames per tile");
    static_assert(KKD % 4 == 0, "KKD must be a multiple of 4");
    const unsigned elemsP = 2048 + KKD + 8;     /* per phase plane */
    const unsigned SPm = (elemsP + 31u) & ~31u;
    /* each phase plane is split into 4 sub-planes by element%4 so a
     * lane's MFMA K-walk (idx = ab+4s) becomes stride-1: one
     * ds_read_b128 feeds 4 K-steps (the b32-per-MFMA A-reads were the
     * round-1 issue-stall, SQ_WAIT_INST_ANY 46% — profiles/pmc_r02) */
    const unsigned SUB = SPm / 4;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    /* two phases resident at a time ([re_v0, re_v1, im_v0, im_v1]) —
     * halves LDS vs all-phase planes, doubling resident blocks/CU;
     * accumulators carry across the two halves, and each half's global
     * loads are issued under the other half's MFMAs. */
    float* planes = (float*)smem;        /* [4][SPm] */
    float* s_rtx = planes + 4u * SPm;    /* [4][KKD+16], 15-zero prologue */

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int r16 = lane & 15;
    const int k4 = lane >> 4;

    for (int i = tid; i < 4 * (KKD + 16); i += 512) {
        int v = i / (KKD + 16), t = i % (KKD + 16);
        s_rtx[i] = (t >= 15 && t < 15 + KKD) ? rtv[v * KKD + (t - 15)] : 0.f;
    }
    __syncthreads();

    const unsigned span = 3 + 4 * elemsP; /* input elements per tile */
    constexpr int NL2 =
        (2 * (2048 + KKD + 8) + 3 + 512 - 1) / 512;
    float2 stgA[NL2], stgB[NL2];
    /* elements of phases {2h, 2h+1}: rel = 3 + 4i + 2h + vloc */
    auto load_half = [&](long long tl, int h, float2 (&stg)[NL2]) {
        const long long ib = tl * 2048 * 4;
#pragma unroll
        for (int j = 0; j < NL2; j++) {
            unsigned idx = (unsigned)(tid + j * 512);
            unsigned rel = 3 + 4 * (idx >> 1) + 2 * h + (idx & 1u);
            long long g = ib + rel;
            stg[j] = (rel < span && g < n_in_valid)
                         ? in[g] : make_float2(0.f, 0.f);
        }
    };
    auto write_half = [&](const float2 (&stg)[NL2]) {
#pragma unroll
        for (int j = 0; j < NL2; j++) {
            unsigned idx = (unsigned)(tid + j * 512);
            unsigned i = idx >> 1, vloc = idx & 1u;
            if (i < elemsP) {
                unsigned d = (i & 3u) * SUB + (i >> 2);
                planes[vloc * SPm + d] = stg[j].x;
                planes[(2 + vloc) * SPm + d] = stg[j].y;
            }
        }
    };
    /* lane (r16,k4)'s K-walk in a sub-plane: sub = k4, dword offset
     * wave*64 + 4*r16 + s — stride-1 in s, 16 B aligned at s%4==0 */
    const unsigned abase = (unsigned)wave * 64 + 4u * r16;
    const unsigned asub = (unsigned)k4 * SUB;
    /* b128 A-reads: one ds_read_b128 pair feeds 8 MFMAs (4 K-steps x
     * re/im), cutting the per-MFMA issue overhead (round-1 b32 reads:
     * SQ_WAIT_INST_ANY-heavy, profiles/pmc_sq_r02.txt). One accumulator
     * pair, re/im alternating: same-C spacing = 2 MFMA issues = 64
     * cyc/SIMD >= the 40-cyc dependent latency, so extra accumulator
     * pairs only cost registers (measured: dual pairs were ~3% slower,
     * forcing 8 waves/SIMD via VGPR caps 2x slower from spills). */
    auto mfma_half = [&](int h, v4f& cre, v4f& cim) {
#pragma unroll
        for (int vloc = 0; vloc < 2; vloc++) {
            const float* pre = planes + (unsigned)vloc * SPm + asub;
            const float* pim = planes + (unsigned)(2 + vloc) * SPm + asub;
            const int v = 2 * h + vloc;
            float bfrag[KKD / 4];
#pragma unroll
            for (int s = 0; s < KKD / 4; s++)
                bfrag[s] = s_rtx[v * (KKD + 16) + 15 + 4 * s + k4 - r16];
            __builtin_amdgcn_s_setprio(1); /* boost MFMA waves over the
                                              FFT/staging phases of
                                              co-resident blocks (T5) */
#pragma unroll
            for (int t = 0; t < (KKD / 4) / 4; t++) {
                float4 ar = *(const float4*)&pre[abase + 4 * t];
                float4 ai = *(const float4*)&pim[abase + 4 * t];
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.x, bfrag[4 * t], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.x, bfrag[4 * t], cim, 0, 0, 0);
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.y, bfrag[4 * t + 1], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.y, bfrag[4 * t + 1], cim, 0, 0, 0);
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.z, bfrag[4 * t + 2], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.z, bfrag[4 * t + 2], cim, 0, 0, 0);
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.w, bfrag[4 * t + 3], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.w, bfrag[4 * t + 3], cim, 0, 0, 0);
            }
#pragma unroll
            for (int s = (KKD / 4) & ~3; s < KKD / 4; s++) {
                float a_re = pre[abase + s];
                float a_im = pim[abase + s];
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    a_re, bfrag[s], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    a_im, bfrag[s], cim, 0, 0, 0);
            }
            __builtin_amdgcn_s_setprio(0);
        }
    };

    load_half(blockIdx.x, 0, stgA);
    for (long long tile = blockIdx.x;
         tile * (long long)2048 < n_out; tile += gridDim.x) {
        const long long out_base = tile * 2048;
        v4f cre = {0.f, 0.f, 0.f, 0.f};
        v4f cim = {0.f, 0.f, 0.f, 0.f};
        write_half(stgA);
        __syncthreads();
        load_half(tile, 1, stgB);     /* in flight under half-0 MFMAs */
        mfma_half(0, cre, cim);
        __syncthreads();
        write_half(stgB);
        __syncthreads();
        if ((tile + gridDim.x) * (long long)2048 < n_out)
            load_half(tile + gridDim.x, 0, stgA); /* under half-1 MFMAs */
        mfma_half(1, cre, cim);
        __syncthreads(); /* phase planes are dead; reuse them as FFT LDS:
                            frame f ping = fbase + f*2048, pong = +1024 */
        float2* fbase = (float2*)planes; /* 4096 float2 = 32 KB */
#pragma unroll
        for (int q = 0; q < 4; q++) {
            int row = k4 * 4 + q;
            int pos = wave * 256 + 16 * row + r16; /* y2 index in tile */
            int fr = pos >> 10, idx = pos & 1023;
            fbase[fr * 2048 + fft_swz((unsigned)idx)] =
                make_float2(cre[q], cim[q]);
        }
        __syncthreads();
        {
            int fl = tid >> 8, tf = tid & 255;
            fft1024_block(fbase + fl * 2048, fbase + fl * 2048 + 1024,
                          twid, tf);
        }
        for (int i = tid; i < 2048; i += 512) {
            long long o = out_base + i;
            if (o < n_out) {
                int fr = i >> 10, idx = i & 1023;
                float2 v = fbase[fr * 2048 + 1024 + fft_swz((unsigned)idx)];
                if (out) out[o] = v; /* null = NullSink'd spectra */
                if (mag_out) mag_out[o] = v.x * v.x + v.y * v.y;
            }
        }
        __syncthreads();
    }
}

/* MFMA-loop microbenchmark (tools/mfma_ubench.py): the decim kernel's
 * inner loop on LDS staged ONCE — no per-tile staging, no FFT, no
 * barriers in the hot loop. Measures the intrinsic ceiling of the
 * b128-fed 16x16x4-f32 loop at this occupancy. */
template <int KKD>
__global__ __launch_bounds__(MDFIR_BLOCK) void k_mfma_ubench_tpl(
    const float* __restrict__ rtv, float2* __restrict__ out, int iters) {
    const unsigned elemsP = MDFIR_TILE + KKD + 8;
    const unsigned SPm = (elemsP + 31u) & ~31u;
    const unsigned SUB = SPm / 4;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* planes = (float*)smem;
    float* s_rtx = planes + 4u * SPm;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int r16 = lane & 15;
    const int k4 = lane >> 4;
    for (int i = tid; i < 4 * (KKD + 16); i += MDFIR_BLOCK) {
        int v = i / (KKD + 16), t = i % (KKD + 16);
        s_rtx[i] = (t >= 15 && t < 15 + KKD) ? rtv[v * KKD + (t - 15)] : 0.f;
    }
    for (int i = tid; i < (int)(4 * SPm); i += MDFIR_BLOCK)
        planes[i] = (float)(i & 255) * 0.001f;
    __syncthreads();
    const unsigned abase = (unsigned)wave * 64 + 4u * r16;
    const unsigned asub = (unsigned)k4 * SUB;
    v4f cre = {0.f, 0.f, 0.f, 0.f};
    v4f cim = {0.f, 0.f, 0.f, 0.f};
    for (int it = 0; it < iters; it++) {
#pragma unroll
        for (int vloc = 0; vloc < 2; vloc++) {
            const float* pre = planes + (unsigned)vloc * SPm + asub;
            const float* pim = planes + (unsigned)(2 + vloc) * SPm + asub;
            const int v = 2 * ((it ^ vloc) & 1) + vloc;
            float bfrag[KKD / 4];
#pragma unroll
            for (int s = 0; s < KKD / 4; s++)
                bfrag[s] = s_rtx[v * (KKD + 16) + 15 + 4 * s + k4 - r16];
#pragma unroll
            for (int t = 0; t < (KKD / 4) / 4; t++) {
                float4 ar = *(const float4*)&pre[abase + 4 * t];
                float4 ai = *(const float4*)&pim[abase + 4 * t];
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.x, bfrag[4 * t], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.x, bfrag[4 * t], cim, 0, 0, 0);
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.y, bfrag[4 * t + 1], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.y, bfrag[4 * t + 1], cim, 0, 0, 0);
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.z, bfrag[4 * t + 2], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.z, bfrag[4 * t + 2], cim, 0, 0, 0);
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.w, bfrag[4 * t + 3], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.w, bfrag[4 * t + 3], cim, 0, 0, 0);
            }
        }
    }
    long long o = (long long)blockIdx.x * MDFIR_BLOCK + tid;
    out[o] = make_float2(cre[0] + cre[1] + cre[2] + cre[3],
                         cim[0] + cim[1] + cim[2] + cim[3]);
}

/* Incremental chain ubench: the real tile loop with components gated by
 * MODE to isolate the integration cost (tools/mfma_ubench.py):
 *   bit0 (1): global staging loads of the real input
 *   bit1 (2): LDS write_half phases + their barriers
 *   bit2 (4): deposit + in-block FFT + output writes
 * MODE 7 == the production kernel shape; MODE 0 ~= the bare loop. */
template <int KKD, int MODE>
__global__ __launch_bounds__(MDFIR_BLOCK) void k_chain_ubench_tpl(
    const float2* __restrict__ in, float2* __restrict__ out,
    const float* __restrict__ rtv, long long n_out, long long n_in_valid,
    const float2* __restrict__ twid) {
    const unsigned elemsP = MDFIR_TILE + KKD + 8;
    const unsigned SPm = (elemsP + 31u) & ~31u;
    const unsigned SUB = SPm / 4;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* planes = (float*)smem;
    float* s_rtx = planes + 4u * SPm;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int r16 = lane & 15;
    const int k4 = lane >> 4;
    for (int i = tid; i < 4 * (KKD + 16); i += MDFIR_BLOCK) {
        int v = i / (KKD + 16), t = i % (KKD + 16);
        s_rtx[i] = (t >= 15 && t < 15 + KKD) ? rtv[v * KKD + (t - 15)] : 0.f;
    }
    for (int i = tid; i < (int)(4 * SPm); i += MDFIR_BLOCK)
        planes[i] = (float)(i & 255) * 0.001f;
    __syncthreads();
    const unsigned span = 3 + 4 * elemsP;
    constexpr int NL2 =
        (2 * (MDFIR_TILE + KKD + 8) + 3 + MDFIR_BLOCK - 1) / MDFIR_BLOCK;
    float2 stgA[NL2], stgB[NL2];
    auto load_half = [&](long long tl, int h, float2 (&stg)[NL2]) {
        if (MODE & 1) {
            const long long ib = tl * MDFIR_TILE * 4;
#pragma unroll
            for (int j = 0; j < NL2; j++) {
                unsigned idx = (unsigned)(tid + j * MDFIR_BLOCK);
                unsigned rel = 3 + 4 * (idx >> 1) + 2 * h + (idx & 1u);
                long long g = ib + rel;
                stg[j] = (rel < span && g < n_in_valid)
                             ? in[g] : make_float2(0.f, 0.f);
            }
        } else {
#pragma unroll
            for (int j = 0; j < NL2; j++)
                stg[j] = make_float2((float)(tid + j + h) * 1e-3f,
                                     (float)(tl & 63) * 1e-3f);
        }
    };
    auto write_half = [&](const float2 (&stg)[NL2]) {
        if (MODE & 2) {
#pragma unroll
            for (int j = 0; j < NL2; j++) {
                unsigned idx = (unsigned)(tid + j * MDFIR_BLOCK);
                unsigned i = idx >> 1, vloc = idx & 1u;
                if (i < elemsP) {
                    unsigned d = (i & 3u) * SUB + (i >> 2);
                    planes[vloc * SPm + d] = stg[j].x;
                    planes[(2 + vloc) * SPm + d] = stg[j].y;
                }
            }
        } else { /* keep stg live without LDS traffic */
            float acc = 0.f;
#pragma unroll
            for (int j = 0; j < NL2; j++) acc += stg[j].x;
            if (acc == 1e30f) planes[tid] = acc; /* never taken */
        }
    };
    const unsigned abase = (unsigned)wave * 64 + 4u * r16;
    const unsigned asub = (unsigned)k4 * SUB;
    auto mfma_half = [&](int h, v4f& cre, v4f& cim) {
#pragma unroll
        for (int vloc = 0; vloc < 2; vloc++) {
            const float* pre = planes + (unsigned)vloc * SPm + asub;
            const float* pim = planes + (unsigned)(2 + vloc) * SPm + asub;
            const int v = 2 * h + vloc;
            float bfrag[KKD / 4];
#pragma unroll
            for (int s = 0; s < KKD / 4; s++)
                bfrag[s] = s_rtx[v * (KKD + 16) + 15 + 4 * s + k4 - r16];
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int t = 0; t < (KKD / 4) / 4; t++) {
                float4 ar = *(const float4*)&pre[abase + 4 * t];
                float4 ai = *(const float4*)&pim[abase + 4 * t];
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.x, bfrag[4 * t], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.x, bfrag[4 * t], cim, 0, 0, 0);
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.y, bfrag[4 * t + 1], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.y, bfrag[4 * t + 1], cim, 0, 0, 0);
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.z, bfrag[4 * t + 2], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.z, bfrag[4 * t + 2], cim, 0, 0, 0);
                cre = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ar.w, bfrag[4 * t + 3], cre, 0, 0, 0);
                cim = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    ai.w, bfrag[4 * t + 3], cim, 0, 0, 0);
            }
            __builtin_amdgcn_s_setprio(0);
        }
    };
    load_half(blockIdx.x, 0, stgA);
    for (long long tile = blockIdx.x;
         tile * (long long)MDFIR_TILE < n_out; tile += gridDim.x) {
        const long long out_base = tile * MDFIR_TILE;
        v4f cre = {0.f, 0.f, 0.f, 0.f};
        v4f cim = {0.f, 0.f, 0.f, 0.f};
        write_half(stgA);
        if (MODE & 2) __syncthreads();
        load_half(tile, 1, stgB);
        mfma_half(0, cre, cim);
        if (MODE & 2) __syncthreads();
        write_half(stgB);
        if (MODE & 2) __syncthreads();
        if ((tile + gridDim.x) * (long long)MDFIR_TILE < n_out)
            load_half(tile + gridDim.x, 0, stgA);
        mfma_half(1, cre, cim);
        if (MODE & 4) {
            __syncthreads();
            float2* ping = (float2*)planes;
            float2* pong = ping + 1024;
#pragma unroll
            for (int q = 0; q < 4; q++) {
                int row = k4 * 4 + q;
                unsigned pos = wave * 256 + 16 * row + r16;
                ping[fft_swz(pos)] = make_float2(cre[q], cim[q]);
            }
            __syncthreads();
            fft1024_block(ping, pong, twid, tid);
            for (int i = tid; i < 1024; i += MDFIR_BLOCK) {
                long long o = out_base + i;
                if (o < n_out) out[o] = pong[fft_swz((unsigned)i)];
            }
            __syncthreads();
        } else {
#pragma unroll
            for (int q = 0; q < 4; q++) {
                int row = k4 * 4 + q;
                long long o =
                    out_base + (long long)wave * 256 + 16 * row + r16;
                if (o < n_out) out[o] = make_float2(cre[q], cim[q]);
            }
            if (MODE & 2) __syncthreads();
        }
    }
}

extern "C" int fsdr_chain_ubench(int mode, double* tflops, void* stream) {
    REQUIRE_GPU();
    const int KKD = 80;
    const long long n_out = 1 << 24; /* decimated outputs (2^26 inputs) */
    const long long n_in = 4 * n_out + 1024;
    unsigned elemsP = MDFIR_TILE + KKD + 8;
    unsigned SPm = (elemsP + 31u) & ~31u;
    size_t lds = (4 * (size_t)SPm + 4 * (KKD + 16)) * sizeof(float);
    float* d_taps = nullptr;
    float2* d_in = nullptr;
    float2* d_out = nullptr;
    float2* d_tw = nullptr;
    HIP_TRY(hipMalloc(&d_taps, 4 * KKD * sizeof(float)));
    HIP_TRY(hipMemset(d_taps, 1, 4 * KKD * sizeof(float)));
    HIP_TRY(hipMalloc(&d_in, n_in * sizeof(float2)));
    HIP_TRY(hipMemset(d_in, 2, n_in * sizeof(float2)));
    HIP_TRY(hipMalloc(&d_out, n_out * sizeof(float2)));
    std::vector<float2> tw(1024);
    for (int k = 0; k < 1024; k++) {
        double a = -2.0 * M_PI * k / 1024.0;
        tw[k] = make_float2((float)cos(a), (float)sin(a));
    }
    HIP_TRY(hipMalloc(&d_tw, 1024 * sizeof(float2)));
    HIP_TRY(hipMemcpy(d_tw, tw.data(), 1024 * sizeof(float2),
                      hipMemcpyHostToDevice));
    hipStream_t st = (hipStream_t)stream;
    int grid = 8192;
    hipEvent_t e0, e1;
    HIP_TRY(hipEventCreate(&e0));
    HIP_TRY(hipEventCreate(&e1));
#define UB_CASE(MV)                                                       \
    case MV:                                                              \
        hipLaunchKernelGGL(HIP_KERNEL_NAME((k_chain_ubench_tpl<80, MV>)), \
                           dim3(grid), dim3(MDFIR_BLOCK), lds, st, d_in,  \
                           d_out, d_taps, n_out, n_in, d_tw);             \
        break;
    for (int rep = 0; rep < 2; rep++) {
        if (rep == 1) HIP_TRY(hipEventRecord(e0, st));
        switch (mode) {
            UB_CASE(0)
            UB_CASE(1)
            UB_CASE(2)
            UB_CASE(3)
            UB_CASE(4)
            UB_CASE(6)
            UB_CASE(7)
            default:
                set_err("mode must be in {0,1,2,3,4,6,7}");
                return FSDR_ERR_INVALID;
        }
        HIP_TRY(hipGetLastError());
    }
    HIP_TRY(hipEventRecord(e1, st));
    HIP_TRY(hipEventSynchronize(e1));
    float ms = 0.f;
    HIP_TRY(hipEventElapsedTime(&ms, e0, e1));
    double fl = (double)n_out * 4 * KKD * 4.0; /* 4 phases x KKD x 4 */
    *tflops = fl / (ms * 1e-3) / 1e12;
    (void)hipEventDestroy(e0);
    (void)hipEventDestroy(e1);
    (void)hipFree(d_taps);
    (void)hipFree(d_in);
    (void)hipFree(d_out);
    (void)hipFree(d_tw);
    return FSDR_OK;
}

extern "C" int fsdr_mfma_ubench(int grid, int iters, double* tflops,
                                void* stream) {
    REQUIRE_GPU();
    const int KKD = 80;
    unsigned elemsP = MDFIR_TILE + KKD + 8;
    unsigned SPm = (elemsP + 31u) & ~31u;
    size_t lds = (4 * (size_t)SPm + 4 * (KKD + 16)) * sizeof(float);
    float* d_taps = nullptr;
    float2* d_out = nullptr;
    HIP_TRY(hipMalloc(&d_taps, 4 * KKD * sizeof(float)));
    HIP_TRY(hipMemset(d_taps, 1, 4 * KKD * sizeof(float)));
    HIP_TRY(hipMalloc(&d_out, (size_t)grid * MDFIR_BLOCK * sizeof(float2)));
    hipStream_t st = (hipStream_t)stream;
    hipEvent_t e0, e1;
    HIP_TRY(hipEventCreate(&e0));
    HIP_TRY(hipEventCreate(&e1));
    hipLaunchKernelGGL(HIP_KERNEL_NAME(k_mfma_ubench_tpl<KKD>), dim3(grid),
                       dim3(MDFIR_BLOCK), lds, st, d_taps, d_out, iters);
    HIP_TRY(hipEventRecord(e0, st));
    hipLaunchKernelGGL(HIP_KERNEL_NAME(k_mfma_ubench_tpl<KKD>), dim3(grid),
                       dim3(MDFIR_BLOCK), lds, st, d_taps, d_out, iters);
    HIP_TRY(hipEventRecord(e1, st));
    HIP_TRY(hipEventSynchronize(e1));
    float ms = 0.f;
    HIP_TRY(hipEventElapsedTime(&ms, e0, e1));
    /* flops: grid blocks x 4 waves x iters x 2 vloc x KKD/4 steps x 2
     * (re+im) MFMAs x 2048 flops */
    double fl = (double)grid * 4 * iters * 2 * (KKD / 4) * 2 * 2048.0;
    *tflops = fl / (ms * 1e-3) / 1e12;
    (void)hipEventDestroy(e0);
    (void)hipEventDestroy(e1);
    (void)hipFree(d_taps);
    (void)hipFree(d_out);
    return FSDR_OK;
}

/* ---- Phase-split decimating FIR, D=4, compile-time taps -------------- *
 * Same math as decimating_fir.rs:80-95 (D=4): y[k] = sum_t x[3+4k+t] *
 * h[T-1-t]. Decompose t = 4u+v: y[k] = sum_v sum_u P_v[k+u] * rt[4u+v]
 * with P_v[i] = x[3+v+4i] and rt[t] = h[T-1-t] — four stride-1 sub-FIRs
 * over de-interleaved phase planes, staged at write time (free). Each
 * sub-FIR then uses the exact k_fir_cf32_tpl structure: conflict-free
 * 16 B lane-stride ds_read_b128 groups, affine offsets, pipelined loads.
 * TPD = per-phase padded tap count (multiple of 4, 4*TPD >= T). Plane
 * stride is 8 mod 32 dwords so the de-interleaving writes are
 * bank-conflict-free. */
#define DFIRT_BLOCK 256
#define DFIRT_R 4
#define DFIRT_TILE (DFIRT_BLOCK * DFIRT_R) /* 1024 decimated outputs */

__device__ __host__ __forceinline__ unsigned dfirt_sp(int tpd) {
    return ((((unsigned)(DFIRT_TILE + tpd + 12)) + 31u) & ~31u) + 8u;
}

template <int TPD>
__global__ __launch_bounds__(DFIRT_BLOCK) void k_fir_decim4_tpl(
    const float2* __restrict__ in, float2* __restrict__ out,
    const float* __restrict__ rtv /* [4][TPD+4], rtv[v][u]=rt[4u+v] */,
    long long n_out, long long n_in_valid) {
    static_assert(TPD % 4 == 0, "TPD must be a multiple of 4");
    const unsigned SP = dfirt_sp(TPD);
    extern __shared__ __attribute__((aligned(16))) char smem[];
    /* two phases resident at a time: [re_v0, re_v1, im_v0, im_v1] — halves
     * the LDS footprint (8 -> 4 planes), doubling blocks/CU vs the
     * all-phases layout (the all-phase version was LDS-capacity-bound at
     * 4 blocks/CU). Accumulators carry across the two halves. */
    float* planes = (float*)smem;   /* [4][SP] */
    float* s_rt = planes + 4u * SP; /* [4][TPD+4] */

    const int tid = threadIdx.x;
    const unsigned span = 3 + 4 * (DFIRT_TILE + TPD + 8); /* input elems */
    for (long long tile = blockIdx.x;
         tile * (long long)DFIRT_TILE < n_out; tile += gridDim.x) {
        const long long out_base = tile * DFIRT_TILE;
        const long long in_base = out_base * 4;
        const unsigned eb = (unsigned)tid * DFIRT_R;
        float2 a01r = make_float2(0.f, 0.f), a23r = a01r;
        float2 a01i = a01r, a23i = a01r;
#pragma unroll
        for (int half = 0; half < 2; half++) {
            /* stage phases v = 2*half, 2*half+1: input elements with
             * (rel-3) % 4 in {2h, 2h+1}; each element = P_v[i] at
             * rel = 3 + 4i + v */
            for (unsigned idx = tid; 2 * idx + 3 < span;
                 idx += DFIRT_BLOCK) {
                unsigned i = idx >> 1, vloc = idx & 1u;
                unsigned rel = 3 + 4 * i + 2 * half + vloc;
                if (rel >= span) continue;
                long long g = in_base + rel;
                float2 x = (g < n_in_valid) ? in[g] : make_float2(0.f, 0.f);
                planes[vloc * SP + i] = x.x;
                planes[(2 + vloc) * SP + i] = x.y;
            }
            if (half == 0)
                for (int i = tid; i < 4 * (TPD + 4); i += DFIRT_BLOCK)
                    s_rt[i] = rtv[i];
            __syncthreads();
#pragma unroll
            for (int vloc = 0; vloc < 2; vloc++) {
                const float* pre = planes + (unsigned)vloc * SP;
                const float* pim = planes + (unsigned)(2 + vloc) * SP;
                const float* rt =
                    s_rt + (unsigned)(2 * half + vloc) * (TPD + 4);
                float4 r0 = *(const float4*)&pre[eb];
                float4 r1 = *(const float4*)&pre[eb + 4];
                float4 i0 = *(const float4*)&pim[eb];
                float4 i1 = *(const float4*)&pim[eb + 4];
                float4 hc = *(const float4*)&rt[0];
                constexpr int NG = TPD / 4;
#pragma unroll 4
                for (int m = 0; m < NG; m++) {
                    const float4 rn = *(const float4*)&pre[eb + 4 * m + 8];
                    const float4 in_ = *(const float4*)&pim[eb + 4 * m + 8];
                    const float4 h4 = *(const float4*)&rt[4 * m + 4];
                    const float wr[8] = {r0.x, r0.y, r0.z, r0.w,
                                         r1.x, r1.y, r1.z, r1.w};
                    const float wi[8] = {i0.x, i0.y, i0.z, i0.w,
                                         i1.x, i1.y, i1.z, i1.w};
                    const float ht[4] = {hc.x, hc.y, hc.z, hc.w};
#pragma unroll
                    for (int tl = 0; tl < 4; tl++) {
                        const float h = ht[tl];
                        a01r.x = fmaf(wr[tl], h, a01r.x);
                        a01r.y = fmaf(wr[tl + 1], h, a01r.y);
                        a23r.x = fmaf(wr[tl + 2], h, a23r.x);
                        a23r.y = fmaf(wr[tl + 3], h, a23r.y);
                        a01i.x = fmaf(wi[tl], h, a01i.x);
                        a01i.y = fmaf(wi[tl + 1], h, a01i.y);
                        a23i.x = fmaf(wi[tl + 2], h, a23i.x);
                        a23i.y = fmaf(wi[tl + 3], h, a23i.y);
                    }
                    r0 = r1; r1 = rn;
                    i0 = i1; i1 = in_;
                    hc = h4;
                }
            }
            __syncthreads(); /* before restaging / next tile */
        }
        const float ar[4] = {a01r.x, a01r.y, a23r.x, a23r.y};
        const float ai[4] = {a01i.x, a01i.y, a23i.x, a23i.y};
#pragma unroll
        for (int j = 0; j < DFIRT_R; j++) {
            long long o = out_base + eb + j;
            if (o < n_out) out[o] = make_float2(ar[j], ai[j]);
        }
    }
}

typedef void (*dfir_tpl_fn)(const float2*, float2*, const float*, long long,
                            long long);

/* Generic decimating FIR (any D) — correctness fallback: one output per
 * lane per iteration, direct reads through L1/L2 (no LDS staging). */
__global__ void k_fir_decim_generic_cf32(const float2* __restrict__ in,
                                         float2* __restrict__ out,
                                         const float* __restrict__ taps,
                                         int n_taps, long long decim,
                                         long long n_out,
                                         long long n_in_valid) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long k = blockIdx.x * (long long)blockDim.x + threadIdx.x;
         k < n_out; k += stride) {
        float sre = 0.f, sim = 0.f;
        long long base = decim - 1 + k * decim;
        for (int t = 0; t < n_taps; t++) {
            float2 x = (base + t < n_in_valid) ? in[base + t]
                                               : make_float2(0.f, 0.f);
            float h = taps[n_taps - 1 - t];
            sre = fmaf(x.x, h, sre);
            sim = fmaf(x.y, h, sim);
        }
        out[k] = make_float2(sre, sim);
    }
}

/* ================= FIR cf32 x cf32 (complex taps) ===================== *
 * FirFilter<Complex32,Complex32,Complex32> — fir.rs:257-277 (stable path):
 * accum + sample*tap with the num_complex multiply. This is the WLAN
 * SyncLong correlator core (examples/wlan/src/sync_long.rs:18-50, 64
 * complex taps). Correctness-first kernel: LDS-staged tile, one output
 * per lane x4 lane-strided (stride-1 LDS reads), taps staged in LDS.
 * An MFMA/window-optimized variant is a later-round item.
 */
__global__ __launch_bounds__(256) void k_fir_ccf32(
    const float2* __restrict__ in, float2* __restrict__ out,
    const float2* __restrict__ taps, int n_taps, long long n_out,
    long long n_in_valid) {
    const unsigned TILE = 1024;
    const unsigned elems = TILE + n_taps - 1;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float2* s_x = (float2*)smem;          /* elems */
    float2* s_h = s_x + elems;            /* reversed taps */
    const int tid = threadIdx.x;
    for (long long tile = blockIdx.x; tile * (long long)TILE < n_out;
         tile += gridDim.x) {
        const long long out_base = tile * TILE;
        for (unsigned i = tid; i < elems; i += 256) {
            long long g = out_base + i;
            s_x[i] = (g < n_in_valid) ? in[g] : make_float2(0.f, 0.f);
        }
        for (int i = tid; i < n_taps; i += 256)
            s_h[i] = taps[n_taps - 1 - i]; /* reversed */
        __syncthreads();
        for (int j = 0; j < 4; j++) {
            unsigned k = tid + j * 256;
            float sre = 0.f, sim = 0.f;
            for (int t = 0; t < n_taps; t++) {
                float2 x = s_x[k + t];
                float2 h = s_h[t];
                sre = fmaf(x.x, h.x, sre);
                sre = fmaf(-x.y, h.y, sre);
                sim = fmaf(x.x, h.y, sim);
                sim = fmaf(x.y, h.x, sim);
            }
            long long o = out_base + k;
            if (o < n_out) out[o] = make_float2(sre, sim);
        }
        __syncthreads();
    }
}

/* Parallel MovingAvg fast path (used when at most one emission occurs,
 * at the end of the processed frames — e.g. the bench's spectrum sink
 * with history == frames): the EMA recurrence is chunked over frames and
 * the per-chunk EMAs composed exactly:
 *   ema(chunk_0..c) = (1-d)^len_c * ema(chunk_0..c-1) + partial_c. */
__global__ void k_moving_avg_chunks(const float* __restrict__ in,
                                    float* __restrict__ partial, int width,
                                    long long frames, int n_chunks,
                                    int chunk_frames, float decay) {
    /* width % 4 == 0 fast shape: one float4 of bins per thread */
    if ((width & 3) == 0) {
        int w4 = width >> 2;
        long long id = blockIdx.x * (long long)blockDim.x + threadIdx.x;
        long long total = (long long)n_chunks * w4;
        long long stride = (long long)gridDim.x * blockDim.x;
        for (; id < total; id += stride) {
            int c = (int)(id / w4);
            int b4 = (int)(id - (long long)c * w4);
            long long f0 = (long long)c * chunk_frames;
            long long f1 = f0 + chunk_frames;
            if (f1 > frames) f1 = frames;
            float4 p = make_float4(0.f, 0.f, 0.f, 0.f);
            const float om = 1.0f - decay;
            for (long long f = f0; f < f1; f++) {
                float4 t = *(const float4*)&in[f * width + 4 * b4];
                p.x = isfinite(t.x) ? om * p.x + decay * t.x : om * p.x;
                p.y = isfinite(t.y) ? om * p.y + decay * t.y : om * p.y;
                p.z = isfinite(t.z) ? om * p.z + decay * t.z : om * p.z;
                p.w = isfinite(t.w) ? om * p.w + decay * t.w : om * p.w;
            }
            *(float4*)&partial[(long long)c * width + 4 * b4] = p;
        }
        return;
    }
    long long id = blockIdx.x * (long long)blockDim.x + threadIdx.x;
    long long total = (long long)n_chunks * width;
    long long stride = (long long)gridDim.x * blockDim.x;
    for (; id < total; id += stride) {
        int c = (int)(id / width);
        int b = (int)(id - (long long)c * width);
        long long f0 = (long long)c * chunk_frames;
        long long f1 = f0 + chunk_frames;
        if (f1 > frames) f1 = frames;
        float p = 0.f;
        for (long long f = f0; f < f1; f++) {
            float t = in[f * width + b];
            if (isfinite(t))
                p = (1.0f - decay) * p + decay * t;
            else
                p *= 1.0f - decay;
        }
        partial[(long long)c * width + b] = p;
    }
}

__global__ void k_moving_avg_combine(const float* __restrict__ partial,
                                     float* __restrict__ avg,
                                     float* __restrict__ out /* nullable */,
                                     int width, long long frames,
                                     int n_chunks, int chunk_frames,
                                     float decay) {
    /* fully parallel closed form: result[b] = om^frames * avg[b] +
     * sum_c om^(frames - end_c) * partial[c][b]; one block per bin,
     * chunks spread over lanes, LDS tree reduction. */
    __shared__ float red[256];
    const int b = blockIdx.x;
    if (b >= width) return;
    const float om = 1.0f - decay;
    float acc = 0.f;
    for (int c = threadIdx.x; c < n_chunks; c += blockDim.x) {
        long long end = (long long)(c + 1) * chunk_frames;
        if (end > frames) end = frames;
        acc += powf(om, (float)(frames - end)) *
               partial[(long long)c * width + b];
    }
    red[threadIdx.x] = acc;
    __syncthreads();
    for (int s = 128; s > 0; s >>= 1) {
        if (threadIdx.x < s) red[threadIdx.x] += red[threadIdx.x + s];
        __syncthreads();
    }
    if (threadIdx.x == 0) {
        float a = powf(om, (float)frames) * avg[b] + red[0];
        avg[b] = a;
        if (out) out[b] = a;
    }
}

/* ================= XlatingFir ========================================= *
 * src/blocks/xlating_fir.rs: DecimatingFir with complex band-pass taps
 * (bpf[i] = e^{i*TAU*offset/fs * i} * taps[i], :79-88) fused with the
 * output Rotator (:91-94,116: phase_incr = -TAU*offset*D/fs applied
 * in-place to the produced samples). Correctness-first kernel (LDS tile,
 * lane-strided outputs); the rotator phase uses the closed form (the
 * reference iterates — tolerance note in tests). */
__global__ __launch_bounds__(256) void k_xlating_decim_ccf32(
    const float2* __restrict__ in, float2* __restrict__ out,
    const float2* __restrict__ taps /* reversed bpf taps */, int n_taps,
    long long decim, long long n_out, long long n_in_valid, double theta,
    float p0r, float p0i) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float2* s_h = (float2*)smem;
    for (int i = threadIdx.x; i < n_taps; i += blockDim.x) s_h[i] = taps[i];
    __syncthreads();
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long k = blockIdx.x * (long long)blockDim.x + threadIdx.x;
         k < n_out; k += stride) {
        float sre = 0.f, sim = 0.f;
        long long base = decim - 1 + k * decim;
        for (int t = 0; t < n_taps; t++) {
            float2 x = (base + t < n_in_valid) ? in[base + t]
                                               : make_float2(0.f, 0.f);
            float2 hh = s_h[t];
            sre = fmaf(x.x, hh.x, sre);
            sre = fmaf(-x.y, hh.y, sre);
            sim = fmaf(x.x, hh.y, sim);
            sim = fmaf(x.y, hh.x, sim);
        }
        /* f64 angle + mod-2pi reduction (see k_rotator) */
        double a = theta * (double)(k + 1);
        a -= 6.283185307179586 * floor(a * 0.15915494309189535);
        float s, cth;
        __sincosf((float)a, &s, &cth);
        float pr = cth * p0r - s * p0i;
        float pi = cth * p0i + s * p0r;
        out[k] = make_float2(sre * pr - sim * pi, sre * pi + sim * pr);
    }
}

/* LDS-tiled XlatingFir (same output math as k_xlating_decim_ccf32): a
 * tile of XT_TILE outputs stages its whole input span into padded SoA
 * planes once; complex bpf taps in LDS. Fallback to the naive kernel
 * when the span exceeds the LDS budget (host decides). */
#define XT_BLOCK 256
#define XT_R 2
#define XT_TILE (XT_BLOCK * XT_R)

__global__ __launch_bounds__(XT_BLOCK) void k_xlating_decim_tiled_ccf32(
    const float2* __restrict__ in, float2* __restrict__ out,
    const float2* __restrict__ taps /* reversed bpf taps */, int n_taps,
    int decim, long long n_out, long long n_in_valid, double theta,
    float p0r, float p0i, int span) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* s_re = (float*)smem;
    float* s_im = s_re + plane_floats((unsigned)span);
    float* s_hr = s_im + plane_floats((unsigned)span);
    float* s_hi = s_hr + n_taps;
    for (int i = threadIdx.x; i < n_taps; i += XT_BLOCK) {
        s_hr[i] = taps[i].x;
        s_hi[i] = taps[i].y;
    }
    const long long tiles = (n_out + XT_TILE - 1) / XT_TILE;
    for (long long tile = blockIdx.x; tile < tiles; tile += gridDim.x) {
        const long long out_base = tile * XT_TILE;
        const long long in_base = decim - 1 + out_base * decim;
        __syncthreads();
        for (int i = threadIdx.x; i < span; i += XT_BLOCK) {
            long long g = in_base + i;
            float2 v = (g < n_in_valid) ? in[g] : make_float2(0.f, 0.f);
            s_re[lds_pad((unsigned)i)] = v.x;
            s_im[lds_pad((unsigned)i)] = v.y;
        }
        __syncthreads();
#pragma unroll
        for (int r = 0; r < XT_R; r++) {
            long long k = out_base + threadIdx.x + r * XT_BLOCK;
            if (k >= n_out) break;
            unsigned rel = (unsigned)((k - out_base) * decim);
            float sre = 0.f, sim = 0.f;
#pragma unroll 4
            for (int t = 0; t < n_taps; t++) {
                unsigned a = lds_pad(rel + (unsigned)t);
                float xr = s_re[a], xi = s_im[a];
                float hr = s_hr[t], hi = s_hi[t];
                sre = fmaf(xr, hr, sre);
                sre = fmaf(-xi, hi, sre);
                sim = fmaf(xr, hi, sim);
                sim = fmaf(xi, hr, sim);
            }
            /* f64 angle + mod-2pi reduction (see k_rotator) */
            double a = theta * (double)(k + 1);
            a -= 6.283185307179586 * floor(a * 0.15915494309189535);
            float s, cth;
            __sincosf((float)a, &s, &cth);
            float pr = cth * p0r - s * p0i;
            float pi = cth * p0i + s * p0r;
            out[k] = make_float2(sre * pr - sim * pi, sre * pi + sim * pr);
        }
    }
}

/* ================= WLAN sync-short helpers ============================ *
 * examples/wlan/src/bin/rx.rs:73-96 autocorrelation chain pieces:
 * a*conj(b) Combine (:81) and the sliding-SUM MovingAverage
 * (moving_average.rs:65-105) with its len-1 zero prologue. */
__global__ void k_cmul_conj(const float2* __restrict__ a,
                            const float2* __restrict__ b,
                            float2* __restrict__ o, long long n) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
         i < n; i += stride) {
        float2 x = a[i], y = b[i];
        o[i] = make_float2(x.x * y.x + x.y * y.y, x.y * y.x - x.x * y.y);
    }
}

/* divide_mag Combine — rx.rs:97: out = |a| / b (cf32, f32) -> f32 */
__global__ void k_divide_mag(const float2* __restrict__ a,
                             const float* __restrict__ b,
                             float* __restrict__ o, long long n) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
         i < n; i += stride) {
        float2 x = a[i];
        o[i] = sqrtf(x.x * x.x + x.y * x.y) / b[i];
    }
}

/* one output per lane: out[i] = sum in[i..i+len) (per float lane);
 * the zero prologue is emitted by the host wrapper */
__global__ void k_moving_sum(const float* __restrict__ in,
                             float* __restrict__ out, int width, int len,
                             long long n_out_items) {
    long long stride = (long long)gridDim.x * blockDim.x;
    long long total = n_out_items * width;
    for (long long id = blockIdx.x * (long long)blockDim.x + threadIdx.x;
         id < total; id += stride) {
        long long i = id / width;
        int q = (int)(id - i * width);
        float sum = 0.f;
        for (int t = 0; t < len; t++) sum += in[(i + t) * width + q];
        out[id] = sum;
    }
}

/* ================= PFB channelizer ==================================== *
 * src/blocks/pfb/channelizer.rs (liquid-dsp scheme), bulk closed form
 * of the round-robin window state for ANY decimation D = N/oversample
 * (channelizer.rs:126-210). Derivation: push p of the input stream
 * lands in window w(p) = (N-1-p) mod N (decrement_base_index); after
 * P = prefill + (k+1)*D pushes (prefill = N*tpf), output step k reads
 * window b's newest-j element x[pmax(b) - j*N] with
 *   pmax(b) = P-1 - ((P-1 - (N-1-b)) mod N)
 * through filter i(b) = (b - base - 1) mod N, base = (N-1-P) mod N:
 *   fft_buf[k][b] = sum_j x[pmax(b) - j*N] * part[i(b)][j]
 * (part = partition_filter_taps, utilities.rs:5-25; the FirFilter's
 * reversed taps turn the oldest->newest window into newest-j order).
 * `p_before` = pushes before this launch's first step (global),
 * `bufbase` = global stream index of in[0] (streaming carries the last
 * N*tpf consumed samples in front). D == N collapses to the maximally
 * decimated form (base stays N-1, i == b). */
__global__ void k_pfb_dots(const float2* __restrict__ in,
                           float2* __restrict__ fb,
                           const float* __restrict__ part /* [N][tpf] */,
                           int N, int tpf, int D, long long steps,
                           long long p_before, long long bufbase) {
    long long id = blockIdx.x * (long long)blockDim.x + threadIdx.x;
    long long total = steps * N;
    long long stride = (long long)gridDim.x * blockDim.x;
    for (; id < total; id += stride) {
        long long k = id / N;
        int b = (int)(id - k * N);
        long long P = p_before + (k + 1) * (long long)D;
        int base = (int)(((N - 1 - P) % N + N) % N);
        int i = ((b - base - 1) % N + N) % N;
        long long pmax =
            P - 1 - (((P - 1 - (N - 1 - b)) % N + N) % N);
        float sre = 0.f, sim = 0.f;
        for (int j = 0; j < tpf; j++) {
            float2 x = in[pmax - (long long)j * N - bufbase];
            float tap = part[i * tpf + j];
            sre = fmaf(x.x, tap, sre);
            sim = fmaf(x.y, tap, sim);
        }
        fb[k * N + b] = make_float2(sre, sim);
    }
}

__global__ void k_pfb_scatter(const float2* __restrict__ ifft,
                              float2* __restrict__ out, int N,
                              long long steps, long long cap) {
    long long id = blockIdx.x * (long long)blockDim.x + threadIdx.x;
    long long total = steps * N;
    long long stride = (long long)gridDim.x * blockDim.x;
    for (; id < total; id += stride) {
        long long k = id / N;
        int cidx = (int)(id - k * N);
        out[cidx * cap + k] = ifft[k * N + cidx];
    }
}

/* ================= MovingAvg ========================================== *
 * src/blocks/moving_avg.rs:79-118: per-bin EMA over WIDTH-sized frames,
 * emit every `history` frames; avg state lives in HBM (stateful block).
 * One lane per bin, sequential over frames (frames are a recurrence). */
__global__ void k_moving_avg(const float* __restrict__ in,
                             float* __restrict__ out,
                             float* __restrict__ avg, int width,
                             long long frames, int i0, int history,
                             float decay) {
    int b = blockIdx.x * blockDim.x + threadIdx.x;
    if (b >= width) return;
    float a = avg[b];
    int i = i0;
    long long prod = 0;
    for (long long f = 0; f < frames; f++) {
        float t = in[f * width + b];
        if (isfinite(t))
            a = (1.0f - decay) * a + decay * t;
        else
            a *= 1.0f - decay;
        if (++i == history) {
            out[prod * width + b] = a;
            i = 0;
            prod++;
        }
    }
    avg[b] = a;
}

/* ================= Rotator ============================================ *
 * Rotator::rotate — crates/futuredsp/src/rotator.rs:23-49: out[i] =
 * in[i] * phase0 * e^{i*theta*(i+1)}. The reference iterates
 * phase *= phase_incr per sample (accumulating f32 rounding); this kernel
 * computes the phase in closed form per sample (sincosf), which tracks
 * the IDEAL rotation — parity vs the oracle is tolerance-bounded by the
 * oracle's own drift (documented in tests). */
__global__ void k_rotator(const float2* __restrict__ in,
                          float2* __restrict__ out, long long n,
                          double theta, float p0r, float p0i) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
         i < n; i += stride) {
        /* angle product + mod-2pi reduction in f64: f32 theta*(i+1) loses
         * ~0.25 rad past i~3e6 and (float)(i+1) is inexact past 2^24 */
        double a = theta * (double)(i + 1);
        a -= 6.283185307179586 * floor(a * 0.15915494309189535);
        float s, c;
        __sincosf((float)a, &s, &c);
        float pr = c * p0r - s * p0i;
        float pi = c * p0i + s * p0r;
        float2 x = in[i];
        out[i] = make_float2(x.x * pr - x.y * pi, x.x * pi + x.y * pr);
    }
}

/* ================= FIR f32 x f32 (plumbing config) ==================== *
 * fir.rs:206-215 semantics; simple LDS-staged kernel (this path is the
 * reference's perf/fir plumbing shape, not the metric). */
__global__ __launch_bounds__(256) void k_fir_f32(
    const float* __restrict__ in, float* __restrict__ out,
    const float* __restrict__ taps, int n_taps, long long n_out,
    long long n_in_valid) {
    const unsigned TILE = 1024;
    const unsigned elems = TILE + n_taps - 1;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* s_x = (float*)smem;
    const int tid = threadIdx.x;
    for (long long tile = blockIdx.x; tile * (long long)TILE < n_out;
         tile += gridDim.x) {
        const long long out_base = tile * TILE;
        for (unsigned i = tid; i < elems; i += 256) {
            long long g = out_base + i;
            s_x[i] = (g < n_in_valid) ? in[g] : 0.f;
        }
        __syncthreads();
        for (int j = 0; j < 4; j++) {
            unsigned k = tid + j * 256; /* lane-strided: stride-1 reads */
            float sum = 0.f;
            for (int t = 0; t < n_taps; t++)
                sum = fmaf(s_x[k + t], taps[n_taps - 1 - t], sum);
            long long o = out_base + k;
            if (o < n_out) out[o] = sum;
        }
        __syncthreads();
    }
}

/* ================= Polyphase resampler, cf32 x f32 ==================== *
 * polyphase_resampling_fir.rs:108-118: y[k] = sum_t i[k*M/L + t] *
 * taps[L*(Tpp-t-1) + (k*M)%L]. Correctness-first kernel: taps staged in
 * LDS (per-lane bank index), one output per lane, inputs via L1/L2. */
__global__ void k_resamp_cf32(const float2* __restrict__ in,
                              float2* __restrict__ out,
                              const float* __restrict__ taps, int n_taps,
                              int interp, int decim, long long n_out,
                              long long n_in_valid) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* s_taps = (float*)smem;
    for (int i = threadIdx.x; i < n_taps; i += blockDim.x)
        s_taps[i] = taps[i];
    __syncthreads();
    const int tpp = n_taps / interp;
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long k = blockIdx.x * (long long)blockDim.x + threadIdx.x;
         k < n_out; k += stride) {
        int bank = (int)((k * decim) % interp);
        long long idx = k * decim / interp;
        float sre = 0.f, sim = 0.f;
        for (int t = 0; t < tpp; t++) {
            float2 x = (idx + t < n_in_valid) ? in[idx + t]
                                              : make_float2(0.f, 0.f);
            float h = s_taps[interp * (tpp - t - 1) + bank];
            sre = fmaf(x.x, h, sre);
            sim = fmaf(x.y, h, sim);
        }
        out[k] = make_float2(sre, sim);
    }
}

/* LDS-tiled resampler (polyphase_resampling_fir.rs:108-118 semantics,
 * identical output math to k_resamp_cf32): a tile of RS_TILE outputs
 * stages its whole input span into padded SoA LDS planes once, then
 * each lane runs its tpp-tap phase dot product from LDS — removes the
 * per-output global gather of the naive kernel. Fallback to
 * k_resamp_cf32 when the span exceeds the LDS budget (host decides). */
#define RS_BLOCK 256
#define RS_R 2
#define RS_TILE (RS_BLOCK * RS_R)

__global__ __launch_bounds__(RS_BLOCK) void k_resamp_tiled_cf32(
    const float2* __restrict__ in, float2* __restrict__ out,
    const float* __restrict__ taps, int n_taps, int interp, int decim,
    long long n_out, long long n_in_valid, int span) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* s_re = (float*)smem;
    float* s_im = s_re + plane_floats((unsigned)span);
    float* s_taps = s_im + plane_floats((unsigned)span);
    for (int i = threadIdx.x; i < n_taps; i += RS_BLOCK)
        s_taps[i] = taps[i];
    const int tpp = n_taps / interp;
    const long long tiles = (n_out + RS_TILE - 1) / RS_TILE;
    for (long long tile = blockIdx.x; tile < tiles; tile += gridDim.x) {
        const long long out_base = tile * RS_TILE;
        const long long in_base = out_base * decim / interp;
        __syncthreads(); /* previous tile's reads done before restage */
        for (int i = threadIdx.x; i < span; i += RS_BLOCK) {
            long long g = in_base + i;
            float2 v = (g < n_in_valid) ? in[g] : make_float2(0.f, 0.f);
            s_re[lds_pad((unsigned)i)] = v.x;
            s_im[lds_pad((unsigned)i)] = v.y;
        }
        __syncthreads();
#pragma unroll
        for (int r = 0; r < RS_R; r++) {
            long long k = out_base + threadIdx.x + r * RS_BLOCK;
            if (k >= n_out) break;
            int bank = (int)((k * decim) % interp);
            unsigned rel = (unsigned)(k * decim / interp - in_base);
            float sre = 0.f, sim = 0.f;
            const float* tb = s_taps + bank;
#pragma unroll 4
            for (int t = 0; t < tpp; t++) {
                unsigned a = lds_pad(rel + (unsigned)t);
                float h = tb[interp * (tpp - t - 1)];
                sre = fmaf(s_re[a], h, sre);
                sim = fmaf(s_im[a], h, sim);
            }
            out[k] = make_float2(sre, sim);
        }
    }
}

/* ================= FFT: radix-4 Stockham in LDS ======================= *
 * Unnormalized DFT, rustfft convention (forward e^{-2pi i kn/N}) — the
 * reference Fft block's math (src/blocks/fft.rs:190-194). Stockham
 * autosort, radix-4 stages (plus one radix-2 stage when log2 N is odd):
 * half the LDS round trips and barriers of radix-2. One or more frames
 * per 256-thread block, ping-pong LDS, twiddle table (float2,
 * W[k] = e^{-2pi i k/N}, k < N, f64-computed on host). Supports
 * fft_shift / normalize per fft.rs:179-210, and an optional fused |X|^2
 * output (the spectrum Apply stage, examples/spectrum cpu.rs:21-28).
 */
__device__ __forceinline__ float2 cmul_tw(float2 a, float2 w, int inverse) {
    if (inverse) w.y = -w.y;
    return cmulf(a, w);
}

/* In-block 1024-pt forward FFT (unnormalized), 256 threads, for the
 * chain's fused decim+FFT kernel. Same radix-4 DIF scheme and fft_swz
 * LDS swizzle as k_fft_stockham. */
__device__ float2* fft_pow2_fwd(float2* a, float2* b,
                                const float2* __restrict__ twid, int n,
                                int tf, int tpf) {
    int scur = 1;
    int ncur = n;
    while (ncur >= 4) {
        const int m4 = ncur >> 2;
        for (int bf = tf; bf < n / 4; bf += tpf) {
            const int p = bf / scur;
            const int q = bf - p * scur;
            const int tw = n / ncur;
            float2 x0 = a[fft_swz(q + scur * p)];
            float2 x1 = a[fft_swz(q + scur * (p + m4))];
            float2 x2 = a[fft_swz(q + scur * (p + 2 * m4))];
            float2 x3 = a[fft_swz(q + scur * (p + 3 * m4))];
            float2 e0 = f2_add(x0, x2), e1 = f2_sub(x0, x2);
            float2 o0 = f2_add(x1, x3), o1 = f2_sub(x1, x3);
            float2 o1r = make_float2(o1.y, -o1.x);
            /* one twiddle load; W^2p, W^3p derived by complex mults
             * (unit-vector products, ~2 ulp — the 3-load version was
             * VMEM-issue-bound inside the fused chain kernel) */
            float2 w1 = twid[(size_t)p * tw];
            float2 w2 = cmulf(w1, w1);
            float2 w3 = cmulf(w2, w1);
            b[fft_swz(q + scur * (4 * p + 0))] = f2_add(e0, o0);
            b[fft_swz(q + scur * (4 * p + 1))] = cmulf(f2_add(e1, o1r), w1);
            b[fft_swz(q + scur * (4 * p + 2))] = cmulf(f2_sub(e0, o0), w2);
            b[fft_swz(q + scur * (4 * p + 3))] = cmulf(f2_sub(e1, o1r), w3);
        }
        float2* t = a; a = b; b = t;
        scur <<= 2;
        ncur >>= 2;
        __syncthreads();
    }
    if (ncur == 2) { /* radix-2 tail for odd log2n */
        for (int bf = tf; bf < n / 2; bf += tpf) {
            const int p = bf / scur;
            const int q = bf - p * scur;
            float2 xa = a[fft_swz(q + scur * p)];
            float2 xb = a[fft_swz(q + scur * (p + 1))];
            b[fft_swz(q + scur * 2 * p)] = f2_add(xa, xb);
            b[fft_swz(q + scur * (2 * p + 1))] =
                cmulf(f2_sub(xa, xb), twid[(size_t)p * (n / 2)]);
        }
        float2* t = a; a = b; b = t;
        __syncthreads();
    }
    return a;
}

__device__ void fft1024_block(float2* ping, float2* pong,
                              const float2* __restrict__ twid,
                              int tf /* thread-in-frame, stride 256 */) {
    (void)fft_pow2_fwd(ping, pong, twid, 1024, tf, 256);
    /* 5 radix-4 stages = 5 ping/pong swaps: result in `pong`. */
}

/* ---- Bluestein (non-pow2 lengths) ------------------------------------ *
 * X[k] = w(k) * IFFT_M(FFT_M(x.w) * B)[k] / M, w the quadratic chirp
 * and B the precomputed transform of the wrapped conj chirp (tables in
 * fsdr_fft_cf32_create). The M-point passes reuse k_fft_stockham. */
__global__ void k_bluestein_pre(const float2* __restrict__ in,
                                float2* __restrict__ out /* frames*M */,
                                const float2* __restrict__ chirp, int n,
                                int M, long long frames, int shift_in) {
    long long total = frames * M;
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long id = blockIdx.x * (long long)blockDim.x + threadIdx.x;
         id < total; id += stride) {
        long long fr = id / M;
        int j = (int)(id - fr * M);
        float2 v = make_float2(0.f, 0.f);
        if (j < n) {
            int src = shift_in ? (j + n / 2) % n : j; /* fft.rs:179-185 */
            v = cmulf(in[fr * n + src], chirp[j]);
        }
        out[id] = v;
    }
}

__global__ void k_bluestein_mul(float2* __restrict__ x,
                                const float2* __restrict__ B, int M,
                                long long frames) {
    long long total = frames * M;
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long id = blockIdx.x * (long long)blockDim.x + threadIdx.x;
         id < total; id += stride)
        x[id] = cmulf(x[id], B[id % M]);
}

__global__ void k_bluestein_post(const float2* __restrict__ conv,
                                 float2* __restrict__ out,
                                 float* __restrict__ mag /* nullable */,
                                 const float2* __restrict__ chirp, int n,
                                 int M, long long frames, int shift_out,
                                 float scale) {
    long long total = frames * n;
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long id = blockIdx.x * (long long)blockDim.x + threadIdx.x;
         id < total; id += stride) {
        long long fr = id / n;
        int k = (int)(id - fr * n);
        int src = shift_out ? (k + n / 2) % n : k; /* fft.rs:196-204 */
        float2 v = cmulf(conv[fr * M + src], chirp[src]);
        v.x *= scale;
        v.y *= scale;
        out[id] = v;
        if (mag) mag[id] = v.x * v.x + v.y * v.y;
    }
}

__global__ __launch_bounds__(256) void k_fft_stockham(
    const float2* __restrict__ in, float2* __restrict__ out,
    float* __restrict__ mag_out /* nullable */,
    const float2* __restrict__ twid, int n, int log2n, int frames_per_block,
    int inverse, int fft_shift, float norm /* 0 = none */,
    long long n_frames) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float2* ping = (float2*)smem;                 /* [fpb][n] */
    float2* pong = ping + (size_t)frames_per_block * n;

    const int tpf = blockDim.x / frames_per_block;   /* threads per frame */
    const int fl = threadIdx.x / tpf;                /* frame within block */
    const int tf = threadIdx.x % tpf;                /* thread within frame */

    for (long long fb = (long long)blockIdx.x * frames_per_block;
         fb < n_frames; fb += (long long)gridDim.x * frames_per_block) {
        const long long frame = fb + fl;
        float2* a = ping + (size_t)fl * n;
        float2* b = pong + (size_t)fl * n;
        if (frame < n_frames) {
            const float2* src = in + frame * n;
            if (inverse && fft_shift) {       /* fft.rs:179-185: shift input */
                for (int i = tf; i < n; i += tpf)
                    a[fft_swz(i)] = src[(i + n / 2) % n];
            } else {
                for (int i = tf; i < n; i += tpf) a[fft_swz(i)] = src[i];
            }
        }
        __syncthreads();
        int scur = 1;
        int ncur = n;
        while (ncur >= 4) {
            const int m4 = ncur >> 2;
            if (frame < n_frames) {
                for (int bf = tf; bf < n / 4; bf += tpf) {
                    const int p = bf / scur;
                    const int q = bf - p * scur;
                    const int tw = n / ncur; /* twiddle stride */
                    float2 x0 = a[fft_swz(q + scur * p)];
                    float2 x1 = a[fft_swz(q + scur * (p + m4))];
                    float2 x2 = a[fft_swz(q + scur * (p + 2 * m4))];
                    float2 x3 = a[fft_swz(q + scur * (p + 3 * m4))];
                    /* DIF radix-4: butterflies first, output twiddles
                     * W^p, W^2p, W^3p on frequencies 1..3 (omega4 = -i
                     * forward, +i inverse). One twiddle load; the
                     * squares/cubes are derived (unit-vector products,
                     * ~2 ulp). */
                    float2 e0 = f2_add(x0, x2), e1 = f2_sub(x0, x2);
                    float2 o0 = f2_add(x1, x3), o1 = f2_sub(x1, x3);
                    float2 o1r = inverse ? make_float2(-o1.y, o1.x)
                                         : make_float2(o1.y, -o1.x);
                    float2 w1 = twid[(size_t)p * tw];
                    if (inverse) w1.y = -w1.y;
                    float2 w2 = cmulf(w1, w1);
                    float2 w3 = cmulf(w2, w1);
                    b[fft_swz(q + scur * (4 * p + 0))] = f2_add(e0, o0);
                    b[fft_swz(q + scur * (4 * p + 1))] =
                        cmulf(f2_add(e1, o1r), w1);
                    b[fft_swz(q + scur * (4 * p + 2))] =
                        cmulf(f2_sub(e0, o0), w2);
                    b[fft_swz(q + scur * (4 * p + 3))] =
                        cmulf(f2_sub(e1, o1r), w3);
                }
            }
            float2* t = a; a = b; b = t;
            scur <<= 2;
            ncur >>= 2;
            __syncthreads();
        }
        if (ncur == 2) { /* final radix-2 stage when log2n is odd */
            if (frame < n_frames) {
                for (int bf = tf; bf < n / 2; bf += tpf) {
                    const int p = bf / scur;
                    const int q = bf - p * scur;
                    float2 xa = a[fft_swz(q + scur * p)];
                    float2 xb = a[fft_swz(q + scur * (p + 1))];
                    b[fft_swz(q + scur * 2 * p)] = f2_add(xa, xb);
                    b[fft_swz(q + scur * (2 * p + 1))] =
                        cmul_tw(f2_sub(xa, xb), twid[(size_t)p * (n / 2)],
                                inverse);
                }
            }
            float2* t = a; a = b; b = t;
            __syncthreads();
        }
        if (frame < n_frames) {
            float2* dst = out + frame * n;
            const bool shift_out = (!inverse) && fft_shift; /* fft.rs:196-204 */
            for (int i = tf; i < n; i += tpf) {
                float2 v = a[fft_swz(shift_out ? (i + n / 2) % n : i)];
                if (norm != 0.f) { v.x *= norm; v.y *= norm; }
                dst[i] = v;
                if (mag_out) mag_out[frame * n + i] = v.x * v.x + v.y * v.y;
            }
        }
        __syncthreads();
    }
}

/* ================= element-wise ======================================= */

__global__ void k_mag2(const float2* __restrict__ in, float* __restrict__ out,
                       long long n) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
         i < n; i += stride) {
        float2 v = in[i];
        out[i] = v.x * v.x + v.y * v.y; /* norm_sqr — spectrum cpu.rs:21-28 */
    }
}

__global__ void k_cmul(const float2* __restrict__ a,
                       const float2* __restrict__ b, float2* __restrict__ o,
                       long long n) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
         i < n; i += stride)
        o[i] = cmulf(a[i], b[i]);
}

/* Deterministic synthetic source: splitmix64 per element, re/im iid
 * uniform[-1,1) (the reference bench convention,
 * crates/futuredsp/benches/benchmarks.rs:23-30). */
__device__ __forceinline__ uint64_t splitmix64(uint64_t x) {
    x += 0x9E3779B97F4A7C15ull;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
    return x ^ (x >> 31);
}

/* lowbias32 (Wellons): full-avalanche 32-bit hash — two 32-bit mults
 * per 32 output bits vs splitmix64's two 64-bit mults per 64 (64-bit
 * integer mults are multi-op on CDNA4; the fill was ALU-capped below
 * write bandwidth). */
__device__ __forceinline__ uint32_t lowbias32(uint32_t x) {
    x ^= x >> 16;
    x *= 0x7FEB352Du;
    x ^= x >> 15;
    x *= 0x846CA68Bu;
    x ^= x >> 16;
    return x;
}

__global__ void k_fill_uniform_cf32(float2* __restrict__ out, long long n,
                                    uint64_t seed, uint64_t offset) {
    /* pair of samples per iteration, one float4 store (the source write
     * is inside the bench's timed region — keep it at write bandwidth).
     * Counter-based: deterministic by (seed, element index). */
    const uint32_t sm =
        (uint32_t)seed * 0x9E3779B9u ^ (uint32_t)(seed >> 32);
    long long np = n >> 1;
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long p = blockIdx.x * (long long)blockDim.x + threadIdx.x;
         p < np; p += stride) {
        long long i = 2 * p;
        uint64_t c64 = 0x5D5D5D5Dull + offset + (uint64_t)i;
        uint32_t c = (uint32_t)c64 ^ (uint32_t)(c64 >> 32) * 0x85EBCA6Bu;
        uint32_t r0 = lowbias32(c * 4u + sm);
        uint32_t r1 = lowbias32(c * 4u + 1u + sm);
        uint32_t r2 = lowbias32(c * 4u + 2u + sm);
        uint32_t r3 = lowbias32(c * 4u + 3u + sm);
        const float s = 2.0f / 16777216.0f;
        float4 v = make_float4((r0 >> 8) * s - 1.0f, (r1 >> 8) * s - 1.0f,
                               (r2 >> 8) * s - 1.0f, (r3 >> 8) * s - 1.0f);
        *(float4*)&out[i] = v;
    }
    if ((n & 1) && blockIdx.x == 0 && threadIdx.x == 0) {
        long long i = n - 1;
        uint64_t h =
            splitmix64(seed ^ (0x5D5D5D5Dull + offset + (uint64_t)i));
        out[i] = make_float2(((uint32_t)h >> 8) * (2.0f / 16777216.0f) -
                                 1.0f,
                             ((uint32_t)(h >> 32) >> 8) *
                                     (2.0f / 16777216.0f) -
                                 1.0f);
    }
}

/* ================= host-side status math ============================== */

static size_t sat_sub(size_t a, size_t b) { return a > b ? a - b : 0; }

/* fir.rs:69-74 */
static fsdr_filter_result fir_status(size_t n_in, size_t nt, size_t n_out) {
    size_t prod = sat_sub(n_in + 1, nt);
    fsdr_filter_result r;
    if (prod > n_out) {
        r.consumed = r.produced = n_out;
        r.status = FSDR_INSUFFICIENT_OUTPUT;
    } else if (prod == n_out) {
        r.consumed = r.produced = prod;
        r.status = FSDR_BOTH_SUFFICIENT;
    } else {
        r.consumed = r.produced = prod;
        r.status = FSDR_INSUFFICIENT_INPUT;
    }
    return r;
}

/* decimating_fir.rs:71-78,94 */
static fsdr_filter_result decim_status(size_t D, size_t n_in, size_t nt,
                                       size_t n_out) {
    size_t consumable = sat_sub(n_in + 1, nt) / D;
    fsdr_filter_result r;
    if (consumable > n_out) {
        r.produced = n_out;
        r.status = FSDR_INSUFFICIENT_OUTPUT;
    } else if (consumable == n_out) {
        r.produced = n_out;
        r.status = FSDR_BOTH_SUFFICIENT;
    } else {
        r.produced = consumable;
        r.status = FSDR_INSUFFICIENT_INPUT;
    }
    r.consumed = r.produced * D;
    return r;
}

/* polyphase_resampling_fir.rs:90-106 */
static fsdr_filter_result resamp_status(size_t L, size_t M, size_t nt_total,
                                        size_t n_in, size_t n_out) {
    size_t nt = nt_total / L;
    size_t prod = sat_sub(sat_sub(n_in + 1, nt) * L, 1) / M;
    prod = (prod / L) * L;
    fsdr_filter_result r;
    if (prod > n_out) {
        r.produced = (n_out / L) * L;
        r.status = FSDR_INSUFFICIENT_OUTPUT;
    } else if (prod == n_out) {
        r.produced = prod;
        r.status = FSDR_BOTH_SUFFICIENT;
    } else {
        r.produced = prod;
        r.status = FSDR_INSUFFICIENT_INPUT;
    }
    r.consumed = (r.produced / L) * M;
    return r;
}

/* ================= filter handles ===================================== */

enum FilterKind { K_FIR_CF32, K_FIR_F32, K_DECIM_CF32, K_RESAMP_CF32,
                  K_FFT_CF32, K_MAG2, K_FIR_CCF32, K_MOVAVG,
                  K_XLATING, K_PFB };

struct fsdr_filter {
    FilterKind kind;
    size_t width = 0;        /* MovingAvg */
    size_t history = 0;
    size_t i_state = 0;
    float decay = 0.f;        /* MovingAvg decay factor */
    float theta = 0.f;        /* XlatingFir rotator increment */
    float rot_re = 1.f, rot_im = 0.f; /* xlating rotator phase state */
    fsdr_filter* sub = nullptr;       /* channelizer's internal IFFT */
    float* d_avg = nullptr;
    size_t n_taps = 0;       /* true tap count (length()) */
    size_t decim = 1, interp = 1;
    size_t fft_len = 0;
    int inverse = 0, fft_shift = 0;
    float norm = 0.f;
    /* Bluestein (non-pow2 fft_len): M = next pow2 >= 2*len-1; chirp
     * w(k) = e^{sigma*pi*i*k^2/len}; B = DFT_M of the wrapped conj
     * chirp. fft.rs is generic over rustfft plan lengths; this closes
     * the pow2-only gap via two pow2 M-point passes per batch. */
    size_t fft_m = 0;
    float2* d_chirp = nullptr;
    float2* d_B = nullptr;
    int n_taps_padded = 0;   /* device taps length (leading zeros) */
    float* d_taps = nullptr;
    int tp_tpl = 0;          /* template tap count (reversed taps) or 0 */
    float* d_rtaps = nullptr; /* reversed taps zero-filled to tp_tpl */
    int kk_mfma = 0;         /* MFMA variant K (>= n_taps+15, %4==0) */
    float* d_mtaps = nullptr; /* reversed taps zero-filled to kk_mfma */
    int kk_mfma32 = 0;       /* 32x32 variant K (>= n_taps+31, even) */
    float* d_mtaps32 = nullptr;
    float2* d_twid = nullptr;
    /* staging buffers for the host-span path */
    void* d_in = nullptr;
    void* d_out = nullptr;
    size_t d_in_bytes = 0, d_out_bytes = 0;
    size_t item_in = 8, item_out = 8;
    /* dedicated kernel scratch (MovingAvg chunk partials, PFB work
     * buffer). MUST be distinct from d_in: on the fsdr_filter_host path
     * d_in holds the staged input while the partials kernel is still
     * reading it. */
    void* d_scratch = nullptr;
    size_t d_scratch_bytes = 0;
    /* PFB streaming state: the last N*tpf CONSUMED samples (the window
     * contents) + total pushes (channelizer.rs round-robin state) */
    void* d_hist = nullptr;
    size_t d_hist_bytes = 0;
    size_t pfb_pushes = 0;
};

static int ensure_dev(void** p, size_t* cur, size_t want) {
    if (*cur >= want) return FSDR_OK;
    if (*p) (void)hipFree(*p);
    *p = nullptr;
    *cur = 0;
    HIP_TRY(hipMalloc(p, want));
    *cur = want;
    return FSDR_OK;
}

/* pad taps to tp ≡ 1 (mod m) with leading zeros (zero taps
 * multiply staged zeros beyond the true window; result unchanged). */
static int upload_taps_padded(fsdr_filter* f, const float* taps, size_t nt,
                              int mod) {
    size_t tp = nt;
    while (tp % mod != 1 % mod) tp++;
    std::vector<float> h(tp, 0.f);
    memcpy(h.data() + (tp - nt), taps, nt * sizeof(float));
    HIP_TRY(hipMalloc(&f->d_taps, tp * sizeof(float)));
    HIP_TRY(hipMemcpy(f->d_taps, h.data(), tp * sizeof(float),
                      hipMemcpyHostToDevice));
    f->n_taps_padded = (int)tp;
    return FSDR_OK;
}

static fsdr_filter* create_common(FilterKind k) {
    if (!have_gpu()) {
        set_err("no HIP device present — cannot create GPU filter");
        return nullptr;
    }
    fsdr_filter* f = new fsdr_filter();
    f->kind = k;
    return f;
}

extern "C" fsdr_filter* fsdr_fir_cf32_create(const float* taps,
                                             size_t n_taps) {
    if (!taps || n_taps == 0) { set_err("null/empty taps"); return nullptr; }
    fsdr_filter* f = create_common(K_FIR_CF32);
    if (!f) return nullptr;
    f->n_taps = n_taps;
    if (upload_taps_padded(f, taps, n_taps, 8) != FSDR_OK) {
        delete f;
        return nullptr;
    }
    /* template variant: reversed taps zero-filled to the smallest
     * instantiated TP >= n_taps */
    static const int tps[] = {17, 33, 65, 129, 257, 513};
    for (int t : tps) {
        if ((size_t)t >= n_taps) { f->tp_tpl = t; break; }
    }
    if (f->tp_tpl) {
        std::vector<float> rt(f->tp_tpl, 0.f);
        for (size_t i = 0; i < n_taps; i++) rt[i] = taps[n_taps - 1 - i];
        if (hipMalloc(&f->d_rtaps, rt.size() * sizeof(float)) != hipSuccess ||
            hipMemcpy(f->d_rtaps, rt.data(), rt.size() * sizeof(float),
                      hipMemcpyHostToDevice) != hipSuccess) {
            set_err("reversed taps upload failed");
            delete f;
            return nullptr;
        }
    }
    static const int kk32s[] = {48, 96, 160, 288, 544};
    for (int k : kk32s)
        if ((size_t)k >= n_taps + 31) { f->kk_mfma32 = k; break; }
    if (f->kk_mfma32) {
        std::vector<float> rt(f->kk_mfma32, 0.f);
        for (size_t i = 0; i < n_taps; i++) rt[i] = taps[n_taps - 1 - i];
        if (hipMalloc(&f->d_mtaps32, rt.size() * sizeof(float)) !=
                hipSuccess ||
            hipMemcpy(f->d_mtaps32, rt.data(), rt.size() * sizeof(float),
                      hipMemcpyHostToDevice) != hipSuccess) {
            set_err("mfma32 taps upload failed");
            delete f;
            return nullptr;
        }
    }
    static const int kks[] = {32, 64, 96, 144, 272, 528};
    for (int k : kks)
        if ((size_t)k >= n_taps + 15) { f->kk_mfma = k; break; }
    if (f->kk_mfma) {
        std::vector<float> rt(f->kk_mfma, 0.f);
        for (size_t i = 0; i < n_taps; i++) rt[i] = taps[n_taps - 1 - i];
        if (hipMalloc(&f->d_mtaps, rt.size() * sizeof(float)) != hipSuccess ||
            hipMemcpy(f->d_mtaps, rt.data(), rt.size() * sizeof(float),
                      hipMemcpyHostToDevice) != hipSuccess) {
            set_err("mfma taps upload failed");
            delete f;
            return nullptr;
        }
    }
    return f;
}

extern "C" fsdr_filter* fsdr_fir_ccf32_create(const fsdr_cf32* taps,
                                              size_t n_taps) {
    if (!taps || n_taps == 0) { set_err("null/empty taps"); return nullptr; }
    fsdr_filter* f = create_common(K_FIR_CCF32);
    if (!f) return nullptr;
    f->n_taps = n_taps;
    f->n_taps_padded = (int)n_taps;
    if (hipMalloc(&f->d_taps, n_taps * sizeof(fsdr_cf32)) != hipSuccess ||
        hipMemcpy(f->d_taps, taps, n_taps * sizeof(fsdr_cf32),
                  hipMemcpyHostToDevice) != hipSuccess) {
        set_err("taps upload failed");
        delete f;
        return nullptr;
    }
    return f;
}

extern "C" fsdr_filter* fsdr_fir_f32_create(const float* taps,
                                            size_t n_taps) {
    if (!taps || n_taps == 0) { set_err("null/empty taps"); return nullptr; }
    fsdr_filter* f = create_common(K_FIR_F32);
    if (!f) return nullptr;
    f->n_taps = n_taps;
    f->item_in = f->item_out = 4;
    std::vector<float> h(taps, taps + n_taps);
    if (hipMalloc(&f->d_taps, n_taps * sizeof(float)) != hipSuccess ||
        hipMemcpy(f->d_taps, h.data(), n_taps * sizeof(float),
                  hipMemcpyHostToDevice) != hipSuccess) {
        set_err("taps upload failed");
        delete f;
        return nullptr;
    }
    f->n_taps_padded = (int)n_taps;
    return f;
}

extern "C" fsdr_filter* fsdr_decim_fir_cf32_create(size_t decimation,
                                                   const float* taps,
                                                   size_t n_taps) {
    if (!taps || n_taps == 0 || decimation == 0) {
        set_err("invalid decimating FIR parameters");
        return nullptr;
    }
    fsdr_filter* f = create_common(K_DECIM_CF32);
    if (!f) return nullptr;
    f->n_taps = n_taps;
    f->decim = decimation;
    int mod = (decimation == 4) ? 16 : 1; /* fast path needs tp%16==1 */
    if (upload_taps_padded(f, taps, n_taps, mod) != FSDR_OK) {
        delete f;
        return nullptr;
    }
    if (decimation == 4 && n_taps <= 512) {
        /* MFMA variant: per-phase K = ceil(T/4)+15 rounded to 4 */
        static const int kkds[] = {20, 32, 48, 80, 144};
        size_t tpd_true = (n_taps + 3) / 4;
        for (int k : kkds)
            if ((size_t)k >= tpd_true + 15) { f->kk_mfma = k; break; }
        if (f->kk_mfma) {
            int kkd = f->kk_mfma;
            std::vector<float> rm(4 * (size_t)kkd, 0.f);
            for (int v = 0; v < 4; v++)
                for (int u = 0; u < kkd; u++) {
                    size_t t = 4 * (size_t)u + v;
                    if (t < n_taps)
                        rm[v * kkd + u] = taps[n_taps - 1 - t];
                }
            if (hipMalloc(&f->d_mtaps, rm.size() * sizeof(float)) !=
                    hipSuccess ||
                hipMemcpy(f->d_mtaps, rm.data(),
                          rm.size() * sizeof(float),
                          hipMemcpyHostToDevice) != hipSuccess) {
                set_err("decim mfma taps upload failed");
                delete f;
                return nullptr;
            }
        }
        /* phase-split template: rtv[v][u] = h[n_taps-1-(4u+v)] */
        static const int tpds[] = {8, 16, 32, 64, 128};
        for (int t : tpds) {
            if ((size_t)(4 * t) >= n_taps) { f->tp_tpl = t; break; }
        }
        int tpd = f->tp_tpl;
        std::vector<float> rtv(4 * (tpd + 4), 0.f);
        for (int v = 0; v < 4; v++)
            for (int u = 0; u < tpd; u++) {
                size_t t = 4 * (size_t)u + v;
                if (t < n_taps)
                    rtv[v * (tpd + 4) + u] = taps[n_taps - 1 - t];
            }
        if (hipMalloc(&f->d_rtaps, rtv.size() * sizeof(float)) !=
                hipSuccess ||
            hipMemcpy(f->d_rtaps, rtv.data(), rtv.size() * sizeof(float),
                      hipMemcpyHostToDevice) != hipSuccess) {
            set_err("rtv upload failed");
            delete f;
            return nullptr;
        }
    }
    return f;
}

extern "C" fsdr_filter* fsdr_resamp_cf32_create(size_t interp, size_t decim,
                                                const float* taps,
                                                size_t n_taps) {
    if (!taps || n_taps == 0 || interp == 0 || decim == 0 ||
        n_taps % interp != 0) {
        /* polyphase_resampling_fir.rs:54-56 assert */
        set_err("invalid resampler parameters (n_taps % interp must be 0)");
        return nullptr;
    }
    fsdr_filter* f = create_common(K_RESAMP_CF32);
    if (!f) return nullptr;
    f->n_taps = n_taps;
    f->interp = interp;
    f->decim = decim;
    if (hipMalloc(&f->d_taps, n_taps * sizeof(float)) != hipSuccess ||
        hipMemcpy(f->d_taps, taps, n_taps * sizeof(float),
                  hipMemcpyHostToDevice) != hipSuccess) {
        set_err("taps upload failed");
        delete f;
        return nullptr;
    }
    f->n_taps_padded = (int)n_taps;
    /* interp==1, decim==4 (the config-3 "4:1" shape): y[k] =
     * sum_t x[4k+t]*h[T-1-t] is the decimating FIR evaluated with its
     * window origin shifted by D-1, so route it to the MFMA decim
     * kernel with the input pointer rebased (launch site). */
    if (interp == 1 && decim == 4 && n_taps <= 512)
        f->sub = fsdr_decim_fir_cf32_create(4, taps, n_taps);
    return f;
}

extern "C" fsdr_filter* fsdr_fft_cf32_create(size_t len, int inverse,
                                             int fft_shift,
                                             const float* normalize) {
    const bool pow2 = (len & (len - 1)) == 0;
    if (pow2 ? (len < 4 || len > 4096) : (len < 2 || len > 2048)) {
        set_err("fft len: pow2 in [4,4096], or any length in [2,2048] "
                "(Bluestein via 2*len-1-padded pow2 passes)");
        return nullptr;
    }
    fsdr_filter* f = create_common(K_FFT_CF32);
    if (!f) return nullptr;
    f->fft_len = len;
    f->inverse = inverse;
    f->fft_shift = fft_shift;
    f->norm = normalize ? *normalize : 0.f;
    f->n_taps = len; /* length() = min_items = len (fft.rs:106-109) */
    size_t tlen = len;
    if (!pow2) {
        size_t M = 4;
        while (M < 2 * len - 1) M <<= 1;
        f->fft_m = M;
        tlen = M; /* twiddles for the M-point passes */
        /* chirp w(k) = e^{sigma*pi*i*k^2/len}, sigma = -1 fwd / +1 inv;
         * k^2 reduced mod 2*len in integers before the f64 angle */
        const double sg = inverse ? 1.0 : -1.0;
        std::vector<float2> w(len);
        std::vector<double> wr(len), wi(len);
        for (size_t k = 0; k < len; k++) {
            unsigned long long r = (k * k) % (2 * len);
            double a = sg * M_PI * (double)r / (double)len;
            wr[k] = cos(a);
            wi[k] = sin(a);
            w[k] = make_float2((float)wr[k], (float)wi[k]);
        }
        /* V_seq = wrapped conj(w); B = forward DFT_M(V_seq) in f64 */
        std::vector<double> vr(M, 0.0), vi(M, 0.0);
        for (size_t j = 0; j < len; j++) {
            vr[j] = wr[j];
            vi[j] = -wi[j];
            if (j) {
                vr[M - j] = wr[j];
                vi[M - j] = -wi[j];
            }
        }
        std::vector<double> cm(M), sm(M);
        for (size_t k = 0; k < M; k++) {
            double a = -2.0 * M_PI * (double)k / (double)M;
            cm[k] = cos(a);
            sm[k] = sin(a);
        }
        std::vector<float2> B(M);
        for (size_t k = 0; k < M; k++) {
            double sre = 0.0, sim = 0.0;
            for (size_t j2 = 0; j2 < M; j2++) {
                if (vr[j2] == 0.0 && vi[j2] == 0.0) continue;
                size_t idx = (k * j2) % M;
                double c = cm[idx], s = sm[idx];
                sre += vr[j2] * c - vi[j2] * s;
                sim += vr[j2] * s + vi[j2] * c;
            }
            B[k] = make_float2((float)sre, (float)sim);
        }
        if (hipMalloc(&f->d_chirp, len * sizeof(float2)) != hipSuccess ||
            hipMemcpy(f->d_chirp, w.data(), len * sizeof(float2),
                      hipMemcpyHostToDevice) != hipSuccess ||
            hipMalloc(&f->d_B, M * sizeof(float2)) != hipSuccess ||
            hipMemcpy(f->d_B, B.data(), M * sizeof(float2),
                      hipMemcpyHostToDevice) != hipSuccess) {
            set_err("bluestein table upload failed");
            fsdr_filter_destroy(f);
            return nullptr;
        }
    }
    /* twiddle table W[k] = e^{-2πik/tlen}, k < tlen, computed in f64
     * (radix-4 needs indices up to 3(tlen/4-1)) */
    std::vector<float2> tw(tlen);
    for (size_t k = 0; k < tlen; k++) {
        double a = -2.0 * M_PI * (double)k / (double)tlen;
        tw[k] = make_float2((float)cos(a), (float)sin(a));
    }
    if (hipMalloc(&f->d_twid, tw.size() * sizeof(float2)) != hipSuccess ||
        hipMemcpy(f->d_twid, tw.data(), tw.size() * sizeof(float2),
                  hipMemcpyHostToDevice) != hipSuccess) {
        set_err("twiddle upload failed");
        fsdr_filter_destroy(f);
        return nullptr;
    }
    return f;
}

extern "C" fsdr_filter* fsdr_mag2_create(void) {
    fsdr_filter* f = create_common(K_MAG2);
    if (!f) return nullptr;
    f->item_out = 4;
    return f;
}

extern "C" fsdr_filter* fsdr_xlating_fir_cf32_create(const float* taps,
                                                     size_t n_taps,
                                                     size_t decimation,
                                                     float offset,
                                                     float sample_rate) {
    /* xlating_fir.rs:44 assert decimation >= 2 */
    if (!taps || n_taps == 0 || decimation < 2) {
        set_err("xlating fir: taps required, decimation must be >= 2");
        return nullptr;
    }
    fsdr_filter* f = create_common(K_XLATING);
    if (!f) return nullptr;
    f->n_taps = n_taps;
    f->decim = decimation;
    /* bpf taps, f32 math identical to from_polar (:79-88), stored
     * REVERSED for the kernel */
    std::vector<float2> bpf(n_taps);
    for (size_t i = 0; i < n_taps; i++) {
        float ang = (float)i * 6.2831853071795864769f * offset / sample_rate;
        float2 rot = make_float2(cosf(ang), sinf(ang));
        size_t src_i = i;
        bpf[n_taps - 1 - src_i] =
            make_float2(rot.x * taps[src_i], rot.y * taps[src_i]);
    }
    if (hipMalloc(&f->d_taps, bpf.size() * sizeof(float2)) != hipSuccess ||
        hipMemcpy(f->d_taps, bpf.data(), bpf.size() * sizeof(float2),
                  hipMemcpyHostToDevice) != hipSuccess) {
        set_err("bpf taps upload failed");
        delete f;
        return nullptr;
    }
    /* rotator increment (:91-94) and unit start phase (rotator.rs:17-19) */
    f->theta = -6.2831853071795864769f * offset * (float)decimation /
               sample_rate;
    f->rot_re = 1.0f;
    f->rot_im = 0.0f;
    f->n_taps_padded = (int)n_taps;
    return f;
}

extern "C" fsdr_filter* fsdr_pfb_channelizer_create(size_t num_channels,
                                                    const float* taps,
                                                    size_t n_taps,
                                                    float oversample_rate) {
    /* channelizer.rs:94-106 asserts */
    if (num_channels <= 2 || !taps || n_taps < num_channels) {
        set_err("pfb: num_channels > 2 and taps.len() >= num_channels");
        return nullptr;
    }
    /* channelizer.rs:100-104: oversample_rate must be N/i, i in [1,N] */
    if (!(oversample_rate > 0.f) ||
        fmodf((float)num_channels, oversample_rate) != 0.f) {
        set_err("pfb: oversample rate must be N/i for i in [1, N]");
        return nullptr;
    }
    if ((num_channels & (num_channels - 1)) != 0 || num_channels < 4 ||
        num_channels > 4096) {
        set_err("pfb: num_channels must be a power of two in [4,4096] "
                "(FFT kernel constraint)");
        return nullptr;
    }
    fsdr_filter* f = create_common(K_PFB);
    if (!f) return nullptr;
    f->width = num_channels;
    /* channelizer.rs:105: decimation_factor = N / oversample_rate */
    f->decim = (size_t)((float)num_channels / oversample_rate);
    size_t tpf = (n_taps + num_channels - 1) / num_channels;
    f->history = tpf;
    f->n_taps = n_taps;
    /* partition_filter_taps (utilities.rs:5-25): part[i][c] = taps[i+c*N],
     * zero-padded */
    std::vector<float> part(num_channels * tpf, 0.f);
    for (size_t i = 0; i < num_channels; i++) {
        size_t cnt = 0;
        for (size_t t = i; t < n_taps; t += num_channels)
            part[i * tpf + cnt++] = taps[t];
    }
    if (hipMalloc(&f->d_taps, part.size() * sizeof(float)) != hipSuccess ||
        hipMemcpy(f->d_taps, part.data(), part.size() * sizeof(float),
                  hipMemcpyHostToDevice) != hipSuccess) {
        set_err("pfb taps upload failed");
        delete f;
        return nullptr;
    }
    f->sub = fsdr_fft_cf32_create(num_channels, 1, 0, nullptr);
    if (!f->sub) {
        delete f;
        return nullptr;
    }
    return f;
}

extern "C" fsdr_filter* fsdr_moving_avg_create(size_t width,
                                               float decay_factor,
                                               size_t history) {
    if (width == 0 || history == 0 || !(decay_factor >= 0.f) ||
        decay_factor > 1.f) {
        /* moving_avg.rs:59-62 assert */
        set_err("decay_factor must be in [0,1], width/history > 0");
        return nullptr;
    }
    fsdr_filter* f = create_common(K_MOVAVG);
    if (!f) return nullptr;
    f->width = width;
    f->history = history;
    f->decay = decay_factor;
    f->item_in = f->item_out = 4;
    f->n_taps = width; /* length() = min_items analogue */
    if (hipMalloc(&f->d_avg, width * sizeof(float)) != hipSuccess ||
        hipMemset(f->d_avg, 0, width * sizeof(float)) != hipSuccess) {
        set_err("avg state alloc failed");
        delete f;
        return nullptr;
    }
    return f;
}

extern "C" size_t fsdr_filter_length(const fsdr_filter* f) {
    return f ? (f->kind == K_MAG2 ? 1 : f->n_taps) : 0;
}

/* item sizes for the flowgraph driver (fsdr_fg.cpp) */
extern "C" size_t fsdr_filter_item_sizes(const fsdr_filter* f,
                                         size_t* out_bytes) {
    if (!f) { if (out_bytes) *out_bytes = 8; return 8; }
    if (out_bytes) *out_bytes = f->item_out;
    return f->item_in;
}

extern "C" void fsdr_filter_destroy(fsdr_filter* f) {
    if (!f) return;
    if (f->sub) fsdr_filter_destroy(f->sub);
    if (f->d_avg) (void)hipFree(f->d_avg);
    if (f->d_taps) (void)hipFree(f->d_taps);
    if (f->d_rtaps) (void)hipFree(f->d_rtaps);
    if (f->d_mtaps) (void)hipFree(f->d_mtaps);
    if (f->d_mtaps32) (void)hipFree(f->d_mtaps32);
    if (f->d_twid) (void)hipFree(f->d_twid);
    if (f->d_chirp) (void)hipFree(f->d_chirp);
    if (f->d_B) (void)hipFree(f->d_B);
    if (f->d_in) (void)hipFree(f->d_in);
    if (f->d_out) (void)hipFree(f->d_out);
    if (f->d_scratch) (void)hipFree(f->d_scratch);
    if (f->d_hist) (void)hipFree(f->d_hist);
    delete f;
}

/* ---- kernel launch helpers ---- */

static int grid_for(long long work_items, int block) {
    long long g = (work_items + block - 1) / block;
    const long long cap = 256 * 32; /* 256 CUs, grid-stride beyond */
    if (g > cap) g = cap;
    if (g < 1) g = 1;
    return (int)g;
}

static int launch_fir_cf32(fsdr_filter* f, const void* d_in, void* d_out,
                           size_t n_out, size_t n_in, hipStream_t st) {
    if (n_out == 0) return FSDR_OK;
    long long tiles = ((long long)n_out + FIR_TILE_OUT - 1) / FIR_TILE_OUT;
    /* one tile per block by default: independent blocks overlap their
     * staging latency with other blocks' compute (a per-block tile loop
     * stalls all 4 waves of the block at each staging barrier) */
    long long cap = 256 * 16;
    if (const char* e = getenv("FSDR_FIR_GRID_CAP")) cap = atoll(e);
    int grid = (int)std::min<long long>(tiles, cap);
    const char* mf32 = getenv("FSDR_FIR_MFMA32");
    if (f->kk_mfma && mf32 && atoi(mf32) != 0) {
        /* 32x32 variant reuses d_mtaps when kk32 <= allocated; host
         * uploads a separate array sized for K = T+31 */
        long long tiles32 =
            ((long long)n_out + MFIR32_TILE - 1) / MFIR32_TILE;
        long long cap32 = 256 * 64;
        if (const char* e = getenv("FSDR_FIR_GRID_CAP")) cap32 = atoll(e);
        int grid32 = (int)std::min<long long>(tiles32, cap32);
        unsigned elems32 = MFIR32_TILE + f->kk_mfma32 + 8;
        size_t lds32 = (2 * (size_t)((elems32 + 31u) & ~31u) +
                        f->kk_mfma32 + 36) * sizeof(float);
#define MFIR32_TPL_CASE(KV)                                                      case KV:                                                                         hipLaunchKernelGGL(HIP_KERNEL_NAME(k_fir_mfma32_tpl<KV>),                                       dim3(grid32), dim3(MFIR32_BLOCK), lds32, st,                                 (const float2*)d_in, (float2*)d_out,                                         f->d_mtaps32, (long long)n_out, (long long)n_in);         break;
        switch (f->kk_mfma32) {
            MFIR32_TPL_CASE(48)
            MFIR32_TPL_CASE(96)
            MFIR32_TPL_CASE(160)
            MFIR32_TPL_CASE(288)
            MFIR32_TPL_CASE(544)
            default:
                set_err("bad mfma32 K");
                return FSDR_ERR_INVALID;
        }
#undef MFIR32_TPL_CASE
        HIP_TRY(hipGetLastError());
        return FSDR_OK;
    }
    const char* mf = getenv("FSDR_FIR_MFMA");
    if (f->kk_mfma && (!mf || atoi(mf) != 0)) {
        unsigned elems = MFIR_TILE + f->kk_mfma + 8;
        size_t lds = (2 * (size_t)((elems + 31u) & ~31u) + f->kk_mfma + 20)
                     * sizeof(float);
#define MFIR_TPL_CASE(KV)                                                        case KV:                                                                         hipLaunchKernelGGL(HIP_KERNEL_NAME(k_fir_mfma_tpl<KV>), dim3(grid),                             dim3(MFIR_BLOCK), lds, st, (const float2*)d_in,                              (float2*)d_out, f->d_mtaps, (long long)n_out,                                (long long)n_in);                                         break;
        switch (f->kk_mfma) {
            MFIR_TPL_CASE(32)
            MFIR_TPL_CASE(64)
            MFIR_TPL_CASE(96)
            MFIR_TPL_CASE(144)
            MFIR_TPL_CASE(272)
            MFIR_TPL_CASE(528)
            default:
                set_err("bad mfma K");
                return FSDR_ERR_INVALID;
        }
#undef MFIR_TPL_CASE
        HIP_TRY(hipGetLastError());
        return FSDR_OK;
    }
    if (f->tp_tpl) {
        unsigned elems = FIR_TILE_OUT + f->tp_tpl + 8;
        size_t lds = (2 * (size_t)((elems + 7u) & ~7u) + f->tp_tpl + 3) *
                     sizeof(float);
#define FIR_TPL_CASE(TPV)                                                        case TPV:                                                                        hipLaunchKernelGGL(HIP_KERNEL_NAME(k_fir_cf32_tpl<TPV>),                                        dim3(grid), dim3(FIR_BLOCK), lds, st,                                        (const float2*)d_in, (float2*)d_out, f->d_rtaps,                             (long long)n_out, (long long)n_in);                       break;
        switch (f->tp_tpl) {
            FIR_TPL_CASE(17)
            FIR_TPL_CASE(33)
            FIR_TPL_CASE(65)
            FIR_TPL_CASE(129)
            FIR_TPL_CASE(257)
            FIR_TPL_CASE(513)
            default:
                set_err("bad template tap count");
                return FSDR_ERR_INVALID;
        }
#undef FIR_TPL_CASE
        HIP_TRY(hipGetLastError());
        return FSDR_OK;
    }
    unsigned elems = FIR_TILE_OUT + f->n_taps_padded - 1 + 6;
    size_t lds = (2 * (size_t)plane_floats(elems) + f->n_taps_padded + 4) *
                 sizeof(float);
    hipLaunchKernelGGL(k_fir_cf32, dim3(grid), dim3(FIR_BLOCK), lds, st,
                       (const float2*)d_in, (float2*)d_out, f->d_taps,
                       f->n_taps_padded, (long long)n_out, (long long)n_in);
    HIP_TRY(hipGetLastError());
    return FSDR_OK;
}

static int launch_decim_cf32(fsdr_filter* f, const void* d_in, void* d_out,
                             size_t n_out, size_t n_in, hipStream_t st) {
    if (n_out == 0) return FSDR_OK;
    const char* dmf = getenv("FSDR_DECIM_MFMA");
    if (f->decim == 4 && f->kk_mfma && (!dmf || atoi(dmf) != 0)) {
        long long tiles = ((long long)n_out + MDFIR_TILE - 1) / MDFIR_TILE;
        long long cap = 256 * 64;
        if (const char* e = getenv("FSDR_FIR_GRID_CAP")) cap = atoll(e);
        int grid = (int)std::min<long long>(tiles, cap);
        unsigned elemsP = MDFIR_TILE + f->kk_mfma + 8;
        size_t lds = (4 * (size_t)((elemsP + 31u) & ~31u) +
                      4 * ((size_t)f->kk_mfma + 16)) * sizeof(float);
#define MDFIR_TPL_CASE(KV)                                                       case KV:                                                                         hipLaunchKernelGGL(HIP_KERNEL_NAME(k_decim4_mfma_tpl<KV>),                                      dim3(grid), dim3(MDFIR_BLOCK), lds, st,                                      (const float2*)d_in, (float2*)d_out, f->d_mtaps,                             (long long)n_out, (long long)n_in);                       break;
        switch (f->kk_mfma) {
            MDFIR_TPL_CASE(20)
            MDFIR_TPL_CASE(32)
            MDFIR_TPL_CASE(48)
            MDFIR_TPL_CASE(80)
            MDFIR_TPL_CASE(144)
            default:
                set_err("bad decim mfma K");
                return FSDR_ERR_INVALID;
        }
#undef MDFIR_TPL_CASE
        HIP_TRY(hipGetLastError());
        return FSDR_OK;
    }
    if (f->decim == 4 && f->tp_tpl) {
        long long tiles = ((long long)n_out + DFIRT_TILE - 1) / DFIRT_TILE;
        long long cap = 256 * 64;
        if (const char* e = getenv("FSDR_FIR_GRID_CAP")) cap = atoll(e);
        int grid = (int)std::min<long long>(tiles, cap);
        size_t lds = (4 * (size_t)dfirt_sp(f->tp_tpl) +
                      4 * ((size_t)f->tp_tpl + 4)) * sizeof(float);
#define DFIR_TPL_CASE(TPV)                                                       case TPV:                                                                        hipLaunchKernelGGL(HIP_KERNEL_NAME(k_fir_decim4_tpl<TPV>),                                      dim3(grid), dim3(DFIRT_BLOCK), lds, st,                                      (const float2*)d_in, (float2*)d_out, f->d_rtaps,                             (long long)n_out, (long long)n_in);                       break;
        switch (f->tp_tpl) {
            DFIR_TPL_CASE(8)
            DFIR_TPL_CASE(16)
            DFIR_TPL_CASE(32)
            DFIR_TPL_CASE(64)
            DFIR_TPL_CASE(128)
            default:
                set_err("bad decim template tap count");
                return FSDR_ERR_INVALID;
        }
#undef DFIR_TPL_CASE
        HIP_TRY(hipGetLastError());
        return FSDR_OK;
    }
    if (f->decim == 4) {
        unsigned elems = DFIR_TILE_OUT * 4 + f->n_taps_padded - 1 + 20;
        size_t lds = (2 * (size_t)plane_floats(elems) + f->n_taps_padded + 4)
                     * sizeof(float);
        long long tiles =
            ((long long)n_out + DFIR_TILE_OUT - 1) / DFIR_TILE_OUT;
        int grid = (int)std::min<long long>(tiles, 256 * 8);
        hipLaunchKernelGGL(k_fir_decim4_cf32, dim3(grid), dim3(DFIR_BLOCK),
                           lds, st, (const float2*)d_in, (float2*)d_out,
                           f->d_taps, f->n_taps_padded, (long long)n_out,
                           (long long)n_in);
    } else {
        hipLaunchKernelGGL(k_fir_decim_generic_cf32,
                           dim3(grid_for((long long)n_out, 256)), dim3(256),
                           0, st, (const float2*)d_in, (float2*)d_out,
                           f->d_taps,
                           f->n_taps_padded, /* == n_taps for generic */
                           (long long)f->decim, (long long)n_out,
                           (long long)n_in);
    }
    HIP_TRY(hipGetLastError());
    return FSDR_OK;
}

static int launch_stockham(const float2* d_in, float2* d_out,
                           float* d_mag, const float2* d_twid, int n,
                           size_t frames, int inverse, int fft_shift,
                           float norm, hipStream_t st) {
    int log2n = 0;
    while ((1 << log2n) < n) log2n++;
    int fpb_base = 1024;
    if (const char* e = getenv("FSDR_FFT_FPB_BASE")) fpb_base = atoi(e);
    int fpb = fpb_base / n;
    if (fpb < 1) fpb = 1;
    if (fpb > 16) fpb = 16;
    size_t lds = 2 * (size_t)fpb * n * sizeof(float2);
    long long blocks = ((long long)frames + fpb - 1) / fpb;
    int grid = (int)std::min<long long>(blocks, 256 * 16);
    hipLaunchKernelGGL(k_fft_stockham, dim3(grid), dim3(256), lds, st,
                       d_in, d_out, d_mag, d_twid, n, log2n, fpb, inverse,
                       fft_shift, norm, (long long)frames);
    HIP_TRY(hipGetLastError());
    return FSDR_OK;
}

static int launch_fft(fsdr_filter* f, const void* d_in, void* d_out,
                      size_t frames, hipStream_t st,
                      float* d_mag = nullptr) {
    if (frames == 0) return FSDR_OK;
    int n = (int)f->fft_len;
    if (f->fft_m) { /* Bluestein: pre -> FFT_M -> *B -> IFFT_M -> post */
        int M = (int)f->fft_m;
        int rc = ensure_dev(&f->d_scratch, &f->d_scratch_bytes,
                            2 * frames * (size_t)M * sizeof(float2));
        if (rc) return rc;
        float2* s1 = (float2*)f->d_scratch;
        float2* s2 = s1 + frames * (size_t)M;
        long long tM = (long long)frames * M;
        hipLaunchKernelGGL(k_bluestein_pre, dim3(grid_for(tM, 256)),
                           dim3(256), 0, st, (const float2*)d_in, s1,
                           f->d_chirp, n, M, (long long)frames,
                           f->inverse && f->fft_shift);
        HIP_TRY(hipGetLastError());
        rc = launch_stockham(s1, s2, nullptr, f->d_twid, M, frames, 0, 0,
                             0.f, st);
        if (rc) return rc;
        hipLaunchKernelGGL(k_bluestein_mul, dim3(grid_for(tM, 256)),
                           dim3(256), 0, st, s2, f->d_B, M,
                           (long long)frames);
        HIP_TRY(hipGetLastError());
        rc = launch_stockham(s2, s1, nullptr, f->d_twid, M, frames, 1, 0,
                             0.f, st);
        if (rc) return rc;
        float scale = (1.0f / (float)M) * (f->norm != 0.f ? f->norm : 1.f);
        hipLaunchKernelGGL(k_bluestein_post,
                           dim3(grid_for((long long)frames * n, 256)),
                           dim3(256), 0, st, s1, (float2*)d_out, d_mag,
                           f->d_chirp, n, M, (long long)frames,
                           (!f->inverse) && f->fft_shift, scale);
        HIP_TRY(hipGetLastError());
        return FSDR_OK;
    }
    return launch_stockham((const float2*)d_in, (float2*)d_out, d_mag,
                           f->d_twid, n, frames, f->inverse, f->fft_shift,
                           f->norm, st);
}

/* Bulk batch FFT: the GPU-native many-frames path (what the chain uses
 * internally). The 32-frames-per-work cap in fsdr_filter_dev mirrors the
 * reference block's work() quantum (fft.rs:56); a batch resident in HBM
 * has no reason to launch 512 times. d_mag optional fused |X|^2. */
extern "C" int fsdr_fft_bulk_dev(fsdr_filter* f, const void* d_in,
                                 void* d_out, void* d_mag, size_t frames,
                                 void* stream) {
    REQUIRE_GPU();
    if (!f || f->kind != K_FFT_CF32) {
        set_err("fft_bulk: not an fft filter");
        return FSDR_ERR_INVALID;
    }
    return launch_fft(f, d_in, d_out, frames, (hipStream_t)stream,
                      (float*)d_mag);
}

extern "C" int fsdr_filter_dev(fsdr_filter* f, const void* d_in, size_t n_in,
                               void* d_out, size_t n_out, void* stream,
                               fsdr_filter_result* r) {
    REQUIRE_GPU();
    if (!f || !r) { set_err("null argument"); return FSDR_ERR_INVALID; }
    hipStream_t st = (hipStream_t)stream;
    switch (f->kind) {
        case K_FIR_CF32: {
            *r = fir_status(n_in, f->n_taps, n_out);
            return launch_fir_cf32(f, d_in, d_out, r->produced, n_in, st);
        }
        case K_FIR_F32: {
            *r = fir_status(n_in, f->n_taps, n_out);
            if (r->produced == 0) return FSDR_OK;
            unsigned elems = 1024 + f->n_taps_padded - 1;
            size_t lds = (size_t)elems * sizeof(float);
            long long tiles = ((long long)r->produced + 1023) / 1024;
            int grid = (int)std::min<long long>(tiles, 256 * 16);
            hipLaunchKernelGGL(k_fir_f32, dim3(grid), dim3(256), lds, st,
                               (const float*)d_in, (float*)d_out, f->d_taps,
                               f->n_taps_padded, (long long)r->produced,
                               (long long)n_in);
            HIP_TRY(hipGetLastError());
            return FSDR_OK;
        }
        case K_DECIM_CF32: {
            *r = decim_status(f->decim, n_in, f->n_taps, n_out);
            return launch_decim_cf32(f, d_in, d_out, r->produced, n_in, st);
        }
        case K_RESAMP_CF32: {
            *r = resamp_status(f->interp, f->decim, f->n_taps, n_in, n_out);
            if (r->produced == 0) return FSDR_OK;
            const char* dmf = getenv("FSDR_DECIM_MFMA");
            if (f->sub && f->sub->kk_mfma && (!dmf || atoi(dmf) != 0)) {
                /* 1:4 shape on the MFMA decim kernel: shift the window
                 * origin back by D-1 (that kernel never reads rel < 3,
                 * so the rebased pointer stays in bounds) */
                return launch_decim_cf32(f->sub,
                                         (const float2*)d_in - 3, d_out,
                                         r->produced, n_in + 3, st);
            }
            size_t tpp = f->n_taps / f->interp;
            long long span = (long long)(RS_TILE - 1) * f->decim /
                                 f->interp + tpp + 2;
            size_t lds_t = (2 * (size_t)plane_floats((unsigned)span) +
                            f->n_taps) * sizeof(float);
            const char* rt = getenv("FSDR_RESAMP_TILED");
            if (lds_t <= 64 * 1024 && (!rt || atoi(rt) != 0)) {
                long long tiles =
                    ((long long)r->produced + RS_TILE - 1) / RS_TILE;
                int grid = (int)std::min<long long>(tiles, 256 * 64);
                hipLaunchKernelGGL(k_resamp_tiled_cf32, dim3(grid),
                                   dim3(RS_BLOCK), lds_t, st,
                                   (const float2*)d_in, (float2*)d_out,
                                   f->d_taps, (int)f->n_taps,
                                   (int)f->interp, (int)f->decim,
                                   (long long)r->produced, (long long)n_in,
                                   (int)span);
                HIP_TRY(hipGetLastError());
                return FSDR_OK;
            }
            size_t lds = f->n_taps * sizeof(float);
            hipLaunchKernelGGL(k_resamp_cf32,
                               dim3(grid_for((long long)r->produced, 256)),
                               dim3(256), lds, st, (const float2*)d_in,
                               (float2*)d_out, f->d_taps, (int)f->n_taps,
                               (int)f->interp, (int)f->decim,
                               (long long)r->produced, (long long)n_in);
            HIP_TRY(hipGetLastError());
            return FSDR_OK;
        }
        case K_FFT_CF32: {
            /* fft.rs:169-171 */
            size_t m = n_in < n_out ? n_in : n_out;
            m = (m / f->fft_len) * f->fft_len;
            size_t cap = f->fft_len * 32;
            if (m > cap) m = cap;
            r->consumed = r->produced = m;
            r->status = FSDR_BOTH_SUFFICIENT;
            return launch_fft(f, d_in, d_out, m / f->fft_len, st);
        }
        case K_FIR_CCF32: {
            *r = fir_status(n_in, f->n_taps, n_out);
            if (r->produced == 0) return FSDR_OK;
            unsigned elems = 1024 + f->n_taps - 1;
            size_t lds = ((size_t)elems + f->n_taps) * sizeof(float2);
            long long tiles = ((long long)r->produced + 1023) / 1024;
            int grid = (int)std::min<long long>(tiles, 256 * 64);
            hipLaunchKernelGGL(k_fir_ccf32, dim3(grid), dim3(256), lds, st,
                               (const float2*)d_in, (float2*)d_out,
                               (const float2*)f->d_taps, (int)f->n_taps,
                               (long long)r->produced, (long long)n_in);
            HIP_TRY(hipGetLastError());
            return FSDR_OK;
        }
        case K_XLATING: {
            *r = decim_status(f->decim, n_in, f->n_taps, n_out);
            if (r->produced == 0) return FSDR_OK;
            long long span = (long long)(XT_TILE - 1) * f->decim +
                             f->n_taps + 2;
            size_t lds_t = (2 * (size_t)plane_floats((unsigned)span) +
                            2 * f->n_taps) * sizeof(float);
            const char* xt = getenv("FSDR_XLATING_TILED");
            if (lds_t <= 64 * 1024 && (!xt || atoi(xt) != 0)) {
                long long tiles =
                    ((long long)r->produced + XT_TILE - 1) / XT_TILE;
                int grid = (int)std::min<long long>(tiles, 256 * 64);
                hipLaunchKernelGGL(k_xlating_decim_tiled_ccf32,
                                   dim3(grid), dim3(XT_BLOCK), lds_t, st,
                                   (const float2*)d_in, (float2*)d_out,
                                   (const float2*)f->d_taps,
                                   (int)f->n_taps, (int)f->decim,
                                   (long long)r->produced, (long long)n_in,
                                   (double)f->theta, f->rot_re, f->rot_im,
                                   (int)span);
                HIP_TRY(hipGetLastError());
            } else {
                size_t lds = f->n_taps * sizeof(float2);
                hipLaunchKernelGGL(
                    k_xlating_decim_ccf32,
                    dim3(grid_for((long long)r->produced, 256)), dim3(256),
                    lds, st, (const float2*)d_in, (float2*)d_out,
                    (const float2*)f->d_taps, (int)f->n_taps,
                    (long long)f->decim, (long long)r->produced,
                    (long long)n_in, f->theta, f->rot_re, f->rot_im);
                HIP_TRY(hipGetLastError());
            }
            { /* advance the rotator phase (closed form, f64) */
                double a = (double)f->theta * (double)r->produced;
                double cs = cos(a), sn = sin(a);
                float nr = (float)(cs * f->rot_re - sn * f->rot_im);
                float ni = (float)(cs * f->rot_im + sn * f->rot_re);
                f->rot_re = nr;
                f->rot_im = ni;
            }
            return FSDR_OK;
        }
        case K_PFB:
            set_err("pfb channelizer uses fsdr_pfb_channelizer_run_dev "
                    "(multi-output block)");
            return FSDR_ERR_INVALID;
        case K_MOVAVG: {
            /* replicate the work() counting loop on the host */
            size_t in_frames = n_in / f->width;
            size_t out_frames = n_out / f->width;
            size_t cons = 0, prod = 0, i = f->i_state;
            while (cons < in_frames && prod < out_frames) {
                if (++i == f->history) { i = 0; prod++; }
                cons++;
            }
            r->consumed = cons * f->width;
            r->produced = prod * f->width;
            r->status = FSDR_BOTH_SUFFICIENT;
            if (cons == 0) return FSDR_OK;
            if (prod <= 1 &&
                (prod == 0 || f->i_state + cons == f->history)) {
                /* parallel fast path: chunked EMA + exact composition */
                int cf = 64; /* measured best (16 and 256 both slower) */
                int nch = (int)((cons + cf - 1) / cf);
                if (nch > 1024) { nch = 1024; cf = (int)((cons + nch - 1) / nch); }
                int rc = ensure_dev(&f->d_scratch, &f->d_scratch_bytes,
                                    (size_t)nch * f->width * 4);
                if (rc) return rc;
                hipLaunchKernelGGL(
                    k_moving_avg_chunks,
                    dim3(grid_for((long long)nch * f->width, 256)),
                    dim3(256), 0, st, (const float*)d_in,
                    (float*)f->d_scratch, (int)f->width, (long long)cons,
                    nch, cf, f->decay);
                HIP_TRY(hipGetLastError());
                hipLaunchKernelGGL(k_moving_avg_combine,
                                   dim3((unsigned)f->width), dim3(256), 0,
                                   st, (const float*)f->d_scratch, f->d_avg,
                                   prod ? (float*)d_out : nullptr,
                                   (int)f->width, (long long)cons, nch, cf,
                                   f->decay);
                HIP_TRY(hipGetLastError());
                f->i_state = i;
                return FSDR_OK;
            }
            hipLaunchKernelGGL(k_moving_avg,
                               dim3((unsigned)((f->width + 255) / 256)),
                               dim3(256), 0, st, (const float*)d_in,
                               (float*)d_out, f->d_avg, (int)f->width,
                               (long long)cons, (int)f->i_state,
                               (int)f->history, f->decay);
            HIP_TRY(hipGetLastError());
            f->i_state = i;
            return FSDR_OK;
        }
        case K_MAG2: {
            size_t m = n_in < n_out ? n_in : n_out; /* apply.rs:108 */
            r->consumed = r->produced = m;
            r->status = FSDR_BOTH_SUFFICIENT;
            if (m == 0) return FSDR_OK;
            hipLaunchKernelGGL(k_mag2, dim3(grid_for((long long)m, 256)),
                               dim3(256), 0, st, (const float2*)d_in,
                               (float*)d_out, (long long)m);
            HIP_TRY(hipGetLastError());
            return FSDR_OK;
        }
    }
    set_err("unknown filter kind");
    return FSDR_ERR_INVALID;
}

extern "C" int fsdr_filter_host(fsdr_filter* f, const void* in, size_t n_in,
                                void* out, size_t n_out,
                                fsdr_filter_result* r) {
    REQUIRE_GPU();
    if (!f || !r) { set_err("null argument"); return FSDR_ERR_INVALID; }
    int rc = ensure_dev(&f->d_in, &f->d_in_bytes, n_in * f->item_in + 64);
    if (rc) return rc;
    rc = ensure_dev(&f->d_out, &f->d_out_bytes, n_out * f->item_out + 64);
    if (rc) return rc;
    if (n_in)
        HIP_TRY(hipMemcpy(f->d_in, in, n_in * f->item_in,
                          hipMemcpyHostToDevice));
    rc = fsdr_filter_dev(f, f->d_in, n_in, f->d_out, n_out, nullptr, r);
    if (rc) return rc;
    HIP_TRY(hipStreamSynchronize(nullptr));
    if (r->produced)
        HIP_TRY(hipMemcpy(out, f->d_out, r->produced * f->item_out,
                          hipMemcpyDeviceToHost));
    return FSDR_OK;
}

extern "C" int fsdr_pfb_channelizer_run_dev(fsdr_filter* f,
                                            const void* d_in, size_t n_in,
                                            void* d_out,
                                            size_t out_cap_per_chan,
                                            void* stream,
                                            size_t* produced_per_chan) {
    REQUIRE_GPU();
    if (!f || f->kind != K_PFB) {
        set_err("not a pfb channelizer");
        return FSDR_ERR_INVALID;
    }
    hipStream_t st = (hipStream_t)stream;
    size_t N = f->width, tpf = f->history, D = f->decim;
    size_t prefill = N * tpf;
    size_t steps = n_in > prefill ? (n_in - prefill) / D : 0;
    if (steps > out_cap_per_chan) steps = out_cap_per_chan;
    if (produced_per_chan) *produced_per_chan = steps;
    if (steps == 0) return FSDR_OK;
    int rc = ensure_dev(&f->d_in, &f->d_in_bytes, steps * N * 8);
    if (rc) return rc;
    int rc2 = ensure_dev(&f->d_out, &f->d_out_bytes, steps * N * 8);
    if (rc2) return rc2;
    hipLaunchKernelGGL(k_pfb_dots,
                       dim3(grid_for((long long)(steps * N), 256)),
                       dim3(256), 0, st, (const float2*)d_in,
                       (float2*)f->d_in, (const float*)f->d_taps, (int)N,
                       (int)tpf, (int)D, (long long)steps,
                       (long long)prefill, (long long)0);
    HIP_TRY(hipGetLastError());
    rc = launch_fft(f->sub, f->d_in, f->d_out, steps, st);
    if (rc) return rc;
    hipLaunchKernelGGL(k_pfb_scatter,
                       dim3(grid_for((long long)(steps * N), 256)),
                       dim3(256), 0, st, (const float2*)f->d_out,
                       (float2*)d_out, (int)N, (long long)steps,
                       (long long)out_cap_per_chan);
    HIP_TRY(hipGetLastError());
    return FSDR_OK;
}

/* Streaming channelizer: stateful like the reference block — carries
 * the round-robin window contents (the last N*tpf CONSUMED samples) and
 * the push count across calls, consuming in D-sample quanta after the
 * N*tpf prefill (channelizer.rs:156-212). The unconsumed chunk
 * remainder stays with the caller (pair with fsdr_ring's
 * release_consumed carry). All work is enqueued on `stream`; use one
 * stream per filter. */
extern "C" int fsdr_pfb_channelizer_stream_dev(
    fsdr_filter* f, const void* d_chunk, size_t n, void* d_out,
    size_t out_cap_per_chan, void* stream, size_t* produced_per_chan,
    size_t* consumed) {
    REQUIRE_GPU();
    if (!f || f->kind != K_PFB) {
        set_err("not a pfb channelizer");
        return FSDR_ERR_INVALID;
    }
    hipStream_t st = (hipStream_t)stream;
    size_t N = f->width, tpf = f->history, D = f->decim;
    size_t prefill = N * tpf, W = N * tpf;
    size_t p = f->pfb_pushes;
    size_t T = p < W ? p : W;
    size_t fill = p < prefill ? std::min(prefill - p, n) : 0;
    size_t steps = (p + n >= prefill) ? (p + n - prefill) / D -
                                            (p > prefill ? (p - prefill) / D
                                                         : 0)
                                      : 0;
    if (steps > out_cap_per_chan) steps = out_cap_per_chan;
    size_t cons = fill + steps * D;
    if (cons > n) cons = n; /* fill-only calls */
    if (produced_per_chan) *produced_per_chan = steps;
    if (consumed) *consumed = cons;
    if (cons == 0) return FSDR_OK;
    /* work buffer = [hist(T), chunk(cons)] */
    int rc = ensure_dev(&f->d_scratch, &f->d_scratch_bytes,
                        (T + cons) * 8);
    if (rc) return rc;
    if (T)
        HIP_TRY(hipMemcpyAsync(f->d_scratch, f->d_hist, T * 8,
                               hipMemcpyDeviceToDevice, st));
    HIP_TRY(hipMemcpyAsync((char*)f->d_scratch + T * 8, d_chunk, cons * 8,
                           hipMemcpyDeviceToDevice, st));
    if (steps) {
        rc = ensure_dev(&f->d_in, &f->d_in_bytes, steps * N * 8);
        if (rc) return rc;
        rc = ensure_dev(&f->d_out, &f->d_out_bytes, steps * N * 8);
        if (rc) return rc;
        hipLaunchKernelGGL(k_pfb_dots,
                           dim3(grid_for((long long)(steps * N), 256)),
                           dim3(256), 0, st, (const float2*)f->d_scratch,
                           (float2*)f->d_in, (const float*)f->d_taps,
                           (int)N, (int)tpf, (int)D, (long long)steps,
                           (long long)(p + fill), (long long)(p - T));
        HIP_TRY(hipGetLastError());
        rc = launch_fft(f->sub, f->d_in, f->d_out, steps, st);
        if (rc) return rc;
        hipLaunchKernelGGL(k_pfb_scatter,
                           dim3(grid_for((long long)(steps * N), 256)),
                           dim3(256), 0, st, (const float2*)f->d_out,
                           (float2*)d_out, (int)N, (long long)steps,
                           (long long)out_cap_per_chan);
        HIP_TRY(hipGetLastError());
    }
    /* new hist = last min(W, T+cons) samples of the work buffer */
    size_t newT = std::min(W, T + cons);
    rc = ensure_dev(&f->d_hist, &f->d_hist_bytes, W * 8);
    if (rc) return rc;
    HIP_TRY(hipMemcpyAsync(f->d_hist,
                           (char*)f->d_scratch + (T + cons - newT) * 8,
                           newT * 8, hipMemcpyDeviceToDevice, st));
    f->pfb_pushes = p + cons;
    return FSDR_OK;
}

extern "C" int fsdr_divide_mag_dev(const void* d_a, size_t n_a,
                                   const void* d_b, size_t n_b,
                                   void* d_out, size_t n_out, void* stream,
                                   size_t* m) {
    REQUIRE_GPU();
    size_t mm = n_a < n_b ? n_a : n_b;
    if (n_out < mm) mm = n_out;
    if (m) *m = mm;
    if (mm == 0) return FSDR_OK;
    hipLaunchKernelGGL(k_divide_mag, dim3(grid_for((long long)mm, 256)),
                       dim3(256), 0, (hipStream_t)stream,
                       (const float2*)d_a, (const float*)d_b,
                       (float*)d_out, (long long)mm);
    HIP_TRY(hipGetLastError());
    return FSDR_OK;
}

extern "C" int fsdr_cmul_conj_dev(const void* d_a, size_t n_a,
                                  const void* d_b, size_t n_b, void* d_out,
                                  size_t n_out, void* stream, size_t* m) {
    REQUIRE_GPU();
    size_t mm = n_a < n_b ? n_a : n_b;
    if (n_out < mm) mm = n_out;
    if (m) *m = mm;
    if (mm == 0) return FSDR_OK;
    hipLaunchKernelGGL(k_cmul_conj, dim3(grid_for((long long)mm, 256)),
                       dim3(256), 0, (hipStream_t)stream,
                       (const float2*)d_a, (const float2*)d_b,
                       (float2*)d_out, (long long)mm);
    HIP_TRY(hipGetLastError());
    return FSDR_OK;
}

/* WLAN MovingAverage, one-shot from fresh state: emits len-1 zero items
 * then sliding sums; is_complex selects cf32 vs f32 items. */
extern "C" int fsdr_wlan_moving_sum_dev(const void* d_in, size_t n_in,
                                        void* d_out, size_t n_out,
                                        size_t len, int is_complex,
                                        void* stream, size_t* produced) {
    REQUIRE_GPU();
    if (len == 0) { set_err("len must be > 0"); return FSDR_ERR_INVALID; }
    hipStream_t st = (hipStream_t)stream;
    size_t w = is_complex ? 2 : 1;
    size_t pad = len - 1 < n_out ? len - 1 : n_out;
    if (pad)
        HIP_TRY(hipMemsetAsync(d_out, 0, pad * w * sizeof(float), st));
    size_t m = n_in + 1 > len ? n_in + 1 - len : 0;
    if (m > n_out - pad) m = n_out - pad;
    if (produced) *produced = pad + m;
    if (m == 0) return FSDR_OK;
    hipLaunchKernelGGL(k_moving_sum,
                       dim3(grid_for((long long)(m * w), 256)), dim3(256),
                       0, st, (const float*)d_in,
                       (float*)d_out + pad * w, (int)w, (int)len,
                       (long long)m);
    HIP_TRY(hipGetLastError());
    return FSDR_OK;
}

extern "C" int fsdr_cmul_dev(const void* d_a, size_t n_a, const void* d_b,
                             size_t n_b, void* d_out, size_t n_out,
                             void* stream, size_t* m) {
    REQUIRE_GPU();
    size_t mm = n_a < n_b ? n_a : n_b; /* combine.rs:115-116 */
    if (n_out < mm) mm = n_out;
    if (m) *m = mm;
    if (mm == 0) return FSDR_OK;
    hipLaunchKernelGGL(k_cmul, dim3(grid_for((long long)mm, 256)), dim3(256),
                       0, (hipStream_t)stream, (const float2*)d_a,
                       (const float2*)d_b, (float2*)d_out, (long long)mm);
    HIP_TRY(hipGetLastError());
    return FSDR_OK;
}

extern "C" int fsdr_cmul_host(const void* a, size_t n_a, const void* b,
                              size_t n_b, void* out, size_t n_out,
                              size_t* m) {
    REQUIRE_GPU();
    size_t mm = n_a < n_b ? n_a : n_b;
    if (n_out < mm) mm = n_out;
    void *da = nullptr, *db = nullptr, *do_ = nullptr;
    HIP_TRY(hipMalloc(&da, (mm ? mm : 1) * 8));
    HIP_TRY(hipMalloc(&db, (mm ? mm : 1) * 8));
    HIP_TRY(hipMalloc(&do_, (mm ? mm : 1) * 8));
    if (mm) {
        HIP_TRY(hipMemcpy(da, a, mm * 8, hipMemcpyHostToDevice));
        HIP_TRY(hipMemcpy(db, b, mm * 8, hipMemcpyHostToDevice));
    }
    int rc = fsdr_cmul_dev(da, mm, db, mm, do_, mm, nullptr, m);
    if (rc == FSDR_OK && mm) {
        HIP_TRY(hipStreamSynchronize(nullptr));
        HIP_TRY(hipMemcpy(out, do_, mm * 8, hipMemcpyDeviceToHost));
    }
    (void)hipFree(da); (void)hipFree(db); (void)hipFree(do_);
    return rc;
}

/* ---- device memory helpers ---- */

extern "C" int fsdr_dev_alloc(void** d_ptr, size_t bytes) {
    REQUIRE_GPU();
    HIP_TRY(hipMalloc(d_ptr, bytes));
    return FSDR_OK;
}
extern "C" int fsdr_dev_free(void* d_ptr) {
    REQUIRE_GPU();
    HIP_TRY(hipFree(d_ptr));
    return FSDR_OK;
}
extern "C" int fsdr_memcpy_h2d(void* d_dst, const void* src, size_t bytes) {
    REQUIRE_GPU();
    HIP_TRY(hipMemcpy(d_dst, src, bytes, hipMemcpyHostToDevice));
    return FSDR_OK;
}
extern "C" int fsdr_memcpy_d2h(void* dst, const void* d_src, size_t bytes) {
    REQUIRE_GPU();
    HIP_TRY(hipMemcpy(dst, d_src, bytes, hipMemcpyDeviceToHost));
    return FSDR_OK;
}
extern "C" int fsdr_fill_uniform_cf32(void* d_ptr, size_t n, uint64_t seed,
                                      uint64_t offset, void* stream) {
    REQUIRE_GPU();
    hipLaunchKernelGGL(k_fill_uniform_cf32,
                       dim3(grid_for((long long)(n / 2 + 1), 256)),
                       dim3(256), 0, (hipStream_t)stream, (float2*)d_ptr,
                       (long long)n, seed, offset);
    HIP_TRY(hipGetLastError());
    return FSDR_OK;
}

/* ================= firdes (host-side) ================================= *
 * Product restatement of the reference tap designers (host code in the
 * reference too): firdes/basic.rs:25-42,310-321,444-459, windows.rs:144,
 * special_funs.rs:22-45. Independent of oracle/ (test infra). */

static double besseli0_h(double x) {
    double t = x / 3.75;
    if (fabs(x) <= 3.75) {
        return 1.0 + 3.5156229 * pow(t, 2.0) + 3.0899424 * pow(t, 4.0) +
               1.2067492 * pow(t, 6.0) + 0.2659732 * pow(t, 8.0) +
               0.0360768 * pow(t, 10.0) + 0.0045813 * pow(t, 12.0);
    }
    return 1.0 / (sqrt(fabs(x)) * exp(-x)) *
           (0.39894228 + 0.01328592 * pow(t, -1.0) + 0.00225319 * pow(t, -2.0) -
            0.00157565 * pow(t, -3.0) + 0.00916281 * pow(t, -4.0) -
            0.02057706 * pow(t, -5.0) + 0.02635537 * pow(t, -6.0) -
            0.01647633 * pow(t, -7.0) + 0.00392377 * pow(t, -8.0));
}

extern "C" double fsdr_kaiser_beta(double max_ripple) {
    double ripple_db = -20.0 * log10(max_ripple);
    if (ripple_db > 50.0) return 0.1102 * (ripple_db - 8.7);
    if (ripple_db >= 21.0)
        return 0.5842 * pow(ripple_db - 21.0, 0.4) +
               0.07886 * (ripple_db - 21.0);
    return 0.0;
}

extern "C" void fsdr_kaiser_window(size_t len, double beta, double* out) {
    double alpha = (double)(len - 1) / 2.0;
    double denom = besseli0_h(beta);
    for (size_t n = 0; n < len; n++) {
        double q = ((double)n - alpha) / alpha;
        out[n] = besseli0_h(beta * sqrt(1.0 - q * q)) / denom;
    }
}

static void firdes_lowpass_h(double cutoff, const double* win, size_t len,
                             double* out) {
    double omega_c = 2.0 * M_PI * cutoff;
    double alpha = (double)(len - 1) / 2.0;
    for (size_t n = 0; n < len; n++) {
        double x = (double)n - alpha;
        double ft = (x == 0.0) ? omega_c / M_PI : sin(omega_c * x) / (M_PI * x);
        out[n] = win[n] * ft;
    }
}

extern "C" size_t fsdr_firdes_kaiser_lowpass_f32(double cutoff,
                                                 double transition_bw,
                                                 double max_ripple,
                                                 float* out, size_t cap) {
    double beta = fsdr_kaiser_beta(max_ripple);
    double ripple_db = -20.0 * log10(max_ripple);
    size_t num_taps =
        (size_t)(ceil((ripple_db - 7.95) / (14.36 * transition_bw)) + 1.0);
    if (!out || cap < num_taps) return num_taps;
    std::vector<double> win(num_taps), taps(num_taps);
    fsdr_kaiser_window(num_taps, beta, win.data());
    double omega_c = (2.0 * cutoff + transition_bw) / 2.0; /* basic.rs:319 */
    firdes_lowpass_h(omega_c, win.data(), num_taps, taps.data());
    for (size_t i = 0; i < num_taps; i++) out[i] = (float)taps[i];
    return num_taps;
}

extern "C" int fsdr_firdes_lowpass_kaiser_n_f32(size_t n_taps, double beta,
                                                double cutoff, float* out) {
    if (!out || n_taps < 2) return FSDR_ERR_INVALID;
    std::vector<double> win(n_taps), taps(n_taps);
    fsdr_kaiser_window(n_taps, beta, win.data());
    firdes_lowpass_h(cutoff, win.data(), n_taps, taps.data());
    for (size_t i = 0; i < n_taps; i++) out[i] = (float)taps[i];
    return FSDR_OK;
}

extern "C" int fsdr_rotator_dev(const void* d_in, void* d_out, size_t n,
                                float phase_incr_angle, float phase0_re,
                                float phase0_im, void* stream,
                                float* final_re, float* final_im) {
    REQUIRE_GPU();
    if (n > 0) {
        hipLaunchKernelGGL(k_rotator, dim3(grid_for((long long)n, 256)),
                           dim3(256), 0, (hipStream_t)stream,
                           (const float2*)d_in, (float2*)d_out, (long long)n,
                           phase_incr_angle, phase0_re, phase0_im);
        HIP_TRY(hipGetLastError());
    }
    if (final_re && final_im) { /* closed-form final phase (f64) */
        double a = (double)phase_incr_angle * (double)n;
        double c = cos(a), s = sin(a);
        *final_re = (float)(c * phase0_re - s * phase0_im);
        *final_im = (float)(c * phase0_im + s * phase0_re);
    }
    return FSDR_OK;
}

/* ================= WLAN rx front end (config 5) ======================= *
 * Host-side state machines mirroring examples/wlan/src/sync_short.rs
 * (:2-5 constants THRESHOLD 0.56 / MIN_GAP 480 / MAX_SAMPLES 540*80;
 * :92-150 Search/Found/Copy loop) and sync_long.rs (:3 SEARCH_WINDOW
 * 320; :18-50 Correlator::sync; :136-178 Sync/Copy). The sequential
 * state machines run on the host exactly like the reference blocks; the
 * SyncLong correlator's 320 64-tap complex dot products run on the GPU
 * (k_fir_ccf32 through an internal filter handle). */

/* the 802.11a long-training correlator taps — sync_long.rs:188-253 */
static const float2 WLAN_LONG[64] = {
    {1.3868f, -0.0000f},  {-0.0455f, 1.0679f},  {0.3528f, 0.9865f},
    {0.8594f, -0.7348f},  {0.1874f, -0.2475f},  {0.5309f, 0.7784f},
    {-1.0218f, 0.4897f},  {-0.3401f, 0.9423f},  {0.8657f, 0.2298f},
    {0.4734f, -0.0362f},  {0.0088f, 1.0207f},   {-1.2142f, 0.4205f},
    {0.2172f, 0.5195f},   {0.5207f, 0.1326f},   {-0.1995f, -1.4259f},
    {1.0583f, 0.0363f},   {0.5547f, 0.5547f},   {0.3277f, -0.8728f},
    {-0.5077f, -0.3488f}, {-1.1650f, -0.5789f}, {0.7297f, -0.8197f},
    {0.6173f, -0.1253f},  {-0.5353f, -0.7214f}, {-0.5011f, 0.1935f},
    {-0.3110f, 1.3392f},  {-1.0818f, 0.1470f},  {-1.1300f, 0.1820f},
    {0.6663f, 0.6571f},   {-0.0249f, -0.4773f}, {-0.8155f, -1.0218f},
    {0.8140f, -0.9396f},  {0.1090f, -0.8662f},  {-1.3868f, -0.0000f},
    {0.1090f, 0.8662f},   {0.8140f, 0.9396f},   {-0.8155f, 1.0218f},
    {-0.0249f, 0.4773f},  {0.6663f, -0.6571f},  {-1.1300f, -0.1820f},
    {-1.0818f, -0.1470f}, {-0.3110f, -1.3392f}, {-0.5011f, -0.1935f},
    {-0.5353f, 0.7214f},  {0.6173f, 0.1253f},   {0.7297f, 0.8197f},
    {-1.1650f, 0.5789f},  {-0.5077f, 0.3488f},  {0.3277f, 0.8728f},
    {0.5547f, -0.5547f},  {1.0583f, -0.0363f},  {-0.1995f, 1.4259f},
    {0.5207f, -0.1326f},  {0.2172f, -0.5195f},  {-1.2142f, -0.4205f},
    {0.0088f, -1.0207f},  {0.4734f, 0.0362f},   {0.8657f, -0.2298f},
    {-0.3401f, -0.9423f}, {-1.0218f, -0.4897f}, {0.5309f, -0.7784f},
    {0.1874f, 0.2475f},   {0.8594f, 0.7348f},   {0.3528f, -0.9865f},
    {-0.0455f, -1.0679f},
};

#define WLAN_THRESHOLD 0.56f
#define WLAN_MIN_GAP 480
#define WLAN_MAX_SAMPLES (540 * 80)
#define WLAN_SEARCH_WINDOW 320

struct fsdr_wlan_rx {
    /* SyncShort state (sync_short.rs:8-12,34) */
    int ss_state = 0;      /* 0 Search, 1 Found, 2 Copy */
    size_t ss_copied = 0;
    float ss_foffset = 0.f;
    bool ss_above = false;
    bool ss_pending = false;
    float ss_pending_freq = 0.f;
    /* SyncLong GPU correlator */
    fsdr_filter* corr = nullptr;
    void* d_win = nullptr;
    void* d_cor = nullptr;
};

extern "C" fsdr_wlan_rx* fsdr_wlan_rx_create(void) {
    if (!have_gpu()) { set_err("no HIP device"); return nullptr; }
    fsdr_wlan_rx* rx = new fsdr_wlan_rx();
    /* FirCC computes y[i] = sum_t in[i+t]*taps[63-t]; the reference
     * correlator is sum_k in[i+k]*LONG[k] (sync_long.rs:21-27), so the
     * filter taps are LONG reversed. */
    fsdr_cf32 rev[64];
    for (int i = 0; i < 64; i++) {
        rev[i].re = WLAN_LONG[63 - i].x;
        rev[i].im = WLAN_LONG[63 - i].y;
    }
    rx->corr = fsdr_fir_ccf32_create(rev, 64);
    if (!rx->corr ||
        hipMalloc(&rx->d_win, (WLAN_SEARCH_WINDOW + 64) * sizeof(float2)) !=
            hipSuccess ||
        hipMalloc(&rx->d_cor, WLAN_SEARCH_WINDOW * sizeof(float2)) !=
            hipSuccess) {
        set_err("wlan rx alloc failed");
        fsdr_filter_destroy(rx->corr);
        if (rx->d_win) (void)hipFree(rx->d_win);
        delete rx;
        return nullptr;
    }
    return rx;
}

extern "C" void fsdr_wlan_rx_destroy(fsdr_wlan_rx* rx) {
    if (!rx) return;
    fsdr_filter_destroy(rx->corr);
    if (rx->d_win) (void)hipFree(rx->d_win);
    if (rx->d_cor) (void)hipFree(rx->d_cor);
    delete rx;
}

/* SyncShort actor loop (sync_short.rs:92-150) over aligned spans of the
 * delayed signal, the 48-sample complex autocorrelation average, and
 * the correlation metric. Emits the frame-sample stream plus
 * "wifi_start" tags (output index + coarse freq offset). Returns
 * produced; *consumed_out = input samples consumed (== n unless out or
 * tag capacity limited). */
extern "C" size_t fsdr_wlan_sync_short_run(
    fsdr_wlan_rx* rx, const fsdr_cf32* sig, const fsdr_cf32* abs48,
    const float* cor, size_t n, fsdr_cf32* out, size_t out_cap,
    size_t* tag_idx, float* tag_freq, size_t tag_cap, size_t* n_tags,
    size_t* consumed_out) {
    size_t i = 0, o = 0, nt = 0;
    while (i < n && o < out_cap) {
        switch (rx->ss_state) {
            case 0: /* Search */
                if (cor[i] > WLAN_THRESHOLD) rx->ss_state = 1;
                break;
            case 1: /* Found */
                if (cor[i] > WLAN_THRESHOLD) {
                    float fo = -atan2f(abs48[i].im, abs48[i].re) / 16.0f;
                    rx->ss_state = 2;
                    rx->ss_copied = 0;
                    rx->ss_foffset = fo;
                    rx->ss_above = false;
                    rx->ss_pending = true;
                    rx->ss_pending_freq = fo;
                } else {
                    rx->ss_state = 0;
                }
                break;
            case 2: { /* Copy(n_copied, f_offset, last_above) */
                if (cor[i] > WLAN_THRESHOLD) {
                    if (rx->ss_above && rx->ss_copied > WLAN_MIN_GAP) {
                        /* resync (sync_short.rs:110-117) */
                        float fo =
                            -atan2f(abs48[i].im, abs48[i].re) / 16.0f;
                        rx->ss_copied = 0;
                        rx->ss_foffset = fo;
                        rx->ss_above = false;
                        rx->ss_pending = true;
                        rx->ss_pending_freq = fo;
                        i++;
                        continue;
                    }
                    rx->ss_above = true;
                } else {
                    rx->ss_above = false;
                }
                if (rx->ss_copied == 0 && rx->ss_pending) {
                    if (nt < tag_cap) {
                        tag_idx[nt] = o;
                        tag_freq[nt] = rx->ss_pending_freq;
                        nt++;
                    }
                    rx->ss_pending = false;
                }
                float ang = rx->ss_foffset * (float)rx->ss_copied;
                float s, c;
                __builtin_sincosf(ang, &s, &c);
                out[o].re = sig[i].re * c - sig[i].im * s;
                out[o].im = sig[i].re * s + sig[i].im * c;
                o++;
                if (rx->ss_copied + 1 == WLAN_MAX_SAMPLES)
                    rx->ss_state = 0;
                else
                    rx->ss_copied++;
                break;
            }
        }
        i++;
    }
    if (n_tags) *n_tags = nt;
    if (consumed_out) *consumed_out = i;
    return o;
}

/* SyncLong over a tagged span (sync_long.rs:96-185): for each
 * "wifi_start" tag, run the GPU 64-tap correlator over the
 * SEARCH_WINDOW, pick the top-2 |cor|^2 peaks (stable order, pair
 * sorted by index — :38-48), copy 128 samples rotated by the fine freq
 * offset, then strip the 16-sample CP from each 80-sample symbol until
 * the next tag (leftover < 80 dropped, like the reference's m<80
 * consume). Needs a GPU. Returns symbols*64 (+128/frame) samples
 * written; frame_off[f] = the correlator offset chosen for frame f. */
extern "C" size_t fsdr_wlan_sync_long_run(
    fsdr_wlan_rx* rx, const fsdr_cf32* in, size_t n,
    const size_t* tag_idx, const float* tag_freq, size_t num_tags,
    fsdr_cf32* out, size_t out_cap, size_t* frame_off, float* frame_freq,
    size_t frame_cap, size_t* num_frames) {
    (void)tag_freq;
    size_t o = 0, nf = 0;
    for (size_t t = 0; t < num_tags; t++) {
        size_t T = tag_idx[t];
        size_t T2 = (t + 1 < num_tags) ? tag_idx[t + 1] : n;
        if (T2 - T < WLAN_SEARCH_WINDOW + 128) continue; /* :143 */
        /* Correlator::sync on in[T .. T+SEARCH_WINDOW+63] (GPU) */
        if (hipMemcpy(rx->d_win, in + T,
                      (WLAN_SEARCH_WINDOW + 63) * sizeof(float2),
                      hipMemcpyHostToDevice) != hipSuccess) {
            set_err("wlan h2d failed");
            break;
        }
        fsdr_filter_result r;
        if (fsdr_filter_dev(rx->corr, rx->d_win, WLAN_SEARCH_WINDOW + 63,
                            rx->d_cor, WLAN_SEARCH_WINDOW, nullptr,
                            &r) != FSDR_OK)
            break;
        (void)hipStreamSynchronize(nullptr);
        float2 corv[WLAN_SEARCH_WINDOW];
        if (hipMemcpy(corv, rx->d_cor,
                      WLAN_SEARCH_WINDOW * sizeof(float2),
                      hipMemcpyDeviceToHost) != hipSuccess)
            break;
        /* top-2 by |cor|^2, stable (first occurrence wins ties), pair
         * ordered by index (:38-44) */
        int i0 = 0, i1 = -1;
        float m0 = -1.f, m1 = -1.f;
        for (int i = 0; i < WLAN_SEARCH_WINDOW; i++) {
            float m = corv[i].x * corv[i].x + corv[i].y * corv[i].y;
            if (m > m0) {
                m1 = m0; i1 = i0;
                m0 = m; i0 = i;
            } else if (m > m1) {
                m1 = m; i1 = i;
            }
        }
        int first = i0 < i1 ? i0 : i1;
        int second = i0 < i1 ? i1 : i0;
        /* freq = arg(cor[first] * conj(cor[second])) / 64  (:46-48) */
        float2 a = corv[first], b = corv[second];
        float pr = a.x * b.x + a.y * b.y;
        float pi = a.y * b.x - a.x * b.y;
        float freq = atan2f(pi, pr) / 64.0f;
        size_t off = (size_t)first;
        if (o + 128 > out_cap) break;
        for (int i = 0; i < 128; i++) { /* :147-151 */
            float s, c;
            __builtin_sincosf((float)i * freq, &s, &c);
            fsdr_cf32 x = in[T + off + i];
            out[o + i].re = x.re * c - x.im * s;
            out[o + i].im = x.re * s + x.im * c;
        }
        o += 128;
        if (nf < frame_cap) {
            if (frame_off) frame_off[nf] = off;
            if (frame_freq) frame_freq[nf] = freq;
            nf++;
        }
        /* Copy state (:162-177): 80-sample symbols -> 64 samples each */
        size_t cur = T + off + 128;
        size_t n_copied = 0;
        while (cur + 80 <= T2 && o + 64 <= out_cap) {
            for (int k = 0; k < 64; k++) {
                float ang =
                    (float)(n_copied * 80 + 128 + 16 + (size_t)k) * freq;
                float s, c;
                __builtin_sincosf(ang, &s, &c);
                fsdr_cf32 x = in[cur + 16 + k];
                out[o + k].re = x.re * c - x.im * s;
                out[o + k].im = x.re * s + x.im * c;
            }
            o += 64;
            cur += 80;
            n_copied++;
        }
    }
    if (num_frames) *num_frames = nf;
    return o;
}

/* ================= chain ============================================== */

struct fsdr_chain {
    fsdr_filter* fir1 = nullptr;
    fsdr_filter* fir2 = nullptr;
    fsdr_filter* fused = nullptr; /* combined-taps decimating FIR */
    fsdr_filter* fft = nullptr;
    float2* d_y1 = nullptr;
    float2* d_y2 = nullptr;
    float2* d_null = nullptr; /* NullSink scratch when caller passes NULL */
    size_t y1_cap = 0, y2_cap = 0, null_cap = 0;
};

extern "C" fsdr_chain* fsdr_chain_create(const float* taps1, size_t n_taps1,
                                         const float* taps2, size_t n_taps2,
                                         size_t decim, size_t fft_len) {
    fsdr_chain* c = new fsdr_chain();
    c->fir1 = fsdr_fir_cf32_create(taps1, n_taps1);
    c->fir2 = fsdr_decim_fir_cf32_create(decim, taps2, n_taps2);
    c->fft = fsdr_fft_cf32_create(fft_len, 0, 0, nullptr);
    /* Algebraic fusion (default on, FSDR_CHAIN_FUSED=0 for the two-stage
     * path): Fir(h1) then DecimatingFir(D, h2) is one DecimatingFir(D, g)
     * with g = h1 (*) h2 (composition of LTI filters; combined taps in
     * f64). Output count is identical: (n+1-(T1+T2-1))/D == the composed
     * two-stage count. Parity vs the two-stage oracle is covered by the
     * chain tests (rel l2 1e-4). Cuts chain arithmetic from
     * (T1*4 + T2) to ~(T1+T2)*1 flops per input sample and removes the
     * y1 HBM round trip (30 -> ~12 B/sample). */
    const char* fz = getenv("FSDR_CHAIN_FUSED");
    if (!fz || atoi(fz) != 0) {
        size_t tg = n_taps1 + n_taps2 - 1;
        std::vector<double> g(tg, 0.0);
        for (size_t a = 0; a < n_taps1; a++)
            for (size_t b = 0; b < n_taps2; b++)
                g[a + b] += (double)taps1[a] * (double)taps2[b];
        std::vector<float> gf(tg);
        for (size_t i = 0; i < tg; i++) gf[i] = (float)g[i];
        c->fused = fsdr_decim_fir_cf32_create(decim, gf.data(), tg);
    }
    if (!c->fir1 || !c->fir2 || !c->fft) {
        fsdr_chain_destroy(c);
        return nullptr;
    }
    return c;
}

extern "C" void fsdr_chain_destroy(fsdr_chain* c) {
    if (!c) return;
    fsdr_filter_destroy(c->fir1);
    fsdr_filter_destroy(c->fir2);
    fsdr_filter_destroy(c->fused);
    fsdr_filter_destroy(c->fft);
    if (c->d_y1) (void)hipFree(c->d_y1);
    if (c->d_y2) (void)hipFree(c->d_y2);
    if (c->d_null) (void)hipFree(c->d_null);
    delete c;
}

extern "C" int fsdr_chain_run_dev(fsdr_chain* c, const void* d_in,
                                  size_t n_in, void* d_out, size_t out_cap,
                                  void* d_mag, size_t mag_cap, void* stream,
                                  size_t* consumed, size_t* produced) {
    REQUIRE_GPU();
    if (!c) { set_err("null chain"); return FSDR_ERR_INVALID; }
    hipStream_t st = (hipStream_t)stream;
    const size_t nt1 = c->fir1->n_taps, nt2 = c->fir2->n_taps;
    const size_t D = c->fir2->decim, L = c->fft->fft_len;
    size_t y1 = sat_sub(n_in + 1, nt1);
    size_t y2 = sat_sub(y1 + 1, nt2) / D;
    size_t frames = y2 / L;
    if (d_out && out_cap < frames * L) frames = out_cap / L;
    if (d_mag && mag_cap < frames * L) frames = mag_cap / L;
    size_t prod = frames * L;
    if (consumed) *consumed = prod * D; /* chain-input samples per frame set */
    if (produced) *produced = prod;
    if (frames == 0) return FSDR_OK;
    int rc;
    const char* ffz = getenv("FSDR_CHAIN_FFTFUSE");
    const bool fft_fusable =
        (L == 64 || L == 128 || L == 256 || L == 512 || L == 1024);
    if (c->fused && fft_fusable && c->fused->kk_mfma &&
        !c->fft->inverse && !c->fft->fft_shift && c->fft->norm == 0.f &&
        (!ffz || atoi(ffz) != 0)) {
        /* single kernel: fused decimating filter + in-block per-frame
         * FFT (+ optional |X|^2) — y2 never touches HBM. Tile = 1024
         * decimated outputs = 1024/L frames. */
        const int KK = c->fused->kk_mfma;
        long long tiles = ((long long)prod + MDFIR_TILE - 1) / MDFIR_TILE;
        long long cap = 8192; /* ~2 tiles/block at 2^26: best measured */
        if (const char* e = getenv("FSDR_FIR_GRID_CAP")) cap = atoll(e);
        int grid = (int)std::min<long long>(tiles, cap);
        unsigned elemsP = MDFIR_TILE + KK + 8;
        size_t lds = (4 * (size_t)((elemsP + 31u) & ~31u) +
                      4 * ((size_t)KK + 16)) * sizeof(float);
        float2* spec_dst = (float2*)d_out; /* null + mag-only: skip the
                                              discarded spectra write */
        if (!spec_dst && !d_mag) { /* NullSink scratch keeps work observable */
            rc = ensure_dev((void**)&c->d_null, &c->null_cap,
                            (prod + 8) * sizeof(float2));
            if (rc) return rc;
            spec_dst = c->d_null;
        }
        const char* b512 = getenv("FSDR_CHAIN_BLOCK512");
        if (b512 && atoi(b512) != 0 && L == 1024 && KK == 80 &&
            prod % 2048 == 0) {
            long long tiles2 = (long long)prod / 2048;
            int grid2 = (int)std::min<long long>(tiles2, cap);
            unsigned eP2 = 2048 + 80 + 8;
            size_t lds2 = (4 * (size_t)((eP2 + 31u) & ~31u) + 4 * (80 + 16))
                          * sizeof(float);
            hipLaunchKernelGGL(HIP_KERNEL_NAME(k_decim4_fft_mfma2_tpl<80>),
                               dim3(grid2), dim3(512), lds2, st,
                               (const float2*)d_in, spec_dst,
                               c->fused->d_mtaps, (long long)prod,
                               (long long)n_in,
                               (const float2*)c->fft->d_twid,
                               (float*)d_mag);
            HIP_TRY(hipGetLastError());
            return FSDR_OK;
        }
        const char* ws = getenv("FSDR_CHAIN_WS");
        /* software-pipelined FFT-in-MFMA-shadow variant: measured ~4%
         * faster than the ap kernel at fft_len 1024 (the serial FFT
         * issues inside the MFMA pipe shadow) — default for the
         * aligned 1024 case; FSDR_CHAIN_WS=0 disables. */
        if ((!ws || atoi(ws) != 0) && L == 1024 &&
            ((uintptr_t)d_in & 15u) == 0) {
            unsigned SPm = (elemsP + 31u) & ~31u;
            size_t lds_ws = (8 * (size_t)SPm + 4 * ((size_t)KK + 16)) *
                                sizeof(float) +
                            1024 * sizeof(float2);
#define CHAIN_WS_CASE(KV)                                                 \
    case KV:                                                              \
        hipLaunchKernelGGL(HIP_KERNEL_NAME(k_decim4_fft_mfma_ws_tpl<KV>), \
                           dim3(grid), dim3(MDFIR_BLOCK), lds_ws, st,     \
                           (const float2*)d_in, spec_dst,                 \
                           c->fused->d_mtaps, (long long)prod,            \
                           (long long)n_in,                               \
                           (const float2*)c->fft->d_twid, (float*)d_mag); \
        break;
            switch (KK) {
                CHAIN_WS_CASE(20)
                CHAIN_WS_CASE(32)
                CHAIN_WS_CASE(48)
                CHAIN_WS_CASE(80)
                CHAIN_WS_CASE(144)
                default:
                    set_err("bad chain mfma K");
                    return FSDR_ERR_INVALID;
            }
#undef CHAIN_WS_CASE
            HIP_TRY(hipGetLastError());
            return FSDR_OK;
        }
        const char* ap = getenv("FSDR_CHAIN_ALLPHASE");
        /* all-phase + aligned-group staging is the default (measured
         * ~2% faster: fewer barriers and half the staging-load
         * instructions; see profiles/pmc_sq_r02.txt). The ap kernel's
         * group staging needs a 16B-aligned input base — ring carry
         * offsets can be 8B-odd — so unaligned inputs (and
         * FSDR_CHAIN_ALLPHASE=0) take the 2-phase-halves kernel. */
        if ((!ap || atoi(ap) != 0) && ((uintptr_t)d_in & 15u) == 0) {
            unsigned SPm = (elemsP + 31u) & ~31u;
            size_t lds_ap =
                (8 * (size_t)SPm + 4 * ((size_t)KK + 16)) * sizeof(float);
#define CHAIN_AP_CASE(KV)                                                 \
    case KV:                                                              \
        hipLaunchKernelGGL(HIP_KERNEL_NAME(k_decim4_fft_mfma_ap_tpl<KV>), \
                           dim3(grid), dim3(MDFIR_BLOCK), lds_ap, st,     \
                           (const float2*)d_in, spec_dst,                 \
                           c->fused->d_mtaps, (long long)prod,            \
                           (long long)n_in,                               \
                           (const float2*)c->fft->d_twid, (float*)d_mag,  \
                           (int)L);                                       \
        break;
            switch (KK) {
                CHAIN_AP_CASE(20)
                CHAIN_AP_CASE(32)
                CHAIN_AP_CASE(48)
                CHAIN_AP_CASE(80)
                CHAIN_AP_CASE(144)
                default:
                    set_err("bad chain mfma K");
                    return FSDR_ERR_INVALID;
            }
#undef CHAIN_AP_CASE
            HIP_TRY(hipGetLastError());
            return FSDR_OK;
        }
        const char* occ8 = getenv("FSDR_CHAIN_OCC8");
        if (occ8 && atoi(occ8) != 0 && KK == 80) {
            /* experiment: force 8 waves/SIMD occupancy (VGPR capped to
             * 64 by the compiler; some spill risk) */
            hipLaunchKernelGGL(
                HIP_KERNEL_NAME((k_decim4_fft_mfma_tpl<80, 8>)),
                dim3(grid), dim3(MDFIR_BLOCK), lds, st,
                (const float2*)d_in, spec_dst, c->fused->d_mtaps,
                (long long)prod, (long long)n_in,
                (const float2*)c->fft->d_twid, (float*)d_mag, (int)L, 0);
            HIP_TRY(hipGetLastError());
            return FSDR_OK;
        }
        int stagger = 0;
        if (const char* sg = getenv("FSDR_CHAIN_STAGGER"))
            stagger = atoi(sg);
#define CHAIN_TPL_CASE(KV)                                                \
    case KV:                                                              \
        hipLaunchKernelGGL(HIP_KERNEL_NAME(k_decim4_fft_mfma_tpl<KV>),    \
                           dim3(grid), dim3(MDFIR_BLOCK), lds, st,        \
                           (const float2*)d_in, spec_dst,                 \
                           c->fused->d_mtaps, (long long)prod,            \
                           (long long)n_in,                               \
                           (const float2*)c->fft->d_twid, (float*)d_mag,  \
                           (int)L, stagger);                              \
        break;
        switch (KK) {
            CHAIN_TPL_CASE(20)
            CHAIN_TPL_CASE(32)
            CHAIN_TPL_CASE(48)
            CHAIN_TPL_CASE(80)
            CHAIN_TPL_CASE(144)
            default:
                set_err("bad chain mfma K");
                return FSDR_ERR_INVALID;
        }
#undef CHAIN_TPL_CASE
        HIP_TRY(hipGetLastError());
        return FSDR_OK;
    }
    rc = ensure_dev((void**)&c->d_y2, &c->y2_cap,
                    (prod + 8) * sizeof(float2));
    if (rc) return rc;
    float2* out2 = (float2*)d_out;
    if (!out2) {
        rc = ensure_dev((void**)&c->d_null, &c->null_cap,
                        (prod + 8) * sizeof(float2));
        if (rc) return rc;
        out2 = c->d_null;
    }
    if (c->fused) {
        rc = launch_decim_cf32(c->fused, d_in, c->d_y2, prod, n_in, st);
        if (rc) return rc;
    } else {
        /* two-stage path: y1 covers the y2 window
         * (y2 needs y1[D-1 + (prod-1)*D + nt2-1]) */
        size_t y1_need = D - 1 + (prod - 1) * D + nt2;
        rc = ensure_dev((void**)&c->d_y1, &c->y1_cap,
                        (y1_need + 8) * sizeof(float2));
        if (rc) return rc;
        rc = launch_fir_cf32(c->fir1, d_in, c->d_y1, y1_need, n_in, st);
        if (rc) return rc;
        rc = launch_decim_cf32(c->fir2, c->d_y1, c->d_y2, prod, y1_need,
                               st);
        if (rc) return rc;
    }
    rc = launch_fft(c->fft, c->d_y2, out2, frames, st, (float*)d_mag);
    return rc;
}

/* ================= ring (Slab-style) ================================== */

struct RingBuf {
    void* host = nullptr;  /* pinned; carry capacity + payload */
    void* dev = nullptr;
    size_t items = 0;      /* valid payload items (after the carry region) */
    hipEvent_t ev = nullptr;      /* copy (H2D [+ carry D2D]) complete */
    hipEvent_t free_ev = nullptr; /* consumer done reading dev */
};

/* Slab-exact streaming ring (slab.rs:110-152,369-399): the reader reports
 * how many items it CONSUMED and the ring carries the unconsumed
 * remainder in front of the next buffer — device-side (one small D2D on
 * the copy stream), so arbitrary chunk sizes stream exactly like the
 * reference block leaving leftovers in the slab buffer. `reserved`
 * is the carry CAPACITY (>= the consumer's worst-case leftover; for a
 * FIR chain that is taps-1 + decim*fft_len). */
struct fsdr_ring {
    size_t n_buffers, items_per_buffer, item_bytes, reserved;
    std::vector<RingBuf> bufs;
    std::deque<int> empty_q, full_q;
    int writer_cur = -1, reader_cur = -1;
    int prev_buf = -1;          /* released buffer still holding the tail */
    size_t prev_tail_off = 0;   /* byte offset of the tail in prev dev buf */
    size_t carry_items = 0;     /* items to prepend at the next acquire */
    size_t presented_items = 0; /* what the current reader_acquire returned */
    size_t presented_off = 0;   /* byte offset of presented span start */
    std::mutex mu;
    std::condition_variable cv;
    hipStream_t copy_stream = nullptr;
};

extern "C" fsdr_ring* fsdr_ring_create(size_t n_buffers,
                                       size_t items_per_buffer,
                                       size_t item_bytes,
                                       size_t reserved_items) {
    if (!have_gpu()) { set_err("no HIP device"); return nullptr; }
    if (n_buffers < 2 || items_per_buffer == 0 || item_bytes == 0) {
        set_err("invalid ring parameters");
        return nullptr;
    }
    fsdr_ring* r = new fsdr_ring();
    r->n_buffers = n_buffers;
    r->items_per_buffer = items_per_buffer;
    r->item_bytes = item_bytes;
    r->reserved = reserved_items;
    if (hipStreamCreate(&r->copy_stream) != hipSuccess) {
        set_err("stream create failed");
        delete r;
        return nullptr;
    }
    size_t bytes = (items_per_buffer + reserved_items) * item_bytes;
    for (size_t i = 0; i < n_buffers; i++) {
        RingBuf b;
        if (hipHostMalloc(&b.host, bytes) != hipSuccess ||
            hipMalloc(&b.dev, bytes) != hipSuccess ||
            hipEventCreate(&b.ev) != hipSuccess ||
            hipEventCreate(&b.free_ev) != hipSuccess) {
            set_err("ring buffer alloc failed");
            r->bufs.push_back(b);
            fsdr_ring_destroy(r);
            return nullptr;
        }
        r->bufs.push_back(b);
        r->empty_q.push_back((int)i);
    }
    return r;
}

extern "C" int fsdr_ring_writer_acquire(fsdr_ring* r, void** host_ptr,
                                        size_t* items) {
    if (!r) return FSDR_ERR_INVALID;
    std::unique_lock<std::mutex> lk(r->mu);
    r->cv.wait(lk, [&] { return !r->empty_q.empty(); });
    r->writer_cur = r->empty_q.front();
    r->empty_q.pop_front();
    RingBuf& b = r->bufs[r->writer_cur];
    *host_ptr = (char*)b.host + r->reserved * r->item_bytes;
    *items = r->items_per_buffer;
    return FSDR_OK;
}

extern "C" int fsdr_ring_writer_commit(fsdr_ring* r, size_t items) {
    if (!r || r->writer_cur < 0) return FSDR_ERR_INVALID;
    RingBuf& b = r->bufs[r->writer_cur];
    b.items = items;
    size_t off = r->reserved * r->item_bytes;
    /* don't overwrite dev while the consumer may still read it (a
     * never-recorded free_ev makes this wait a no-op) */
    HIP_TRY(hipStreamWaitEvent(r->copy_stream, b.free_ev, 0));
    HIP_TRY(hipMemcpyAsync((char*)b.dev + off, (char*)b.host + off,
                           items * r->item_bytes, hipMemcpyHostToDevice,
                           r->copy_stream));
    HIP_TRY(hipEventRecord(b.ev, r->copy_stream));
    {
        std::lock_guard<std::mutex> lk(r->mu);
        r->full_q.push_back(r->writer_cur);
        r->writer_cur = -1;
    }
    r->cv.notify_all();
    return FSDR_OK;
}

extern "C" int fsdr_ring_reader_acquire(fsdr_ring* r, void** dev_ptr,
                                        size_t* items) {
    if (!r) return FSDR_ERR_INVALID;
    std::unique_lock<std::mutex> lk(r->mu);
    r->cv.wait(lk, [&] { return !r->full_q.empty(); });
    r->reader_cur = r->full_q.front();
    r->full_q.pop_front();
    lk.unlock();
    RingBuf& b = r->bufs[r->reader_cur];
    size_t ib = r->item_bytes;
    size_t off = (r->reserved - r->carry_items) * ib;
    if (r->carry_items > 0 && r->prev_buf >= 0) {
        /* prepend the previous buffer's unconsumed tail (slab.rs:369-399),
         * after the consumer's reads of it have drained */
        RingBuf& pb = r->bufs[r->prev_buf];
        HIP_TRY(hipStreamWaitEvent(r->copy_stream, pb.free_ev, 0));
        HIP_TRY(hipMemcpyAsync((char*)b.dev + off,
                               (char*)pb.dev + r->prev_tail_off,
                               r->carry_items * ib,
                               hipMemcpyDeviceToDevice, r->copy_stream));
        HIP_TRY(hipEventRecord(b.ev, r->copy_stream));
    }
    if (r->prev_buf >= 0) {
        std::lock_guard<std::mutex> lk2(r->mu);
        r->empty_q.push_back(r->prev_buf);
        r->prev_buf = -1;
        r->cv.notify_all();
    }
    HIP_TRY(hipEventSynchronize(b.ev));
    r->presented_items = r->carry_items + b.items;
    r->presented_off = off;
    r->carry_items = 0;
    *dev_ptr = (char*)b.dev + off;
    *items = r->presented_items;
    return FSDR_OK;
}

/* Reader reports how much of the presented span it consumed; the
 * remainder is carried in front of the next buffer. `stream` is the
 * consumer's compute stream (the carry copy and buffer reuse are ordered
 * after work already enqueued there); NULL = default stream. */
extern "C" int fsdr_ring_reader_release_consumed(fsdr_ring* r,
                                                 size_t consumed,
                                                 void* stream) {
    if (!r || r->reader_cur < 0) return FSDR_ERR_INVALID;
    if (consumed > r->presented_items) {
        set_err("release_consumed: consumed > presented");
        return FSDR_ERR_INVALID;
    }
    size_t un = r->presented_items - consumed;
    if (un > r->reserved) {
        set_err("release_consumed: unconsumed tail exceeds ring "
                "reserved capacity");
        return FSDR_ERR_INVALID;
    }
    RingBuf& b = r->bufs[r->reader_cur];
    HIP_TRY(hipEventRecord(b.free_ev, (hipStream_t)stream));
    r->prev_buf = r->reader_cur;
    r->prev_tail_off = r->presented_off + consumed * r->item_bytes;
    r->carry_items = un;
    r->reader_cur = -1;
    return FSDR_OK;
}

extern "C" int fsdr_ring_reader_release(fsdr_ring* r) {
    return fsdr_ring_reader_release_consumed(r, r ? r->presented_items : 0,
                                             nullptr);
}

extern "C" void fsdr_ring_destroy(fsdr_ring* r) {
    if (!r) return;
    for (auto& b : r->bufs) {
        if (b.host) (void)hipHostFree(b.host);
        if (b.dev) (void)hipFree(b.dev);
        if (b.ev) (void)hipEventDestroy(b.ev);
        if (b.free_ev) (void)hipEventDestroy(b.free_ev);
    }
    if (r->copy_stream) (void)hipStreamDestroy(r->copy_stream);
    delete r;
}

/* ================= D2H return ring ==================================== *
 * The reverse direction of fsdr_ring: a device-side producer fills ring
 * buffers, async D2H on the ring's copy stream hands them to a pinned
 * host consumer — the vulkan d2h.rs Writer::submit / host Reader pair
 * (d2h.rs:66,247-268) with the circuit's empty-buffer recirculation
 * (d2h.rs:284-299). Writer: acquire(stream) (returns a device buffer;
 * the caller's stream is made to wait for that buffer's previous D2H)
 * -> fill on the stream -> commit(items, stream) (enqueues D2H after
 * the producer's work). Reader: acquire (waits for the copy; yields the
 * pinned host span) -> release (recycles). SPSC. */

struct fsdr_ring_d2h {
    size_t n_buffers, items_per_buffer, item_bytes;
    std::vector<RingBuf> bufs; /* free_ev = producer-done event */
    std::deque<int> empty_q, full_q;
    int writer_cur = -1, reader_cur = -1;
    std::mutex mu;
    std::condition_variable cv;
    hipStream_t copy_stream = nullptr;
};

extern "C" fsdr_ring_d2h* fsdr_ring_d2h_create(size_t n_buffers,
                                               size_t items_per_buffer,
                                               size_t item_bytes) {
    if (!have_gpu()) { set_err("no HIP device"); return nullptr; }
    if (n_buffers < 2 || items_per_buffer == 0 || item_bytes == 0) {
        set_err("invalid ring parameters");
        return nullptr;
    }
    fsdr_ring_d2h* r = new fsdr_ring_d2h();
    r->n_buffers = n_buffers;
    r->items_per_buffer = items_per_buffer;
    r->item_bytes = item_bytes;
    if (hipStreamCreate(&r->copy_stream) != hipSuccess) {
        set_err("stream create failed");
        delete r;
        return nullptr;
    }
    size_t bytes = items_per_buffer * item_bytes;
    for (size_t i = 0; i < n_buffers; i++) {
        RingBuf b;
        if (hipHostMalloc(&b.host, bytes) != hipSuccess ||
            hipMalloc(&b.dev, bytes) != hipSuccess ||
            hipEventCreate(&b.ev) != hipSuccess ||
            hipEventCreate(&b.free_ev) != hipSuccess) {
            set_err("ring buffer alloc failed");
            r->bufs.push_back(b);
            fsdr_ring_d2h_destroy(r);
            return nullptr;
        }
        r->bufs.push_back(b);
        r->empty_q.push_back((int)i);
    }
    return r;
}

extern "C" int fsdr_ring_d2h_writer_acquire(fsdr_ring_d2h* r,
                                            void** dev_ptr, size_t* items,
                                            void* stream) {
    if (!r) return FSDR_ERR_INVALID;
    std::unique_lock<std::mutex> lk(r->mu);
    r->cv.wait(lk, [&] { return !r->empty_q.empty(); });
    r->writer_cur = r->empty_q.front();
    r->empty_q.pop_front();
    lk.unlock();
    RingBuf& b = r->bufs[r->writer_cur];
    /* the producer must not overwrite dev before its previous D2H is
     * drained (no-op for a never-recorded event) */
    HIP_TRY(hipStreamWaitEvent((hipStream_t)stream, b.ev, 0));
    *dev_ptr = b.dev;
    *items = r->items_per_buffer;
    return FSDR_OK;
}

extern "C" int fsdr_ring_d2h_writer_commit(fsdr_ring_d2h* r, size_t items,
                                           void* stream) {
    if (!r || r->writer_cur < 0) return FSDR_ERR_INVALID;
    RingBuf& b = r->bufs[r->writer_cur];
    b.items = items;
    HIP_TRY(hipEventRecord(b.free_ev, (hipStream_t)stream));
    HIP_TRY(hipStreamWaitEvent(r->copy_stream, b.free_ev, 0));
    HIP_TRY(hipMemcpyAsync(b.host, b.dev, items * r->item_bytes,
                           hipMemcpyDeviceToHost, r->copy_stream));
    HIP_TRY(hipEventRecord(b.ev, r->copy_stream));
    {
        std::lock_guard<std::mutex> lk(r->mu);
        r->full_q.push_back(r->writer_cur);
        r->writer_cur = -1;
    }
    r->cv.notify_all();
    return FSDR_OK;
}

extern "C" int fsdr_ring_d2h_reader_acquire(fsdr_ring_d2h* r,
                                            void** host_ptr,
                                            size_t* items) {
    if (!r) return FSDR_ERR_INVALID;
    std::unique_lock<std::mutex> lk(r->mu);
    r->cv.wait(lk, [&] { return !r->full_q.empty(); });
    r->reader_cur = r->full_q.front();
    r->full_q.pop_front();
    lk.unlock();
    RingBuf& b = r->bufs[r->reader_cur];
    HIP_TRY(hipEventSynchronize(b.ev));
    *host_ptr = b.host;
    *items = b.items;
    return FSDR_OK;
}

extern "C" int fsdr_ring_d2h_reader_release(fsdr_ring_d2h* r) {
    if (!r || r->reader_cur < 0) return FSDR_ERR_INVALID;
    {
        std::lock_guard<std::mutex> lk(r->mu);
        r->empty_q.push_back(r->reader_cur);
        r->reader_cur = -1;
    }
    r->cv.notify_all();
    return FSDR_OK;
}

extern "C" void fsdr_ring_d2h_destroy(fsdr_ring_d2h* r) {
    if (!r) return;
    for (auto& b : r->bufs) {
        if (b.host) (void)hipHostFree(b.host);
        if (b.dev) (void)hipFree(b.dev);
        if (b.ev) (void)hipEventDestroy(b.ev);
        if (b.free_ev) (void)hipEventDestroy(b.free_ev);
    }
    if (r->copy_stream) (void)hipStreamDestroy(r->copy_stream);
    delete r;
}
