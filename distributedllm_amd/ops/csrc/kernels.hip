// CDNA4 (gfx950) kernels for the layer-sliced LLaMA inference engine.
//
// Replaces the vendored ggml CPU kernels of the reference
// (/root/reference/distllm/tensor_processor.cpp:474-809 builds the graph this
// file executes natively; SURVEY.md §2.5 maps each op K1-K15 to a kernel
// here). Design notes with the measurements that drove each choice:
// docs/kernels.md.
//
// Production decode path (q4_0/q4_1/f16 weights): the MFMA dequant-GEMM
// family around `wave_tile_kloop` — `mfma_f32_16x16x32_f16` tiles with the
// 4-bit dequantization done in exact packed-f16 VALU (0x6400|n == 1024+n),
// JT = 1/2/4 column tiles (up to 64 tokens) amortizing the HBM weight
// stream, optional RT row tiles amortizing the L2 B-panel, software-
// pipelined non-temporal dwordx4 weight loads, and RMSNorm fused into the
// consumers' B-fragment builds via a sumsq side channel. Split-K variants
// write plain f32 slabs combined by fused reduce/finish passes.
//
// MFMA-path weight layout in HBM (repacked at load; python side in
// engine/slice_engine.py::repack_mfma):
//   q4   : data   u32[R][nbp/4][4 ws][16 i][4 kb]  (one dwordx4 per lane
//          covers 4 consecutive K-blocks; nbp = nb padded to 4)
//          scales f16 (alpha, beta)[R][nbp/4][16 i][4 kb]
//   f16  : data   f16[R][cols/8][16 i][8]
// plus one prefetch batch of tail slack on every allocation.
//
// A legacy scalar GEMV family (wave_row_dot) serves W_F32 test models.
//
// Wavefront = 64 (CDNA); blocks are 256 threads = 4 waves.

#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <hip/hip_bf16.h>
#include <math.h>
#include <stdint.h>

#include "kernels.h"

#define WAVE 64
#define NWAVES 4
#define BLOCK 256

// ----------------------------------------------------- MFMA fragment types

typedef __attribute__((ext_vector_type(8))) _Float16 f16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4;

union ABFrag {
    uint32_t u[4];
    f16x8 v;
};

// packed-f16 helpers for the MFMA side-channels / dequant
__device__ __forceinline__ uint32_t pack_f16(float lo, float hi) {
    union { __half2 h; uint32_t u; } c;
    c.h = __floats2half2_rn(lo, hi);
    return c.u;
}
__device__ __forceinline__ __half2 u2h2(uint32_t w) {
    union { uint32_t u; __half2 h; } c;
    c.u = w;
    return c.h;
}
__device__ __forceinline__ uint32_t h22u(__half2 h) {
    union { __half2 h; uint32_t u; } c;
    c.h = h;
    return c.u;
}

// ---------------------------------------------------------------- reductions

__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off);
    return v;  // valid in lane 0
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off));
    return v;
}

// --------------------------------------------------------------- row x vec
// One WAVE computes dot(row r of W, x_t) for T tokens (T <= TMAX).
// x: [T][cols] f32 (row-major, ld = cols). Results valid in lane 0.
//
// Coalescing is the whole game here (decode GEMV = HBM-bound weight
// stream): every lane covers 4 CONSECUTIVE weights, so the wave's x loads
// are one contiguous 1 KiB float4 transaction, the q4 nibble loads one
// contiguous 128 B u32 transaction, and the f16 scale loads one 16 B span.
// (The naive block-per-lane layout put each lane's x on its own cache line
// — 64 transactions per wave instruction — and measured 25x off the
// bandwidth roofline; rocprof evidence in profiles/.)

template <int TMAX>
__device__ __forceinline__ void wave_row_dot_q4_0(
    const __half* __restrict__ scales, const uint8_t* __restrict__ qs,
    int row, int cols, const float* __restrict__ x, int T,
    float* __restrict__ acc /* [TMAX] */) {
    const int nb = cols >> 5;  // 32 weights per block
    const __half* srow = scales + (size_t)row * nb;
    const uint8_t* qrow = qs + (size_t)row * nb * 16;
    const int lane = threadIdx.x & (WAVE - 1);
    const int bofs = lane >> 3;        // block within the 8-block group
    const int p = (lane & 7) * 4;      // weight span [p, p+4) in the block
    const int qoff = p & 15;           // byte offset of the span's u32
    const int shift = (p >= 16) ? 4 : 0;  // high or low nibbles
#pragma unroll
    for (int t = 0; t < TMAX; ++t) acc[t] = 0.0f;
    const int ngroups = (nb + 7) >> 3;
    for (int g = 0; g < ngroups; ++g) {
        const int b = g * 8 + bofs;
        float d = 0.0f;
        uint32_t q = 0x88888888u;  // dequantizes to 0 with the -8 bias
        if (b < nb) {
            d = __half2float(srow[b]);
            q = *reinterpret_cast<const uint32_t*>(qrow + b * 16 + qoff);
        }
        const float w0 = (float)((int)((q >> (shift + 0)) & 0xF) - 8);
        const float w1 = (float)((int)((q >> (shift + 8)) & 0xF) - 8);
        const float w2 = (float)((int)((q >> (shift + 16)) & 0xF) - 8);
        const float w3 = (float)((int)((q >> (shift + 24)) & 0xF) - 8);
        const int xoff = b * 32 + p;
#pragma unroll
        for (int t = 0; t < TMAX; ++t) {
            if (t >= T) continue;
            float4 xv = make_float4(0.f, 0.f, 0.f, 0.f);
            if (b < nb)
                xv = *reinterpret_cast<const float4*>(
                    x + (size_t)t * cols + xoff);
            float s = fmaf(w0, xv.x, 0.0f);
            s = fmaf(w1, xv.y, s);
            s = fmaf(w2, xv.z, s);
            s = fmaf(w3, xv.w, s);
            acc[t] = fmaf(d, s, acc[t]);
        }
    }
#pragma unroll
    for (int t = 0; t < TMAX; ++t)
        if (t < T) acc[t] = wave_reduce_sum(acc[t]);
}

template <int TMAX>
__device__ __forceinline__ void wave_row_dot_q4_1(
    const __half* __restrict__ scales, const uint8_t* __restrict__ qs,
    int row, int cols, const float* __restrict__ x, int T,
    float* __restrict__ acc) {
    const int nb = cols >> 5;
    const __half* srow = scales + (size_t)row * nb * 2;  // (d, m) pairs
    const uint8_t* qrow = qs + (size_t)row * nb * 16;
    const int lane = threadIdx.x & (WAVE - 1);
    const int bofs = lane >> 3;
    const int p = (lane & 7) * 4;
    const int qoff = p & 15;
    const int shift = (p >= 16) ? 4 : 0;
#pragma unroll
    for (int t = 0; t < TMAX; ++t) acc[t] = 0.0f;
    const int ngroups = (nb + 7) >> 3;
    for (int g = 0; g < ngroups; ++g) {
        const int b = g * 8 + bofs;
        float d = 0.0f, m = 0.0f;
        uint32_t q = 0;
        if (b < nb) {
            const __half2 dm =
                *reinterpret_cast<const __half2*>(srow + b * 2);
            d = __half2float(__low2half(dm));
            m = __half2float(__high2half(dm));
            q = *reinterpret_cast<const uint32_t*>(qrow + b * 16 + qoff);
        }
        const float w0 = (float)((q >> (shift + 0)) & 0xF);
        const float w1 = (float)((q >> (shift + 8)) & 0xF);
        const float w2 = (float)((q >> (shift + 16)) & 0xF);
        const float w3 = (float)((q >> (shift + 24)) & 0xF);
        const int xoff = b * 32 + p;
#pragma unroll
        for (int t = 0; t < TMAX; ++t) {
            if (t >= T) continue;
            float4 xv = make_float4(0.f, 0.f, 0.f, 0.f);
            if (b < nb)
                xv = *reinterpret_cast<const float4*>(
                    x + (size_t)t * cols + xoff);
            float s = fmaf(w0, xv.x, 0.0f);
            s = fmaf(w1, xv.y, s);
            s = fmaf(w2, xv.z, s);
            s = fmaf(w3, xv.w, s);
            const float sx = (xv.x + xv.y) + (xv.z + xv.w);
            acc[t] += d * s + m * sx;
        }
    }
#pragma unroll
    for (int t = 0; t < TMAX; ++t)
        if (t < T) acc[t] = wave_reduce_sum(acc[t]);
}

template <int TMAX>
__device__ __forceinline__ void wave_row_dot_f16(
    const __half* __restrict__ data, int row, int cols,
    const float* __restrict__ x, int T, float* __restrict__ acc) {
    const __half* wrow = data + (size_t)row * cols;
    const int lane = threadIdx.x & (WAVE - 1);
#pragma unroll
    for (int t = 0; t < TMAX; ++t) acc[t] = 0.0f;
    // 4 consecutive halves per lane -> 256 weights per wave iteration;
    // w loads 8 B/lane (512 B contiguous), x loads 16 B/lane (1 KiB).
    const int niter = (cols + 255) >> 8;
    for (int g = 0; g < niter; ++g) {
        const int c = g * 256 + lane * 4;
        float w0 = 0.f, w1 = 0.f, w2 = 0.f, w3 = 0.f;
        if (c < cols) {
            const uint2 raw = *reinterpret_cast<const uint2*>(wrow + c);
            const __half2* h2 = reinterpret_cast<const __half2*>(&raw);
            const float2 a = __half22float2(h2[0]);
            const float2 b = __half22float2(h2[1]);
            w0 = a.x; w1 = a.y; w2 = b.x; w3 = b.y;
        }
#pragma unroll
        for (int t = 0; t < TMAX; ++t) {
            if (t >= T) continue;
            float4 xv = make_float4(0.f, 0.f, 0.f, 0.f);
            if (c < cols)
                xv = *reinterpret_cast<const float4*>(
                    x + (size_t)t * cols + c);
            float s = fmaf(w0, xv.x, 0.0f);
            s = fmaf(w1, xv.y, s);
            s = fmaf(w2, xv.z, s);
            s = fmaf(w3, xv.w, s);
            acc[t] += s;
        }
    }
#pragma unroll
    for (int t = 0; t < TMAX; ++t)
        if (t < T) acc[t] = wave_reduce_sum(acc[t]);
}

template <int TMAX>
__device__ __forceinline__ void wave_row_dot_f32(
    const float* __restrict__ data, int row, int cols,
    const float* __restrict__ x, int T, float* __restrict__ acc) {
    const float* wrow = data + (size_t)row * cols;
    const int lane = threadIdx.x & (WAVE - 1);
#pragma unroll
    for (int t = 0; t < TMAX; ++t) acc[t] = 0.0f;
    const int niter = (cols + 255) >> 8;
    for (int g = 0; g < niter; ++g) {
        const int c = g * 256 + lane * 4;
        float4 w4 = make_float4(0.f, 0.f, 0.f, 0.f);
        if (c < cols)
            w4 = *reinterpret_cast<const float4*>(wrow + c);
#pragma unroll
        for (int t = 0; t < TMAX; ++t) {
            if (t >= T) continue;
            float4 xv = make_float4(0.f, 0.f, 0.f, 0.f);
            if (c < cols)
                xv = *reinterpret_cast<const float4*>(
                    x + (size_t)t * cols + c);
            float s = fmaf(w4.x, xv.x, 0.0f);
            s = fmaf(w4.y, xv.y, s);
            s = fmaf(w4.z, xv.z, s);
            s = fmaf(w4.w, xv.w, s);
            acc[t] += s;
        }
    }
#pragma unroll
    for (int t = 0; t < TMAX; ++t)
        if (t < T) acc[t] = wave_reduce_sum(acc[t]);
}

template <int WT, int TMAX>
__device__ __forceinline__ void wave_row_dot(
    const WMat& w, int row, const float* __restrict__ x, int T,
    float* __restrict__ acc) {
    if (WT == W_Q4_0)
        wave_row_dot_q4_0<TMAX>((const __half*)w.scales, (const uint8_t*)w.data,
                                row, w.cols, x, T, acc);
    else if (WT == W_Q4_1)
        wave_row_dot_q4_1<TMAX>((const __half*)w.scales, (const uint8_t*)w.data,
                                row, w.cols, x, T, acc);
    else if (WT == W_F16)
        wave_row_dot_f16<TMAX>((const __half*)w.data, row, w.cols, x, T, acc);
    else
        wave_row_dot_f32<TMAX>((const float*)w.data, row, w.cols, x, T, acc);
}

// ------------------------------------------------------------------ RMSNorm
// y[t] = x[t] * rsqrt(mean(x[t]^2) + eps) * w     (K2 of SURVEY §2.5;
// reference: ggml_rms_norm + ggml_mul, tensor_processor.cpp:555-570)

__global__ void k_rmsnorm(const float* __restrict__ x,
                          const float* __restrict__ w,
                          float* __restrict__ y, int E, float eps) {
    const int t = blockIdx.x;
    const float* xt = x + (size_t)t * E;
    float* yt = y + (size_t)t * E;
    __shared__ float red[NWAVES];

    float ss = 0.0f;
    for (int i = threadIdx.x; i < (E >> 2); i += BLOCK) {
        const float4 v = reinterpret_cast<const float4*>(xt)[i];
        ss += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
    }
    ss = wave_reduce_sum(ss);
    const int wid = threadIdx.x / WAVE;
    if ((threadIdx.x & (WAVE - 1)) == 0) red[wid] = ss;
    __syncthreads();
    float total = 0.0f;
#pragma unroll
    for (int i = 0; i < NWAVES; ++i) total += red[i];
    const float scale = rsqrtf(total / (float)E + eps);

    for (int i = threadIdx.x; i < (E >> 2); i += BLOCK) {
        const float4 v = reinterpret_cast<const float4*>(xt)[i];
        const float4 g = reinterpret_cast<const float4*>(w)[i];
        float4 o;
        o.x = v.x * scale * g.x;
        o.y = v.y * scale * g.y;
        o.z = v.z * scale * g.z;
        o.w = v.w * scale * g.w;
        reinterpret_cast<float4*>(yt)[i] = o;
    }
}

// ------------------------------------------------- QKV + RoPE + KV append
// Fuses K3+K4+K5 of SURVEY §2.5 (reference: ggml_mul_mat wq/wk/wv +
// ggml_rope_inplace + ggml_cpy into cache, tensor_processor.cpp:579-622).
// Grid: (3E/4) blocks; block computes 4 consecutive rows of one of
// {wq, wk, wv}. RoPE (GGML mode 0: rotate adjacent pairs (2i, 2i+1) within
// each head, theta = pos * base^(-2i/D)) is applied in the epilogue to q and
// k rows; k/v rows are converted to f16 and appended to the cache at
// [seq[t]][pos[t]].

template <int WT, int TMAX>
__global__ void k_qkv_rope_append(
    WMat wq, WMat wk, WMat wv, const float* __restrict__ xn,
    float* __restrict__ q_buf, __half* __restrict__ k_cache,
    __half* __restrict__ v_cache, const int* __restrict__ pos,
    const int* __restrict__ seq, const float* __restrict__ inv_freq,
    int E, int D, int n_ctx, int T) {
    const int r_global = blockIdx.x * 4;  // in [0, 3E)
    const int mat = r_global / E;         // 0=q 1=k 2=v
    const int row0 = r_global % E;
    const int wid = threadIdx.x / WAVE;
    const int row = row0 + wid;

    float acc[TMAX];
    const WMat& w = (mat == 0) ? wq : (mat == 1) ? wk : wv;
    wave_row_dot<WT, TMAX>(w, row, xn, T, acc);

    __shared__ float ybuf[4][TMAX];
    if ((threadIdx.x & (WAVE - 1)) == 0) {
#pragma unroll
        for (int t = 0; t < TMAX; ++t)
            if (t < T) ybuf[wid][t] = acc[t];
    }
    __syncthreads();

    if (mat == 2) {
        // v rows: straight f16 append, thread handles (row r, token t)
        const int idx = threadIdx.x;
        if (idx < 4 * T) {
            const int rr = idx / T, t = idx % T;
            const int e = row0 + rr;
            __half* dst = v_cache +
                ((size_t)seq[t] * n_ctx + pos[t]) * E + e;
            *dst = __float2half(ybuf[rr][t]);
        }
        return;
    }
    // q/k rows: rope on pairs (row0+2p, row0+2p+1)
    const int idx = threadIdx.x;
    if (idx < 2 * T) {
        const int p = idx / T, t = idx % T;
        const int e = row0 + 2 * p;
        const int d = e % D;
        const float theta = (float)pos[t] * inv_freq[d >> 1];
        float c, s;
        __sincosf(theta, &s, &c);
        const float x0 = ybuf[2 * p][t], x1 = ybuf[2 * p + 1][t];
        const float o0 = x0 * c - x1 * s;
        const float o1 = x0 * s + x1 * c;
        if (mat == 0) {
            q_buf[(size_t)t * E + e] = o0;
            q_buf[(size_t)t * E + e + 1] = o1;
        } else {
            __half* dst = k_cache +
                ((size_t)seq[t] * n_ctx + pos[t]) * E + e;
            dst[0] = __float2half(o0);
            dst[1] = __float2half(o1);
        }
    }
}

// ------------------------------------------------------ streaming attention
// Fuses K6+K7+K8 of SURVEY §2.5 (reference: ggml_mul_mat KxQ + scale +
// diag_mask_inf + soft_max + VxP, tensor_processor.cpp:645-700) into one
// online-softmax kernel: O(1) memory in context length, never materializes
// the [P+N, N, H] score tensor. Grid: (T, H); token t attends over cache
// entries [0, pos[t]] of sequence seq[t] — causality for prefill rows comes
// from their per-row positions.

// 8 halves [d0, d0+8) of `row` as floats — UNCONDITIONAL 16 B load
// (the KV allocation carries 16 halves of tail pad, and a lane-varying
// tail branch would execute BOTH paths with exec masking — the guarded
// first version added 8 masked scalar loads to every wide load and
// measured slower than 2 B loads). Elements past D return whatever the
// next row holds: always finite (the cache is zero-initialized and only
// ever written with real values); callers discard d >= D.
__device__ __forceinline__ void load_voct(const __half* __restrict__ row,
                                          int d0, float f[8]) {
    const uint4 u = *reinterpret_cast<const uint4*>(row + d0);
    const __half2* h2 = reinterpret_cast<const __half2*>(&u);
    const float2 a = __half22float2(h2[0]);
    const float2 b = __half22float2(h2[1]);
    const float2 c = __half22float2(h2[2]);
    const float2 d = __half22float2(h2[3]);
    f[0] = a.x; f[1] = a.y; f[2] = b.x; f[3] = b.y;
    f[4] = c.x; f[5] = c.y; f[6] = d.x; f[7] = d.y;
}

// FUSEQKV (decode only — every token its own sequence, so no block needs
// another token's new KV row): the prologue sums the qkv split-K slabs
// for this (token, head) slice, applies RoPE, appends the K/V cache row
// itself and stages q in LDS — replacing the separate k_qkv_finish pass
// (each such small kernel costs ~5 us at this chain shape).
template <bool FUSEQKV>
__global__ void k_attention(
    const float* __restrict__ q_buf, const __half* __restrict__ k_cache,
    __half* __restrict__ v_cache, float* __restrict__ out,
    unsigned short* __restrict__ out_prep, const int* __restrict__ pos,
    const int* __restrict__ seq, int E, int Ekv, int D, int n_ctx,
    int jtw, const float* __restrict__ qkv_slab, int ks,
    const float* __restrict__ inv_freq) {
    const int t = blockIdx.x;
    const int h = blockIdx.y;
    const int J = pos[t] + 1;
    // GQA: q head h reads kv head h / (H / Hkv); KV rows are Ekv wide
    const int grp = (int)gridDim.y / (Ekv / D);
    const int hk = h / grp;
    const size_t base = (size_t)seq[t] * n_ctx * Ekv + hk * D;
    const float inv_sqrt_d = rsqrtf((float)D);

    extern __shared__ float smem[];
    float* lds_q = smem;          // [D]
    float* lds_p = smem + D;      // [BLOCK]
    float* lds_red = lds_p + BLOCK;  // [NWAVES]

    if (FUSEQKV) {
        // qkv_slab layout: f32[(E + 2*Ekv)/16 gtiles][ks][64 tok][16
        // rows]; q rows then k rows then v rows. With GQA the `grp`
        // blocks sharing a kv head all append the SAME row bytes —
        // redundant but benign.
        const int p = J - 1;
        const size_t koff = (size_t)(E >> 4) * ks * 64 * 16;
        const size_t voff = koff + (size_t)(Ekv >> 4) * ks * 64 * 16;
        __half* kv_k =
            const_cast<__half*>(k_cache) + base + (size_t)p * Ekv;
        __half* kv_v = v_cache + base + (size_t)p * Ekv;
        for (int d = threadIdx.x; d < D; d += BLOCK) {
            const int e = h * D + d;        // row in q space
            const int ek = hk * D + d;      // row in kv space
            float q = 0.f, k = 0.f, v = 0.f;
            for (int c = 0; c < ks; ++c) {
                q += qkv_slab[(((size_t)(e >> 4) * ks + c) * 64 + t) * 16 +
                              (e & 15)];
                const size_t offk =
                    (((size_t)(ek >> 4) * ks + c) * 64 + t) * 16 +
                    (ek & 15);
                k += qkv_slab[koff + offk];
                v += qkv_slab[voff + offk];
            }
            lds_q[d] = q;
            lds_p[d] = k;  // staged for the pair rotation below
            kv_v[d] = __float2half(v);
        }
        __syncthreads();
        // RoPE on (even, odd) pairs; h*D is even so e % D == d
        for (int d2 = threadIdx.x; d2 < D / 2; d2 += BLOCK) {
            const int d0 = 2 * d2;
            const float theta = (float)p * inv_freq[d2];
            float sn, cs;
            __sincosf(theta, &sn, &cs);
            const float q0 = lds_q[d0], q1 = lds_q[d0 + 1];
            const float k0 = lds_p[d0], k1 = lds_p[d0 + 1];
            lds_q[d0] = (q0 * cs - q1 * sn) * inv_sqrt_d;
            lds_q[d0 + 1] = (q0 * sn + q1 * cs) * inv_sqrt_d;
            kv_k[d0] = __float2half(k0 * cs - k1 * sn);
            kv_k[d0 + 1] = __float2half(k0 * sn + k1 * cs);
        }
        // zero-pad lds_q to the score loop's octet width
        for (int d = D + (int)threadIdx.x; d < ((D + 7) & ~7); d += BLOCK)
            lds_q[d] = 0.0f;
        // same-block visibility of the new K/V row: stores above are this
        // CU's own L1 write-through; __syncthreads orders them before the
        // scan loop's loads (cross-CU coherence is not needed — decode
        // tokens are distinct sequences, no other block reads this row)
        __syncthreads();
    } else {
        const int Dp = (D + 7) & ~7;  // scores read whole octets
        for (int d = threadIdx.x; d < Dp; d += BLOCK)
            lds_q[d] = (d < D)
                ? q_buf[(size_t)t * E + h * D + d] * inv_sqrt_d
                : 0.0f;
        __syncthreads();
    }

    // Flash-style per-wave streaming (zero __syncthreads in the loop):
    // wave w owns 4-row groups g ≡ w (mod 4); within the wave, lane
    // (sub = lane>>4, oct = lane&15) covers row (g*4 + sub)'s d-octet
    // oct*8 — one 16 B load each for K and V per row, the row's dot
    // reduced by a 4-step in-16 shuffle butterfly, q held in registers.
    // Each wave keeps private (m, l, o8) partials; one flash merge at
    // the end. (The earlier block-synchronized form paid 3 barriers +
    // 2 cross-wave reductions per 256-row chunk and scalar-width V
    // loads before that — the deep-decode ladder in
    // profiles/r2_deep_decode_ladder.md tracks the steps.)
    const int wid = threadIdx.x / WAVE;
    const int lane0 = threadIdx.x & (WAVE - 1);
    const int sub0 = lane0 >> 4;        // row within the 4-row group
    const int oct0 = lane0 & 15;        // d-octet within the row
    const int d0 = oct0 * 8;
    float q8[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) {
        const int d = d0 + e;
        if (FUSEQKV) {
            q8[e] = (d < D) ? lds_q[d] : 0.0f;  // prescaled in prologue
        } else {
            q8[e] = (d < D)
                ? q_buf[(size_t)t * E + h * D + d] * inv_sqrt_d
                : 0.0f;
        }
    }
    float m = -INFINITY, l = 0.0f;
    float o8[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) o8[e] = 0.0f;

    // two 4-row groups per iteration: both K loads issue before either
    // dot needs them, and the 8 rows share one max/rescale pass
    for (int g = wid; g * 4 < J; g += 2 * NWAVES) {
        const int j1 = g * 4 + sub0;
        const int g2i = g + NWAVES;
        const int j2 = g2i * 4 + sub0;
        const bool ok1 = j1 < J;
        const bool ok2 = (g2i * 4 < J) && (j2 < J);
        float s1 = 0.0f, s2 = 0.0f;
        if (d0 < D) {
            float f1[8], f2[8];
            load_voct(k_cache + base + (size_t)(ok1 ? j1 : 0) * Ekv, d0,
                      f1);
            load_voct(k_cache + base + (size_t)(ok2 ? j2 : 0) * Ekv, d0,
                      f2);
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                s1 = fmaf(f1[e], q8[e], s1);
                s2 = fmaf(f2[e], q8[e], s2);
            }
        }
        // full dots: butterfly over the 16 octet lanes
#pragma unroll
        for (int sh = 1; sh <= 8; sh <<= 1) {
            s1 += __shfl_xor(s1, sh);
            s2 += __shfl_xor(s2, sh);
        }
        if (!ok1) s1 = -INFINITY;
        if (!ok2) s2 = -INFINITY;
        // max over all 8 rows (sub groups at lane bits 4-5)
        float gm = fmaxf(s1, s2);
        gm = fmaxf(gm, __shfl_xor(gm, 16));
        gm = fmaxf(gm, __shfl_xor(gm, 32));
        if (gm > m) {  // rescale partials (wave-uniform branch)
            const float alpha = (m == -INFINITY) ? 0.0f : __expf(m - gm);
#pragma unroll
            for (int e = 0; e < 8; ++e) o8[e] *= alpha;
            l *= alpha;
            m = gm;
        }
        const float p1 = ok1 ? __expf(s1 - m) : 0.0f;
        const float p2 = ok2 ? __expf(s2 - m) : 0.0f;
        float psum = (p1 + p2) + __shfl_xor(p1 + p2, 16);
        psum += __shfl_xor(psum, 32);
        l += psum;
        if (d0 < D) {
            float f1[8], f2[8];
            // masked rows contribute p = 0 (V row 0 is a safe address)
            load_voct(v_cache + base + (size_t)(ok1 ? j1 : 0) * Ekv, d0,
                      f1);
            load_voct(v_cache + base + (size_t)(ok2 ? j2 : 0) * Ekv, d0,
                      f2);
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                o8[e] = fmaf(p1, f1[e], o8[e]);
                o8[e] = fmaf(p2, f2[e], o8[e]);
            }
        }
    }

    // flash merge of the 4 waves' (m, l) x 16 (wave, sub) o8 partials
    {
        float* lds_m = lds_red;               // [NWAVES]
        float* lds_l = lds_red + NWAVES;      // [NWAVES]
        float* lds_o = lds_l + NWAVES;        // [16][128]
        const int gidx = wid * 4 + sub0;
#pragma unroll
        for (int e = 0; e < 8; ++e) lds_o[gidx * 128 + d0 + e] = o8[e];
        if (lane0 == 0) {
            lds_m[wid] = m;
            lds_l[wid] = l;
        }
        __syncthreads();
        if (threadIdx.x < D) {
            const int d = threadIdx.x;
            float mstar = -INFINITY;
#pragma unroll
            for (int w = 0; w < NWAVES; ++w)
                mstar = fmaxf(mstar, lds_m[w]);
            float lstar = 0.0f, num = 0.0f;
#pragma unroll
            for (int w = 0; w < NWAVES; ++w) {
                const float mw = lds_m[w];
                const float sc =
                    (mw == -INFINITY) ? 0.0f : __expf(mw - mstar);
                lstar += lds_l[w] * sc;
                num += sc * (lds_o[(w * 4 + 0) * 128 + d] +
                             lds_o[(w * 4 + 1) * 128 + d] +
                             lds_o[(w * 4 + 2) * 128 + d] +
                             lds_o[(w * 4 + 3) * 128 + d]);
            }
            const float v = num / lstar;
            const int e = h * D + d;
            out[(size_t)t * E + e] = v;
            if (out_prep != nullptr) {
                // f16 B-layout side-channel for the wo MFMA consumer
                union { __half h; unsigned short u; } c;
                c.h = __float2half(v);
                out_prep[(((size_t)(e >> 3) * jtw + (t >> 4)) * 16 +
                          (t & 15)) * 8 + (e & 7)] = c.u;
            }
        }
    }
}

// ------------------------------------------------- prefill attention (QT=16)
// Q-tiled streaming attention for the large-M prefill path: one block
// handles QT=16 query tokens of a (head, 16-token tile), so each K/V row
// is read ONCE per 16 queries instead of once per query — 16x less HBM
// traffic than k_attention's decode shape (which is exactly bandwidth-
// bound at large T: 22 ms of a 48 ms span-1024 forward, profiles/).
// Mixed admission streams: a tile may span several sequences; the block
// processes each same-seq segment separately (causality per token via
// its own pos, exactly like k_attention).
__global__ void k_attn_prefill(
    const float* __restrict__ q_buf, const __half* __restrict__ k_cache,
    const __half* __restrict__ v_cache, float* __restrict__ out,
    unsigned short* __restrict__ out_prep, const int* __restrict__ pos,
    const int* __restrict__ seq, int E, int Ekv, int D, int n_ctx,
    int jtw, int T) {
    constexpr int QT = 16;
    const int t0 = blockIdx.x * QT;
    const int h = blockIdx.y;
    const int nq = min(QT, T - t0);
    const float inv_sqrt_d = rsqrtf((float)D);

    extern __shared__ float smem[];
    float* lds_q = smem;                  // [QT][D] prescaled q
    float* lds_s = lds_q + QT * D;        // [QT][BLOCK] scores -> p
    float* lds_mred = lds_s + QT * BLOCK; // [QT] chunk m_new
    float* lds_pred = lds_mred + QT;      // [QT] chunk sum_p
    __shared__ int lds_pos[QT];
    __shared__ int lds_seq[QT];

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    if (tid < nq) {
        lds_pos[tid] = pos[t0 + tid];
        lds_seq[tid] = seq[t0 + tid];
    }
    // lds_mred carries each q's running max ACROSS chunks (the reducing
    // wave's threads don't all run the PV stage, so the register copy
    // m[q] is only authoritative on threads < D)
    if (tid < QT) lds_mred[tid] = -INFINITY;
    for (int i = tid; i < nq * D; i += BLOCK) {
        const int q = i / D, d = i % D;
        lds_q[q * D + d] =
            q_buf[(size_t)(t0 + q) * E + h * D + d] * inv_sqrt_d;
    }
    __syncthreads();

    float o[QT];      // thread d accumulates output[d] for each q
    float m[QT], l[QT];
#pragma unroll
    for (int q = 0; q < QT; ++q) {
        o[q] = 0.0f;
        m[q] = -INFINITY;
        l[q] = 0.0f;
    }

    // Every per-q register array below is indexed ONLY by the fully
    // unrolled compile-time q (runtime indexing would spill o/m/l to
    // scratch — the first version did exactly that and ran 1.7x SLOWER
    // than the untiled kernel at 455 ms/span-1024-run).
    int a = 0;
    while (a < nq) {  // same-sequence segments of the tile
        const int sseq = lds_seq[a];
        int b = a + 1;
        while (b < nq && lds_seq[b] == sseq) ++b;
        int jmax = 0;
        for (int q = a; q < b; ++q) jmax = max(jmax, lds_pos[q]);
        const int J = jmax + 1;
        const int au = __builtin_amdgcn_readfirstlane(a);
        const int bu = __builtin_amdgcn_readfirstlane(b);
        const size_t base = (size_t)sseq * n_ctx * E + h * D;

        for (int j0 = 0; j0 < J; j0 += BLOCK) {
            const int jj = j0 + tid;
            // scores: thread jj reads K row once, dots all segment q's
            if (jj < J) {
                const __half* krow = k_cache + base + (size_t)jj * E;
                const int nh2 = D >> 2;
                const float2* k2 = reinterpret_cast<const float2*>(krow);
                float acc[QT];
#pragma unroll
                for (int q = 0; q < QT; ++q) acc[q] = 0.0f;
                for (int c = 0; c < nh2; ++c) {
                    const float2 raw = k2[c];
                    const __half2* hh =
                        reinterpret_cast<const __half2*>(&raw);
                    const float2 ka = __half22float2(hh[0]);
                    const float2 kb = __half22float2(hh[1]);
#pragma unroll
                    for (int q = 0; q < QT; ++q) {
                        if (q < au || q >= bu) continue;
                        const float* qr = lds_q + q * D + c * 4;
                        float s = fmaf(ka.x, qr[0], 0.0f);
                        s = fmaf(ka.y, qr[1], s);
                        s = fmaf(kb.x, qr[2], s);
                        s = fmaf(kb.y, qr[3], s);
                        acc[q] += s;
                    }
                }
#pragma unroll
                for (int q = 0; q < QT; ++q) {
                    if (q < au || q >= bu) continue;
                    lds_s[q * BLOCK + tid] =
                        (jj <= lds_pos[q]) ? acc[q] : -INFINITY;
                }
            } else {
#pragma unroll
                for (int q = 0; q < QT; ++q) {
                    if (q < au || q >= bu) continue;
                    lds_s[q * BLOCK + tid] = -INFINITY;
                }
            }
            __syncthreads();
            // per-q online-softmax bookkeeping: wave w owns q = w (mod 4)
            for (int q = au + wid; q < bu; q += NWAVES) {
                float* row = lds_s + q * BLOCK;
                float wm = -INFINITY;
#pragma unroll
                for (int i = 0; i < BLOCK / WAVE; ++i)
                    wm = fmaxf(wm, row[i * WAVE + lane]);
                wm = wave_reduce_max(wm);
                wm = __shfl(wm, 0);
                const float m_old = lds_mred[q];
                const float m_new = fmaxf(m_old, wm);
                float ws = 0.0f;
#pragma unroll
                for (int i = 0; i < BLOCK / WAVE; ++i) {
                    const float p = __expf(row[i * WAVE + lane] - m_new);
                    row[i * WAVE + lane] = p;
                    ws += p;
                }
                ws = wave_reduce_sum(ws);
                if (lane == 0) {
                    lds_mred[q] = m_new;
                    lds_pred[q] = ws;
                }
            }
            __syncthreads();
            // V accumulation: thread d sums p[q][j] * V[j][d]
            if (tid < D) {
                const int jlim = min(BLOCK, J - j0);
                const __half* vcol =
                    v_cache + base + (size_t)j0 * Ekv + tid;
#pragma unroll
                for (int q = 0; q < QT; ++q) {
                    if (q < au || q >= bu) continue;
                    const float m_new = lds_mred[q];
                    const float alpha =
                        (m[q] == -INFINITY) ? 0.0f : __expf(m[q] - m_new);
                    o[q] *= alpha;
                    l[q] = l[q] * alpha + lds_pred[q];
                    m[q] = m_new;
                }
                for (int jc = 0; jc < jlim; ++jc) {
                    const float v = __half2float(vcol[(size_t)jc * Ekv]);
#pragma unroll
                    for (int q = 0; q < QT; ++q) {
                        if (q < au || q >= bu) continue;
                        o[q] = fmaf(lds_s[q * BLOCK + jc], v, o[q]);
                    }
                }
            }
            __syncthreads();  // lds_s reuse next chunk
        }
        a = b;
    }
    if (tid < D) {
#pragma unroll
        for (int q = 0; q < QT; ++q) {
            if (q >= nq) break;
            const float v = o[q] / l[q];
            const int e = h * D + tid;
            const int t = t0 + q;
            out[(size_t)t * E + e] = v;
            if (out_prep != nullptr) {
                union { __half h; unsigned short u; } c;
                c.h = __float2half(v);
                out_prep[(((size_t)(e >> 3) * jtw + (t >> 4)) * 16 +
                          (t & 15)) * 8 + (e & 7)] = c.u;
            }
        }
    }
}

// --------------------------------------- prefill flash attention (MFMA)
// Matrix-core flash attention for the large-M prefill path (D <= 128):
// one block = one (16-query tile, head); each of the 4 waves streams a
// strided subset of the 16-row KV tiles FULLY IN REGISTERS — scores via
// mfma_f32_16x16x32_f16 (A = K tile, B = the staged Q tile, C rows = j,
// cols = q), online softmax on the 4 score rows each lane holds, then
// PV via mfma_f32_16x16x16_f16 where the score C-layout (lane: q =
// l&15, j = (l>>4)*4+jj) IS the B-operand layout (cols q, k-dim j) —
// no transpose, no LDS, no syncs in the inner loop. Partial (m, l, O)
// merge across waves once at the end. The VALU fallback k_attn_prefill
// (below) covers D > 128 and measured ~25x off issue-roofline — this
// kernel exists because 16x16 tiles of dot products belong on MFMA.
typedef __attribute__((ext_vector_type(4))) _Float16 f16x4;

union BFrag4 {
    uint32_t u[2];
    f16x4 v;
};

// 8 halves [d0, d0+8) of `row`, zero-padded past D
__device__ __forceinline__ f16x8 load_kfrag8(const __half* __restrict__ row,
                                             int d0, int D) {
    ABFrag a;
    if (d0 + 8 <= D) {
        const uint4 u = *reinterpret_cast<const uint4*>(row + d0);
        a.u[0] = u.x; a.u[1] = u.y; a.u[2] = u.z; a.u[3] = u.w;
    } else {
        __half tmp[8];
#pragma unroll
        for (int e = 0; e < 8; ++e)
            tmp[e] = (d0 + e < D) ? row[d0 + e] : __half(0.0f);
        const uint4 u = *reinterpret_cast<const uint4*>(tmp);
        a.u[0] = u.x; a.u[1] = u.y; a.u[2] = u.z; a.u[3] = u.w;
    }
    return a.v;
}

__global__ __launch_bounds__(BLOCK) void k_attn_prefill_mfma(
    const float* __restrict__ q_buf, const __half* __restrict__ k_cache,
    const __half* __restrict__ v_cache, float* __restrict__ out,
    unsigned short* __restrict__ out_prep, const int* __restrict__ pos,
    const int* __restrict__ seq, int E, int Ekv, int D, int n_ctx,
    int jtw, int T) {
    constexpr int QT = 16;
    constexpr int DCMAX = 8;            // D <= 128 = 8 chunks of 16
    const int t0 = blockIdx.x * QT;
    const int h = blockIdx.y;
    const int nq = min(QT, T - t0);
    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int lq = lane & 15;           // this lane's query column
    const int lg = lane >> 4;           // lane group (j/k sub-span)
    const int ndc = (D + 15) >> 4;      // 16-wide d chunks (PV / output)
    const float inv_sqrt_d = rsqrtf((float)D);

    __shared__ int lds_pos[QT];
    __shared__ int lds_seq[QT];
    // per-wave partials for the final merge
    __shared__ float lds_m[NWAVES][QT];
    __shared__ float lds_l[NWAVES][QT];
    __shared__ float lds_o[NWAVES][QT][DCMAX * 16];

    if (tid < QT) {
        lds_pos[tid] = (t0 + tid < T) ? pos[t0 + tid] : 0;
        lds_seq[tid] = (t0 + tid < T) ? seq[t0 + tid] : -1;
    }
    __syncthreads();

    // stage Q as score-MFMA B fragments: lane (col q=lq, k-span lg*8
    // within each 32-d chunk), prescaled by 1/sqrt(D), zero past D/nq
    ABFrag qf[4];
#pragma unroll
    for (int c = 0; c < 4; ++c) {
        __half tmp[8];
        const int d0 = c * 32 + lg * 8;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
            float v = 0.0f;
            if (t0 + lq < T && d0 + e < D)
                v = q_buf[(size_t)(t0 + lq) * E + h * D + d0 + e] *
                    inv_sqrt_d;
            tmp[e] = __float2half(v);
        }
        const uint4 u = *reinterpret_cast<const uint4*>(tmp);
        qf[c].u[0] = u.x; qf[c].u[1] = u.y;
        qf[c].u[2] = u.z; qf[c].u[3] = u.w;
    }

    float m = -INFINITY, l = 0.0f;      // running state for column lq
    f32x4 oacc[DCMAX];
#pragma unroll
    for (int dc = 0; dc < DCMAX; ++dc)
        oacc[dc] = f32x4{0.f, 0.f, 0.f, 0.f};

    const int pos_q = lds_pos[min(lq, nq - 1)];
    int a = 0;
    while (a < nq) {  // same-sequence segments of the query tile
        const int sseq = lds_seq[a];
        int b = a + 1;
        while (b < nq && lds_seq[b] == sseq) ++b;
        int jmax = 0;
        for (int q = a; q < b; ++q) jmax = max(jmax, lds_pos[q]);
        const int J = jmax + 1;
        const bool active = (lq >= a && lq < b);
        const int hk = h / ((int)gridDim.y / (Ekv / D));  // GQA kv head
        const size_t base = (size_t)sseq * n_ctx * Ekv + hk * D;

        // wave w streams tiles w, w+4, w+8, ... of this segment
        for (int tj = wid * 16; tj < J; tj += NWAVES * 16) {
            // scores: A = K rows (lane: row j=lq of the tile, k-span lg)
            const bool jrow_ok = (tj + lq) < J;  // tile tail: no OOB read
            const __half* krow =
                k_cache + base + (size_t)(jrow_ok ? tj + lq : 0) * Ekv;
            f32x4 sc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int c = 0; c < 4; ++c) {
                if (c * 32 >= D) break;
                f16x8 kf = load_kfrag8(krow, c * 32 + lg * 8, D);
                if (!jrow_ok) kf = f16x8{0, 0, 0, 0, 0, 0, 0, 0};
                sc = __builtin_amdgcn_mfma_f32_16x16x32_f16(kf, qf[c].v,
                                                            sc, 0, 0, 0);
            }
            // lane now holds s[j = tj + lg*4 + jj][q = lq]; mask + max
            float s[4], tm = -INFINITY;
#pragma unroll
            for (int jj = 0; jj < 4; ++jj) {
                const int j = tj + lg * 4 + jj;
                s[jj] = (active && j < J && j <= pos_q) ? sc[jj]
                                                        : -INFINITY;
                tm = fmaxf(tm, s[jj]);
            }
            tm = fmaxf(tm, __shfl_xor(tm, 16));
            tm = fmaxf(tm, __shfl_xor(tm, 32));
            // wave-uniform skip only (divergent MFMA is illegal); lanes
            // whose q saw nothing yet keep (m=-inf, l=0, o=0) via
            // dead-lane handling below — never compute -inf - -inf
            if (!__any(tm != -INFINITY)) continue;
            const float m_new = fmaxf(m, tm);
            const bool dead = (m_new == -INFINITY);
            const float alpha =
                (dead || m == -INFINITY) ? (dead ? 1.0f : 0.0f)
                                         : __expf(m - m_new);
            BFrag4 p;
            float ts = 0.0f;
#pragma unroll
            for (int jj = 0; jj < 4; ++jj) {
                const float pv = dead ? 0.0f : __expf(s[jj] - m_new);
                ts += pv;
                reinterpret_cast<__half*>(p.u)[jj] = __float2half(pv);
            }
            ts += __shfl_xor(ts, 16);
            ts += __shfl_xor(ts, 32);
            l = l * alpha + ts;
            m = m_new;
#pragma unroll
            for (int dc = 0; dc < DCMAX; ++dc) {
                if (dc >= ndc) break;
                oacc[dc][0] *= alpha; oacc[dc][1] *= alpha;
                oacc[dc][2] *= alpha; oacc[dc][3] *= alpha;
            }
            // PV: A = V^T (lane: row d = dc*16+lq, k-span j = lg*4+e);
            // B = p (exactly the score C layout). Masked j rows carry
            // p = 0; KV memory is zero-initialized so V rows past J
            // are finite.
#pragma unroll
            for (int dc = 0; dc < DCMAX; ++dc) {
                if (dc >= ndc) break;
                const int d = dc * 16 + lq;
                BFrag4 vt;
#pragma unroll
                for (int e = 0; e < 4; ++e) {
                    const int j = tj + lg * 4 + e;
                    __half v(0.0f);
                    if (d < D && j < J)
                        v = v_cache[base + (size_t)j * Ekv + d];
                    reinterpret_cast<__half*>(vt.u)[e] = v;
                }
                oacc[dc] = __builtin_amdgcn_mfma_f32_16x16x16f16(
                    vt.v, p.v, oacc[dc], 0, 0, 0);
            }
        }
        a = b;
    }

    // merge the 4 waves' partials (each lane group holds redundant
    // copies of (m, l) for its q; group 0 writes)
    if (lg == 0) {
        lds_m[wid][lq] = m;
        lds_l[wid][lq] = l;
    }
#pragma unroll
    for (int dc = 0; dc < DCMAX; ++dc) {
        if (dc >= ndc) break;
#pragma unroll
        for (int jj = 0; jj < 4; ++jj)
            lds_o[wid][lq][dc * 16 + lg * 4 + jj] = oacc[dc][jj];
    }
    __syncthreads();
    // thread (q, d) recombines: 256 threads cover 16 q x 16 d per pass
    const int qq = tid >> 4;
    for (int d = tid & 15; d < D; d += 16) {
        if (qq >= nq) break;
        float mstar = -INFINITY;
#pragma unroll
        for (int w = 0; w < NWAVES; ++w)
            mstar = fmaxf(mstar, lds_m[w][qq]);
        if (mstar == -INFINITY) mstar = 0.0f;  // no unmasked rows
        float lstar = 0.0f, ostar = 0.0f;
#pragma unroll
        for (int w = 0; w < NWAVES; ++w) {
            const float mw = lds_m[w][qq];
            const float sc = (mw == -INFINITY) ? 0.0f
                                               : __expf(mw - mstar);
            lstar += lds_l[w][qq] * sc;
            ostar += lds_o[w][qq][d] * sc;
        }
        const float v = ostar / fmaxf(lstar, 1e-20f);
        const int e = h * D + d;
        const int t = t0 + qq;
        out[(size_t)t * E + e] = v;
        if (out_prep != nullptr) {
            union { __half h; unsigned short u; } cvt;
            cvt.h = __float2half(v);
            out_prep[(((size_t)(e >> 3) * jtw + (t >> 4)) * 16 +
                      (t & 15)) * 8 + (e & 7)] = cvt.u;
        }
    }
}

// --------------------------------------------------------------- GEMV(+res)
// K9/K10/K13/K14 of SURVEY §2.5: output projection / FFN down / lm_head,
// with the residual add fused into the epilogue.

template <int WT, int TMAX, bool RES>
__global__ void k_gemv(WMat w, const float* __restrict__ x,
                       const float* __restrict__ res,
                       float* __restrict__ y, int T) {
    const int row = blockIdx.x * 4 + threadIdx.x / WAVE;
    float acc[TMAX];
    wave_row_dot<WT, TMAX>(w, row, x, T, acc);
    if ((threadIdx.x & (WAVE - 1)) == 0) {
#pragma unroll
        for (int t = 0; t < TMAX; ++t) {
            if (t >= T) continue;
            float v = acc[t];
            if (RES) v += res[(size_t)t * w.rows + row];
            y[(size_t)t * w.rows + row] = v;
        }
    }
}

// ------------------------------------------------------------ w1/w3 SwiGLU
// K11+K12 of SURVEY §2.5 (reference: ggml_mul_mat w1/w3 + ggml_silu +
// ggml_mul, tensor_processor.cpp:732-751): g = silu(w1·x) * (w3·x).

template <int WT, int TMAX>
__global__ void k_ffn_gate(WMat w1, WMat w3, const float* __restrict__ xn,
                           float* __restrict__ g, int T) {
    const int row = blockIdx.x * 4 + threadIdx.x / WAVE;
    float a1[TMAX], a3[TMAX];
    wave_row_dot<WT, TMAX>(w1, row, xn, T, a1);
    wave_row_dot<WT, TMAX>(w3, row, xn, T, a3);
    if ((threadIdx.x & (WAVE - 1)) == 0) {
#pragma unroll
        for (int t = 0; t < TMAX; ++t) {
            if (t >= T) continue;
            const float v1 = a1[t];
            const float silu = v1 / (1.0f + __expf(-v1));
            g[(size_t)t * w1.rows + row] = silu * a3[t];
        }
    }
}

// -------------------------------------------------------- embedding gather
// K1 of SURVEY §2.5 (ggml_get_rows, tensor_processor.cpp:1767): dequantize
// row tok[t] of the embedding table into f32 activations.

template <int WT>
__global__ void k_embed(WMat tab, const int* __restrict__ tokens,
                        float* __restrict__ out, int E) {
    const int t = blockIdx.x;
    const int row = tokens[t];
    float* dst = out + (size_t)t * E;
    if (WT == W_Q4_0) {
        const int nb = E >> 5;
        const __half* srow = (const __half*)tab.scales + (size_t)row * nb;
        const uint8_t* qrow = (const uint8_t*)tab.data + (size_t)row * nb * 16;
        for (int b = threadIdx.x; b < nb; b += BLOCK) {
            const float d = __half2float(srow[b]);
            const uint4 packed = *reinterpret_cast<const uint4*>(qrow + b * 16);
            const uint32_t w[4] = {packed.x, packed.y, packed.z, packed.w};
#pragma unroll
            for (int i = 0; i < 4; ++i) {
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    const int lo = (int)((w[i] >> (8 * j)) & 0xF) - 8;
                    const int hi = (int)((w[i] >> (8 * j + 4)) & 0xF) - 8;
                    dst[(b << 5) + i * 4 + j] = d * (float)lo;
                    dst[(b << 5) + 16 + i * 4 + j] = d * (float)hi;
                }
            }
        }
    } else if (WT == W_Q4_1) {
        const int nb = E >> 5;
        const __half* srow = (const __half*)tab.scales + (size_t)row * nb * 2;
        const uint8_t* qrow = (const uint8_t*)tab.data + (size_t)row * nb * 16;
        for (int b = threadIdx.x; b < nb; b += BLOCK) {
            const float d = __half2float(srow[b * 2]);
            const float mm = __half2float(srow[b * 2 + 1]);
            const uint4 packed = *reinterpret_cast<const uint4*>(qrow + b * 16);
            const uint32_t w[4] = {packed.x, packed.y, packed.z, packed.w};
#pragma unroll
            for (int i = 0; i < 4; ++i) {
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    const int lo = (int)((w[i] >> (8 * j)) & 0xF);
                    const int hi = (int)((w[i] >> (8 * j + 4)) & 0xF);
                    dst[(b << 5) + i * 4 + j] = d * (float)lo + mm;
                    dst[(b << 5) + 16 + i * 4 + j] = d * (float)hi + mm;
                }
            }
        }
    } else if (WT == W_F16) {
        const __half* wrow = (const __half*)tab.data + (size_t)row * E;
        for (int i = threadIdx.x; i < E; i += BLOCK)
            dst[i] = __half2float(wrow[i]);
    } else {
        const float* wrow = (const float*)tab.data + (size_t)row * E;
        for (int i = threadIdx.x; i < E; i += BLOCK) dst[i] = wrow[i];
    }
}

// ------------------------------------------------------------------- argmax
// K15 of SURVEY §2.5 (sample_next_token greedy argmax,
// tensor_processor.cpp:1894-1908), on-device so greedy decode never copies
// the [V] logits to host.

// Two-stage: grid (T, NPART) partial blocks each reduce a V/NPART chunk
// and atomicMax a packed u64 key (ordered-f32 ‖ bit-inverted index — the
// index inversion makes ties resolve to the SMALLEST index, matching the
// reference's first-max scan); a tiny decode kernel unpacks. One block per
// row (the previous form) used only T CUs of 256 and measured 40 µs at
// T=4, V=32000.
__device__ __forceinline__ uint32_t ordered_f32(float v) {
    union { float f; uint32_t u; } c;
    c.f = v;
    return (c.u & 0x80000000u) ? ~c.u : (c.u | 0x80000000u);
}

__global__ void k_argmax_part(const float* __restrict__ logits,
                              unsigned long long* __restrict__ keys, int V) {
    const int t = blockIdx.x;
    const float* lt = logits + (size_t)t * V;
    const int chunk = (V + gridDim.y - 1) / gridDim.y;
    const int i0 = blockIdx.y * chunk;
    const int i1 = min(V, i0 + chunk);
    float best = -INFINITY;
    int bi = i0;
    for (int i = i0 + threadIdx.x; i < i1; i += BLOCK) {
        const float v = lt[i];
        if (v > best || (v == best && i < bi)) {
            best = v;
            bi = i;
        }
    }
    unsigned long long key =
        ((unsigned long long)ordered_f32(best) << 32) | (uint32_t)(~bi);
    __shared__ unsigned long long sk[BLOCK];
    sk[threadIdx.x] = key;
    __syncthreads();
    for (int stride = BLOCK / 2; stride > 0; stride >>= 1) {
        if (threadIdx.x < stride)
            sk[threadIdx.x] = max(sk[threadIdx.x], sk[threadIdx.x + stride]);
        __syncthreads();
    }
    if (threadIdx.x == 0) atomicMax(keys + t, sk[0]);
}

__global__ void k_argmax_finish(const unsigned long long* __restrict__ keys,
                                int* __restrict__ out, int T) {
    const int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t < T) out[t] = (int)~(uint32_t)(keys[t] & 0xFFFFFFFFu);
}

// ======================================================================
// MFMA dequant-GEMM family — the production decode path.
//
// Matrix cores do the FLOPs; the VALU only unpacks. One
// mfma_f32_16x16x32_bf16 consumes exactly one q4_0 block column (K=32) of
// a 16-row tile, with the 4-bit dequantization folded in two ways:
//   * nibbles become exact bf16 integers by OR-ing into the 0x4300
//     exponent pattern (bf16(0x4300 | n) == 128 + n), so A carries (n+128)
//     and the per-block scale/bias is applied to the 16x16 MFMA result:
//     w = d*(n-8)  =>  d*(D - 136*sum(B_col));
//   * the repacked nibble word interleaves weights so the three OR+AND+SHR
//     per word produce the bf16x8 A-fragment directly (see repack in
//     engine/slice_engine.py).
// Activations ride a bf16 side-channel ("xprep", layout [cols/8][16][8] =
// exactly the B-fragment gather) written by the producing kernel's
// epilogue; the f32 residual stream stays f32. RMSNorm is fused into the
// consumer's B-fragment build (sumsq side-channel written by the previous
// residual epilogue), so a decode layer is 5 kernels total.
//
// Block = 4 waves; each wave owns a quarter of the K blocks of ONE 16-row
// tile; partial accumulators combine through LDS; wave 0 runs the fused
// epilogue (residual add + sumsq atomics + xprep/rope/cache/silu writes).

// (f16x8 / f32x4 / u32x4 / ABFrag / pack_f16 helpers are defined near
// the top of the file — shared with the prefill attention kernels)

// write-through 16 B store (sc0 sc1): the split-K slab publish leaves no
// dirty lines in the writer XCD's L2, so the consumer kernel's reads are
// not cross-XCD dirty-line snoops (cdna_hip_programming.md §5 split-K
// recipe: write-through slab stores beat plain stores + boundary flush)
__device__ __forceinline__ void store_f4_wt(float* addr, float4 v) {
    f32x4 w = {v.x, v.y, v.z, v.w};
    asm volatile("global_store_dwordx4 %0, %1, off sc0 sc1\n\ts_nop 1"
                 :: "v"(addr), "v"(w) : "memory");
}

// Per-wave state for the K loop: lane (i = l&15 row, ks = l>>4 k-span).
// For q4 weights every wave range is aligned to the 4-K-block load group
// (the repacked layout packs 4 consecutive K-blocks per dwordx4); ranges
// live in the PADDED block count (repack zero-pads nb to a multiple of 4
// with alpha=beta=0 blocks that contribute exact zeros).
struct KLoop {
    int lane, i, ks;
    int kb0, kb1;  // this wave's K-block range
    // split a [b0, b1) block range (grid-level split-K) across the waves;
    // align=4 for q4 group loads, 1 for f16
    __device__ void init_range(int b0, int b1, int align, int nwaves = NWAVES) {
        lane = threadIdx.x & (WAVE - 1);
        i = lane & 15;
        ks = lane >> 4;
        const int wid = threadIdx.x / WAVE;
        const int am = align - 1;
        int per = (b1 - b0 + nwaves - 1) / nwaves;
        per = (per + am) & ~am;
        // readfirstlane: wid is wave-uniform but derived from threadIdx,
        // so without this the compiler treats every K loop bounded by
        // kb0/kb1 as lane-divergent and wraps each iteration in exec-mask
        // bookkeeping (saveexec/or/andn2 + implicit-def accvgpr moves)
        kb0 = __builtin_amdgcn_readfirstlane(min(b1, b0 + wid * per));
        kb1 = __builtin_amdgcn_readfirstlane(min(b1, kb0 + per));
    }
};

// Dequantize one repacked nibble word into an f16 A fragment with pure
// packed-f16 VALU (no unpack-to-f32 shift/and per half — the bf16 variant
// of this dequant measured VALU-issue-bound):
//   0x6400 | n is EXACTLY the f16 value 1024+n, so
//   q4_0: w = alpha * ((1024+n) - 1032)        (n-8 exact in f16)
//   q4_1: w = alpha * ((1024+n) - 1024) + beta (beta = m)
// per word: and/shr/or unpack + one v_pk_add_f16 + one v_pk_fma_f16; the
// result is the correctly-rounded f16 of the true q4 value (single
// rounding). ab = (beta_f16 << 16) | alpha_f16 per (row, block).
template <int WT>
__device__ __forceinline__ void a_frag_q4(uint32_t q, uint32_t ab,
                                          ABFrag& a) {
    const __half2 abh = u2h2(ab);
    const __half2 alpha2 = __half2half2(__low2half(abh));
    const __half2 beta2 = __half2half2(__high2half(abh));
    // f16 bit patterns: -1032 = 0xE408, -1024 = 0xE400
    const uint32_t csub = (WT == W_Q4_0) ? 0xE408E408u : 0xE400E400u;
    const __half2 c2 = u2h2(csub);
    uint32_t w[4];
    w[0] = 0x64006400u | (q & 0x000F000Fu);
    w[1] = 0x64006400u | ((q >> 4) & 0x000F000Fu);
    w[2] = 0x64006400u | ((q >> 8) & 0x000F000Fu);
    w[3] = 0x64006400u | ((q >> 12) & 0x000F000Fu);
#pragma unroll
    for (int i = 0; i < 4; ++i) {
        const __half2 v = __hadd2(u2h2(w[i]), c2);
        a.u[i] = h22u(__hfma2(v, alpha2, beta2));
    }
}

// Byte-stream A fragment (W_Q8B): 8 re-biased u8 weights arrive as two
// u32 with byte order [w0,w2,w1,w3] / [w4,w6,w5,w7], so the same mask
// trick as q4 yields f16 pairs: 0x6400|b == f16(1024+b) exactly for
// b in [0,255], then w = alpha*((1024+b) - 1152) + beta (csub -1152 =
// 0xE480; repack re-biases q8_0 by +128 and q5_x by +112 so one csub
// serves all three formats).
__device__ __forceinline__ void a_frag_q8(uint32_t qA, uint32_t qB,
                                          uint32_t ab, ABFrag& a) {
    const __half2 abh = u2h2(ab);
    const __half2 alpha2 = __half2half2(__low2half(abh));
    const __half2 beta2 = __half2half2(__high2half(abh));
    const __half2 c2 = u2h2(0xE480E480u);  // f16 -1152
    uint32_t w[4];
    w[0] = 0x64006400u | (qA & 0x00FF00FFu);
    w[1] = 0x64006400u | ((qA >> 8) & 0x00FF00FFu);
    w[2] = 0x64006400u | (qB & 0x00FF00FFu);
    w[3] = 0x64006400u | ((qB >> 8) & 0x00FF00FFu);
#pragma unroll
    for (int i = 0; i < 4; ++i) {
        const __half2 v = __hadd2(u2h2(w[i]), c2);
        a.u[i] = h22u(__hfma2(v, alpha2, beta2));
    }
}

// One wave's software-pipelined K loop over NM matrices sharing the B
// panel (NM=2 for the FFN's w1/w3) and JT 16-token column tiles sharing
// the A (weight) stream. JT is THE batched-decode lever: the HBM weight
// stream and the dequant VALU are paid once for JT*16 tokens, so decode
// tokens/s scales nearly linearly in JT while the kernel wall barely
// moves (memory-bound).
//
// Pipeline shape (all learned from SQ counters / emitted waits):
//  * rotated two-deep UNCONDITIONAL prefetch of the weight batch — a
//    branch around loads makes hipcc drain vmcnt inside the body; the
//    last iteration overreads ONE batch (tail slack in every allocation),
//  * the (L2-resident) B-panel loads of batch g issue BEFORE the HBM
//    weight loads of batch g+1: vmcnt retires in issue order, so B loads
//    issued after the prefetch could not complete before it,
//  * per-batch pointers advance by constant strides (no per-load 64-bit
//    address math),
//  * wave-uniform loop bounds via readfirstlane (no exec-mask loop),
//  * accumulation is MFMA C-chained over two alternating accumulator
//    sets (covers dependent-accumulator latency).
// acc[n][jt][jj] ends with rows (l>>4)*4+jj, col jt*16 + (l&15).
// RT = row tiles per wave: the wave's B fragments feed RT A-tile streams,
// dividing the (L2-heavy) B-panel re-read traffic and the B-build VALU by
// RT. acc[rt][n][jt][jj].
// jtw/jt0: token-panel stride and first 16-token tile of this block in the
// xprep side channel (layout [kc][jtw][16][8]). Decode callers pass
// (JT, 0); the prefill *_mt kernels tile a T-wide panel with jtw = ceil(T/16)
// padded to the 4-tile group and jt0 = mtile*JT.
template <int WT, bool NORM, int NM, int JT, int RT = 1, int PF = 4,
          int NW = NWAVES>
__device__ __forceinline__ void wave_tile_kloop(
    const WMat2* const* ws, int tile_row,
    const unsigned short* __restrict__ xprep,
    const unsigned short* __restrict__ normprep,
    const float* __restrict__ ss_in, float eps,
    float acc[RT][NM][JT][4], int b0, int b1, int jtw = JT, int jt0 = 0) {
    KLoop kl;
    kl.init_range(b0, b1, (WT == W_F16) ? 1 : 4, NW);
    const int nb0 = ws[0]->cols >> 5;
    const int nb = (WT == W_F16) ? nb0 : ((nb0 + 3) & ~3);  // padded count
    // u32 per lane per K-block: 1 for q4 nibbles, 2 for the byte stream
    constexpr int QW = (WT == W_Q8B || WT == W_Q8B16) ? 2 : 1;
    // scale planes: W_Q8B16 carries one (alpha, beta) pair per 16
    // weights — lane k-span ks selects its half-plane (ks >> 1)
    constexpr int ABW = (WT == W_Q8B16) ? 2 : 1;
    const f32x4 zero = {0.f, 0.f, 0.f, 0.f};
    // With >=4 independent MFMAs per K-block (JT/NM/RT product) the
    // accumulator reuse distance already covers the MFMA dependent
    // latency — and the runtime `parity ? c1 : c0` reference select is
    // poison: the compiler materialized it as per-MFMA cndmask + AGPR
    // read/write round-trips (338 of 445 inner-loop instructions in the
    // JT=4 ffn kernel). Only the low-parallelism shapes keep the pair.
    constexpr int NP = (NM * JT * RT >= 4) ? 1 : 2;
    f32x4 c0[RT][NM][JT], c1[NP == 2 ? RT : 1][NM][JT];
#pragma unroll
    for (int rt = 0; rt < RT; ++rt)
#pragma unroll
        for (int n = 0; n < NM; ++n)
#pragma unroll
            for (int jt = 0; jt < JT; ++jt) {
                c0[rt][n][jt] = zero;
                if (NP == 2) c1[rt][n][jt] = zero;
            }

    // per-column-tile RMSNorm scale, hoisted (column j = lane&15 of tile jt)
    __half2 scale2[JT];
    if (NORM) {
        const float inv_cols = 1.0f / (float)ws[0]->cols;
#pragma unroll
        for (int jt = 0; jt < JT; ++jt)
            scale2[jt] = __float2half2_rn(
                rsqrtf(ss_in[(jt0 + jt) * 16 + kl.i] * inv_cols + eps));
    }

    const uint32_t* qp[RT][NM];
    const uint32_t* abp[RT][NM];
    const unsigned short* tp[RT][NM];
#pragma unroll
    for (int rt = 0; rt < RT; ++rt)
#pragma unroll
        for (int n = 0; n < NM; ++n) {
            const int tr = tile_row * RT + rt;
            qp[rt][n] = (const uint32_t*)ws[n]->data +
                        ((size_t)tr * nb + kl.kb0) * 64 * QW +
                        (kl.ks * 16 + kl.i) * 4 * QW;
            abp[rt][n] = (const uint32_t*)ws[n]->scales +
                         ((size_t)tr * nb + kl.kb0) * 16 * ABW +
                         (ABW == 2 ? (kl.ks >> 1) * 64 : 0) + kl.i * 4;
            tp[rt][n] = (const unsigned short*)ws[n]->data +
                        ((size_t)tr * (ws[n]->cols >> 3)) * 128 +
                        ((size_t)(kl.kb0 * 4 + kl.ks) * 16 + kl.i) * 8;
        }
    // xprep layout: element (kc, jt, j, e) at ((kc*jtw + jt)*16 + j)*8 + e
    const unsigned short* xp =
        xprep + ((size_t)(kl.kb0 * 4 + kl.ks) * jtw + jt0) * 128 + kl.i * 8;
    const unsigned short* np =
        normprep + (NORM ? (size_t)(kl.kb0 * 4 + kl.ks) * 8 : 0);

    struct Batch {  // weight stream only (HBM, nt, double-buffered)
        u32x4 q[(PF / 4) * QW][RT][NM], ab[PF / 4][RT][NM];
        uint4 aw[PF][RT][NM];
    };
    // NAMED buffers, never indexed by a runtime value (a runtime select
    // sends the array to scratch: measured 240-336 B/lane and 3x slower).
    Batch bufA, bufB;
    struct XPanel {  // B-panel side channel (L2-resident, per iteration)
        uint4 xb[PF][JT];
        uint4 nbv[PF];
    };

    auto load_w = [&](Batch& bt) {
#pragma unroll
        for (int u4 = 0; u4 < PF / 4; ++u4) {
#pragma unroll
            for (int rt = 0; rt < RT; ++rt)
#pragma unroll
                for (int n = 0; n < NM; ++n) {
                    if (WT == W_F16) {
#pragma unroll
                        for (int v = 0; v < 4; ++v)
                            bt.aw[u4 * 4 + v][rt][n] =
                                *reinterpret_cast<const uint4*>(
                                    tp[rt][n] + (u4 * 4 + v) * 512);
                    } else {
#pragma unroll
                        for (int qi = 0; qi < QW; ++qi)
                            bt.q[u4 * QW + qi][rt][n] =
                                __builtin_nontemporal_load(
                                    reinterpret_cast<const u32x4*>(
                                        qp[rt][n]) + u4 * 64 * QW + qi);
                        bt.ab[u4][rt][n] = __builtin_nontemporal_load(
                            reinterpret_cast<const u32x4*>(abp[rt][n]) +
                            u4 * 16 * ABW);
                    }
                }
        }
#pragma unroll
        for (int rt = 0; rt < RT; ++rt)
#pragma unroll
            for (int n = 0; n < NM; ++n) {
                if (WT == W_F16) {
                    tp[rt][n] += PF * 512;
                } else {
                    qp[rt][n] += PF * 64 * QW;
                    abp[rt][n] += PF * 16 * ABW;
                }
            }
    };

    auto load_x = [&](XPanel& px) {
#pragma unroll
        for (int u = 0; u < PF; ++u) {
#pragma unroll
            for (int jt = 0; jt < JT; ++jt)
                px.xb[u][jt] = *reinterpret_cast<const uint4*>(
                    xp + ((size_t)u * 4 * jtw + jt) * 128);  // 4 kc per kb
            if (NORM)
                px.nbv[u] = *reinterpret_cast<const uint4*>(np + u * 32);
        }
        xp += (size_t)PF * 4 * jtw * 128;
        if (NORM) np += PF * 32;
    };

    auto compute_one = [&](int parity, const uint32_t q[RT][NM][QW],
                           const uint32_t ab[RT][NM],
                           const uint4 aw[RT][NM], const uint4 xb[JT],
                           const uint4& nbv) {
        ABFrag b[JT];
#pragma unroll
        for (int jt = 0; jt < JT; ++jt) {
            if (NORM) {
                const uint32_t xw[4] = {xb[jt].x, xb[jt].y, xb[jt].z,
                                        xb[jt].w};
                const uint32_t nw[4] = {nbv.x, nbv.y, nbv.z, nbv.w};
#pragma unroll
                for (int w = 0; w < 4; ++w) {
                    const __half2 v = __hmul2(u2h2(xw[w]), u2h2(nw[w]));
                    b[jt].u[w] = h22u(__hmul2(v, scale2[jt]));
                }
            } else {
                b[jt].u[0] = xb[jt].x;
                b[jt].u[1] = xb[jt].y;
                b[jt].u[2] = xb[jt].z;
                b[jt].u[3] = xb[jt].w;
            }
        }
#pragma unroll
        for (int rt = 0; rt < RT; ++rt)
#pragma unroll
            for (int n = 0; n < NM; ++n) {
                ABFrag a;
                if (WT == W_F16) {
                    a.u[0] = aw[rt][n].x; a.u[1] = aw[rt][n].y;
                    a.u[2] = aw[rt][n].z; a.u[3] = aw[rt][n].w;
                } else if (WT == W_Q8B || WT == W_Q8B16) {
                    a_frag_q8(q[rt][n][0], q[rt][n][QW - 1], ab[rt][n],
                              a);
                } else {
                    a_frag_q4<WT>(q[rt][n][0], ab[rt][n], a);
                }
#pragma unroll
                for (int jt = 0; jt < JT; ++jt) {
                    f32x4& c = (NP == 2 && parity) ? c1[rt][n][jt]
                                                   : c0[rt][n][jt];
                    c = __builtin_amdgcn_mfma_f32_16x16x32_f16(a.v, b[jt].v,
                                                               c, 0, 0, 0);
                }
            }
    };

    auto compute_batch = [&](Batch& bt, XPanel& px) {
#pragma unroll
        for (int u = 0; u < PF; ++u) {
            uint32_t q[RT][NM][QW], ab[RT][NM];
#pragma unroll
            for (int rt = 0; rt < RT; ++rt)
#pragma unroll
                for (int n = 0; n < NM; ++n) {
                    if (WT == W_Q8B || WT == W_Q8B16) {
                        // lane's 8 u32 per group: [blk0.A, blk0.B, ...]
                        q[rt][n][0] = bt.q[u / 2][rt][n][(u % 2) * 2];
                        q[rt][n][QW - 1] =
                            bt.q[u / 2][rt][n][(u % 2) * 2 + 1];
                    } else {
                        q[rt][n][0] = bt.q[u / 4][rt][n][u % 4];
                    }
                    ab[rt][n] = bt.ab[u / 4][rt][n][u % 4];
                }
            compute_one(u & 1, q, ab, bt.aw[u], px.xb[u], px.nbv[u]);
        }
    };

    // Per iteration: [load B-panel (L2) -> prefetch next weights (HBM) ->
    // compute]. The B-panel loads retire first (vmcnt is in-order and they
    // are issued before the HBM batch), so compute's wait never drains the
    // prefetch; one live XPanel keeps register pressure down (a fully
    // double-buffered B panel measured 9% SLOWER end to end).
    const int nfull = (kl.kb1 - kl.kb0) / PF;
    if (nfull > 0) {
        load_w(bufA);
        int it = 0;
        while (true) {
            XPanel pxA;
            load_x(pxA);
            load_w(bufB);        // prefetch next weights (may overread 1)
            compute_batch(bufA, pxA);
            if (++it == nfull) break;
            XPanel pxB;
            load_x(pxB);
            load_w(bufA);
            compute_batch(bufB, pxB);
            if (++it == nfull) break;
        }
        // load_w ran nfull+1 times (the trailing prefetch) but only
        // nfull batches were consumed — rewind one batch so the tail
        // loop reads the right K-blocks. (Latent until the prefill _mt
        // kernels: 4-aligned q4 ranges never leave a tail, and the f16
        // slab kernels' per-wave ranges are < PF so nfull == 0.)
#pragma unroll
        for (int rt = 0; rt < RT; ++rt)
#pragma unroll
            for (int n = 0; n < NM; ++n) {
                if (WT == W_F16) {
                    tp[rt][n] -= PF * 512;
                } else {
                    qp[rt][n] -= PF * 64 * QW;
                    abp[rt][n] -= PF * 16 * ABW;
                }
            }
    }
    for (int g = kl.kb0 + nfull * PF; g < kl.kb1; ++g) {
        uint32_t q[RT][NM][QW], ab[RT][NM];
        uint4 aw[RT][NM];
        uint4 xb[JT], nbv;
#pragma unroll
        for (int jt = 0; jt < JT; ++jt)
            xb[jt] = *reinterpret_cast<const uint4*>(xp + jt * 128);
        xp += (size_t)4 * jtw * 128;
        if (NORM) {
            nbv = *reinterpret_cast<const uint4*>(np);
            np += 32;
        }
#pragma unroll
        for (int rt = 0; rt < RT; ++rt)
#pragma unroll
            for (int n = 0; n < NM; ++n) {
                if (WT == W_F16) {
                    aw[rt][n] = *reinterpret_cast<const uint4*>(tp[rt][n]);
                    tp[rt][n] += 512;
                } else {
#pragma unroll
                    for (int qi = 0; qi < QW; ++qi)
                        q[rt][n][qi] =
                            __builtin_nontemporal_load(qp[rt][n] + qi);
                    qp[rt][n] += 64 * QW;
                    ab[rt][n] = __builtin_nontemporal_load(abp[rt][n]);
                    abp[rt][n] += 16;
                }
            }
        compute_one(g & 1, q, ab, aw, xb, nbv);
    }
#pragma unroll
    for (int rt = 0; rt < RT; ++rt)
#pragma unroll
        for (int n = 0; n < NM; ++n)
#pragma unroll
            for (int jt = 0; jt < JT; ++jt)
#pragma unroll
                for (int jj = 0; jj < 4; ++jj)
                    acc[rt][n][jt][jj] =
                        (NP == 2) ? c0[rt][n][jt][jj] + c1[rt][n][jt][jj]
                                  : c0[rt][n][jt][jj];
}

// LDS combine of the NW waves' partial accumulators; wave 0 ends with
// the full 16x16 tile (4 rows x 1 col per lane). NACC = accumulator sets.
template <int NACC, int NW = NWAVES>
__device__ __forceinline__ void combine_acc(float acc[NACC][4],
                                            float* lds /* (NW-1)*64*4*NACC */) {
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x & (WAVE - 1);
    if (wid > 0) {
        float* dst = lds + (((wid - 1) * 64 + lane) * 4) * NACC;
#pragma unroll
        for (int n = 0; n < NACC; ++n)
#pragma unroll
            for (int jj = 0; jj < 4; ++jj) dst[n * 4 + jj] = acc[n][jj];
    }
    __syncthreads();
    if (wid == 0) {
#pragma unroll
        for (int w = 0; w < NW - 1; ++w) {
            const float* src = lds + ((w * 64 + lane) * 4) * NACC;
#pragma unroll
            for (int n = 0; n < NACC; ++n)
#pragma unroll
                for (int jj = 0; jj < 4; ++jj) acc[n][jj] += src[n * 4 + jj];
        }
    }
}

__device__ __forceinline__ float norm_scale(const float* ss, int j, int cols,
                                            float eps) {
    return rsqrtf(ss[j] / (float)cols + eps);
}

// ------------------------------------------------------------- k_prep_x
// sumsq + bf16 xprep of an incoming f32 activation block (forward entry,
// split-K gemm epilogue, logits entry). Grid: (T, C) — C column chunks per
// row so the launch fills more than T CUs; ss[t] accumulates via atomics
// and MUST be zeroed before the launch (a T-block single-chunk version
// measured 4.6 µs — launch+latency bound on 16 CUs of 256).
__global__ void k_prep_x(const float* __restrict__ x,
                         unsigned short* __restrict__ xprep,
                         float* __restrict__ ss, int cols, int jtw) {
    const int t = blockIdx.x;
    const float* xt = x + (size_t)t * cols;
    __shared__ float red[NWAVES];
    const int nkc = cols >> 3;
    const int per = (nkc + gridDim.y - 1) / gridDim.y;
    const int kc0 = blockIdx.y * per;
    const int kc1 = min(nkc, kc0 + per);
    float sum = 0.0f;
    for (int kc = kc0 + threadIdx.x; kc < kc1; kc += BLOCK) {
        const float4 a = *reinterpret_cast<const float4*>(xt + kc * 8);
        const float4 b = *reinterpret_cast<const float4*>(xt + kc * 8 + 4);
        sum += a.x * a.x + a.y * a.y + a.z * a.z + a.w * a.w;
        sum += b.x * b.x + b.y * b.y + b.z * b.z + b.w * b.w;
        uint4 o;
        o.x = pack_f16(a.x, a.y);
        o.y = pack_f16(a.z, a.w);
        o.z = pack_f16(b.x, b.y);
        o.w = pack_f16(b.z, b.w);
        *reinterpret_cast<uint4*>(
            xprep + (((size_t)kc * jtw + (t >> 4)) * 16 + (t & 15)) * 8) = o;
    }
    sum = wave_reduce_sum(sum);
    const int wid = threadIdx.x / WAVE;
    if ((threadIdx.x & (WAVE - 1)) == 0) red[wid] = sum;
    __syncthreads();
    if (threadIdx.x == 0)
        atomicAdd(ss + t, red[0] + red[1] + red[2] + red[3]);
}

// ------------------------------------------------------- k_reduce_prep
// Split-K reducer + side-channel prep in one pass: y[t] += sum of the KS
// slab partials, then sumsq + f16 xprep of the result (replaces atomic
// split-K partials + a separate k_prep_x — the 819K atomicAdd/kernel of
// the atomic form dominated at JT=4). Grid (T, C); ss zeroed upstream.
__global__ void k_reduce_prep(float* __restrict__ y,
                              const float* __restrict__ slab, int ks,
                              unsigned short* __restrict__ xprep,
                              float* __restrict__ ss, int cols, int jtw) {
    const int t = blockIdx.x;
    float* yt = y + (size_t)t * cols;
    __shared__ float red[NWAVES];
    const int nkc = cols >> 3;
    const int per = (nkc + gridDim.y - 1) / gridDim.y;
    const int kc0 = blockIdx.y * per;
    const int kc1 = min(nkc, kc0 + per);
    float sum = 0.0f;
    for (int kc = kc0 + threadIdx.x; kc < kc1; kc += BLOCK) {
        float4 a = *reinterpret_cast<const float4*>(yt + kc * 8);
        float4 b = *reinterpret_cast<const float4*>(yt + kc * 8 + 4);
        const int e0 = kc * 8;            // 8 consecutive rows
        const int tile = e0 >> 4;         // 16-row tile
        const int i0 = e0 & 15;           // 0 or 8 within the tile
        const float* sbase =
            slab + ((size_t)tile * ks * 64 + t) * 16 + i0;
        const size_t kstride = (size_t)64 * 16;
        for (int k = 0; k < ks; ++k) {
            const float4 p0 = *reinterpret_cast<const float4*>(
                sbase + k * kstride);
            const float4 p1 = *reinterpret_cast<const float4*>(
                sbase + k * kstride + 4);
            a.x += p0.x; a.y += p0.y; a.z += p0.z; a.w += p0.w;
            b.x += p1.x; b.y += p1.y; b.z += p1.z; b.w += p1.w;
        }
        *reinterpret_cast<float4*>(yt + kc * 8) = a;
        *reinterpret_cast<float4*>(yt + kc * 8 + 4) = b;
        sum += a.x * a.x + a.y * a.y + a.z * a.z + a.w * a.w;
        sum += b.x * b.x + b.y * b.y + b.z * b.z + b.w * b.w;
        uint4 o;
        o.x = pack_f16(a.x, a.y);
        o.y = pack_f16(a.z, a.w);
        o.z = pack_f16(b.x, b.y);
        o.w = pack_f16(b.z, b.w);
        *reinterpret_cast<uint4*>(
            xprep + (((size_t)kc * jtw + (t >> 4)) * 16 + (t & 15)) * 8) = o;
    }
    sum = wave_reduce_sum(sum);
    const int wid = threadIdx.x / WAVE;
    if ((threadIdx.x & (WAVE - 1)) == 0) red[wid] = sum;
    __syncthreads();
    if (threadIdx.x == 0)
        atomicAdd(ss + t, red[0] + red[1] + red[2] + red[3]);
}

// ------------------------------------------------------------- k_gemm16
// MODE GM_ATOMIC: grid-level split-K — gridDim.y blocks per row-tile each
// cover nb/gridDim.y K-blocks and atomicAdd their partial tile into y
// (which already holds the residual stream); a following k_prep_x pass
// rebuilds the sumsq/xprep side-channels. Used when rows/16 alone cannot
// fill 256 CUs (wo: E/16 = 200 blocks).
template <int WT, int MODE, int JT, int RT = 1>
__global__ __launch_bounds__(BLOCK) void k_gemm16(
    WMat2 w, const unsigned short* __restrict__ bprep,
    const unsigned short* __restrict__ normprep,
    const float* __restrict__ ss_in, float eps, float* __restrict__ y,
    unsigned short* __restrict__ xprep_out, float* __restrict__ ss_out,
    int T) {
    constexpr bool NORM = (MODE == GM_NORM_PLAIN);
    const int lane = threadIdx.x & (WAVE - 1);
    const int j = lane & 15;
    float acc[RT][1][JT][4];
    const WMat2* ws[1] = {&w};
    const int nbt = w.cols >> 5;
    const int nbk = (WT == W_F16) ? nbt : ((nbt + 3) & ~3);  // padded
    int b0 = 0, b1 = nbk;
    if (MODE == GM_ATOMIC || MODE == GM_SLAB) {
        // split in 4-aligned units so q4 group loads stay whole
        int per = (nbk + gridDim.y - 1) / gridDim.y;
        per = (per + 3) & ~3;
        b0 = min(nbk, (int)blockIdx.y * per);
        b1 = min(nbk, b0 + per);
    }
    wave_tile_kloop<WT, NORM, 1, JT, RT>(ws, blockIdx.x, bprep, normprep,
                                         ss_in, eps, acc, b0, b1);
    __shared__ float lds[3 * 64 * 4 * JT * RT];
    combine_acc<RT * JT>(reinterpret_cast<float(*)[4]>(acc), lds);
    if (threadIdx.x >= WAVE) return;
    const int r0 = blockIdx.x * 16 + (lane >> 4) * 4;  // RT==1 modes
    if (MODE == GM_ATOMIC) {
#pragma unroll
        for (int jt = 0; jt < JT; ++jt) {
            const int j2 = jt * 16 + j;
            if (j2 < T) {
#pragma unroll
                for (int jj = 0; jj < 4; ++jj)
                    atomicAdd(y + (size_t)j2 * w.rows + r0 + jj,
                              acc[0][0][jt][jj]);
            }
        }
        return;
    }
    if (MODE == GM_SLAB) {
        // write-through-store the partial tile(s); k_reduce_prep sums the
        // KS slices at the next kernel boundary (no atomics, no
        // pre-zeroing). Slab layout: f32[R][KS][kMaxTok][16 rows]; `y` is
        // reused as the slab base. With RT>1 a block covers RT row tiles
        // sharing its B-panel reads.
#pragma unroll
        for (int rt = 0; rt < RT; ++rt) {
            float* slab = y +
                ((size_t)(blockIdx.x * RT + rt) * gridDim.y + blockIdx.y) *
                    64 * 16;
#pragma unroll
            for (int jt = 0; jt < JT; ++jt) {
                const int j2 = jt * 16 + j;
                float4 v;
                v.x = acc[rt][0][jt][0];
                v.y = acc[rt][0][jt][1];
                v.z = acc[rt][0][jt][2];
                v.w = acc[rt][0][jt][3];
                store_f4_wt(slab + ((size_t)j2 * 16) + (lane >> 4) * 4, v);
            }
        }
        return;
    }
    // wave-0 fused epilogue: rows r0..r0+3 per row tile, col jt*16+j
#pragma unroll
    for (int rt = 0; rt < RT; ++rt) {
    const int r0 = (blockIdx.x * RT + rt) * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int jt = 0; jt < JT; ++jt) {
        const int j2 = jt * 16 + j;
        float sq = 0.0f;
#pragma unroll
        for (int jj = 0; jj < 4; ++jj) {
            const int row = r0 + jj;
            float v = acc[rt][0][jt][jj];
            if (MODE == GM_RES_SQ) {
                if (j2 < T) {
                    v += y[(size_t)j2 * w.rows + row];
                    y[(size_t)j2 * w.rows + row] = v;
                    sq += v * v;
                }
            } else {
                if (j2 < T) y[(size_t)j2 * w.rows + row] = v;
            }
            acc[rt][0][jt][jj] = v;
        }
        if (MODE == GM_RES_SQ && xprep_out != nullptr && j2 < T) {
            // 4 consecutive rows -> one aligned 8 B f16x4 chunk of xprep
            uint2 o;
            o.x = pack_f16(acc[rt][0][jt][0], acc[rt][0][jt][1]);
            o.y = pack_f16(acc[rt][0][jt][2], acc[rt][0][jt][3]);
            *reinterpret_cast<uint2*>(
                xprep_out + (((size_t)(r0 >> 3) * JT + jt) * 16 + j) * 8 +
                (r0 & 7)) = o;
        }
        if (MODE == GM_RES_SQ && ss_out != nullptr) {
            float s2 = sq;
            s2 += __shfl_xor(s2, 16);
            s2 += __shfl_xor(s2, 32);
            if (lane < 16 && j2 < T) atomicAdd(ss_out + j2, s2);
        }
    }
    }
}

// ------------------------------------------------------------- k_qkv16
// QKV projections on MFMA + fused input RMSNorm + RoPE + KV append.
// GQA: wk/wv have Ekv = Hkv*D rows (< E); KV cache rows are Ekv wide.
// Tile space is [0, E/16) for wq then [.., +Ekv/16) each for wk/wv.
template <int WT, int JT, int RT = 1>
__global__ __launch_bounds__(BLOCK) void k_qkv16(
    WMat2 wq, WMat2 wk, WMat2 wv, const unsigned short* __restrict__ xprep,
    const unsigned short* __restrict__ normprep,
    const float* __restrict__ ss_in, float eps, float* __restrict__ q_buf,
    __half* __restrict__ k_cache, __half* __restrict__ v_cache,
    const int* __restrict__ pos, const int* __restrict__ seq,
    const float* __restrict__ inv_freq, int E, int Ekv, int D, int n_ctx,
    int T) {
    const int tq = (E >> 4) / RT;
    const int tk = (Ekv >> 4) / RT;
    const int mat = (blockIdx.x < tq) ? 0
                    : (blockIdx.x < tq + tk) ? 1 : 2;
    const int tile = blockIdx.x - ((mat == 0) ? 0 : (mat == 1) ? tq
                                                               : tq + tk);
    const WMat2& w = (mat == 0) ? wq : (mat == 1) ? wk : wv;
    const int lane = threadIdx.x & (WAVE - 1);
    const int j = lane & 15;
    float acc[RT][1][JT][4];
    const WMat2* ws[1] = {&w};
    const int nbe = (WT == W_F16) ? (E >> 5) : (((E >> 5) + 3) & ~3);
    wave_tile_kloop<WT, true, 1, JT, RT>(ws, tile, xprep, normprep, ss_in,
                                         eps, acc, 0, nbe);
    __shared__ float lds[3 * 64 * 4 * RT * JT];
    combine_acc<RT * JT>(reinterpret_cast<float(*)[4]>(acc), lds);
    if (threadIdx.x >= WAVE) return;
#pragma unroll
    for (int rt = 0; rt < RT; ++rt) {
    const int r0 = (tile * RT + rt) * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int jt = 0; jt < JT; ++jt) {
        const int j2 = jt * 16 + j;
        if (j2 >= T) continue;
        const int p = pos[j2];
        if (mat == 2) {  // V rows: straight f16 cache append
            __half* dst =
                v_cache + ((size_t)seq[j2] * n_ctx + p) * Ekv + r0;
#pragma unroll
            for (int jj = 0; jj < 4; ++jj)
                dst[jj] = __float2half(acc[rt][0][jt][jj]);
            continue;
        }
        // q/k: RoPE on in-lane pairs (rows r0+2q2, r0+2q2+1)
#pragma unroll
        for (int q2 = 0; q2 < 2; ++q2) {
            const int e = r0 + 2 * q2;
            const int d = e % D;
            const float theta = (float)p * inv_freq[d >> 1];
            float sn, cs;
            __sincosf(theta, &sn, &cs);
            const float x0 = acc[rt][0][jt][2 * q2];
            const float x1 = acc[rt][0][jt][2 * q2 + 1];
            const float o0 = x0 * cs - x1 * sn;
            const float o1 = x0 * sn + x1 * cs;
            if (mat == 0) {
                q_buf[(size_t)j2 * E + e] = o0;
                q_buf[(size_t)j2 * E + e + 1] = o1;
            } else {
                __half* dst =
                    k_cache + ((size_t)seq[j2] * n_ctx + p) * Ekv + e;
                dst[0] = __float2half(o0);
                dst[1] = __float2half(o1);
            }
        }
    }
    }
}

// ------------------------------------------------------------- k_ffn16
// w1 + w3 against the same B panel + fused input RMSNorm + SwiGLU; emits
// the gate product straight into gprep (f16 B-layout over F).
template <int WT, int JT, int RT = 1>
__global__ __launch_bounds__(BLOCK) void k_ffn16(
    WMat2 w1, WMat2 w3, const unsigned short* __restrict__ xprep,
    const unsigned short* __restrict__ normprep,
    const float* __restrict__ ss_in, float eps,
    unsigned short* __restrict__ gprep, int T) {
    const int lane = threadIdx.x & (WAVE - 1);
    const int j = lane & 15;
    float acc[RT][2][JT][4];
    const WMat2* ws[2] = {&w1, &w3};
    const int nbf = (WT == W_F16) ? (w1.cols >> 5)
                                  : (((w1.cols >> 5) + 3) & ~3);
    wave_tile_kloop<WT, true, 2, JT, RT>(ws, blockIdx.x, xprep, normprep,
                                         ss_in, eps, acc, 0, nbf);
    __shared__ float lds[3 * 64 * 4 * RT * 2 * JT];
    combine_acc<RT * 2 * JT>(reinterpret_cast<float(*)[4]>(acc), lds);
    if (threadIdx.x >= WAVE) return;
#pragma unroll
    for (int rt = 0; rt < RT; ++rt) {
        const int r0 = (blockIdx.x * RT + rt) * 16 + (lane >> 4) * 4;
#pragma unroll
        for (int jt = 0; jt < JT; ++jt) {
            const int j2 = jt * 16 + j;
            if (j2 >= T) continue;
            float g[4];
#pragma unroll
            for (int jj = 0; jj < 4; ++jj) {
                const float v1 = acc[rt][0][jt][jj];
                const float silu = v1 / (1.0f + __expf(-v1));
                g[jj] = silu * acc[rt][1][jt][jj];
            }
            uint2 o;
            o.x = pack_f16(g[0], g[1]);
            o.y = pack_f16(g[2], g[3]);
            *reinterpret_cast<uint2*>(
                gprep + (((size_t)(r0 >> 3) * JT + jt) * 16 + j) * 8 +
                (r0 & 7)) = o;
        }
    }
}

// ------------------------------------------------- slab qkv/ffn variants
// For models whose tile counts underfill the chip (3B: qkv 600 tiles, ffn
// 540), the fused kernels run at ~2.3 blocks/CU with the full B panel
// re-read per tile. These variants use RT=2 row tiles per wave (halves
// the B-panel L2 traffic and B-build VALU) plus grid-level split-K into
// slabs; the finish kernels sum the slabs and run the rope/cache (qkv) or
// SwiGLU (ffn) epilogue.

template <int WT, int JT, int RT>
__global__ __launch_bounds__(BLOCK) void k_qkv16_slab(
    WMat2 wq, WMat2 wk, WMat2 wv, const unsigned short* __restrict__ xprep,
    const unsigned short* __restrict__ normprep,
    const float* __restrict__ ss_in, float eps, float* __restrict__ slab,
    int E, int Ekv, int T) {
    const int tq = (E >> 4) / RT;
    const int tk = (Ekv >> 4) / RT;
    const int mat = (blockIdx.x < tq) ? 0
                    : (blockIdx.x < tq + tk) ? 1 : 2;
    const int stile = blockIdx.x - ((mat == 0) ? 0 : (mat == 1) ? tq
                                                                : tq + tk);
    const WMat2& w = (mat == 0) ? wq : (mat == 1) ? wk : wv;
    const int lane = threadIdx.x & (WAVE - 1);
    const int j = lane & 15;
    float acc[RT][1][JT][4];
    const WMat2* ws[1] = {&w};
    const int nbe = (WT == W_F16) ? (E >> 5) : (((E >> 5) + 3) & ~3);
    int per = (nbe + gridDim.y - 1) / gridDim.y;
    per = (per + 3) & ~3;
    const int b0 = min(nbe, (int)blockIdx.y * per);
    const int b1 = min(nbe, b0 + per);
    wave_tile_kloop<WT, true, 1, JT, RT>(ws, stile, xprep, normprep, ss_in,
                                         eps, acc, b0, b1);
    __shared__ float lds[3 * 64 * 4 * RT * JT];
    combine_acc<RT * JT>(reinterpret_cast<float(*)[4]>(acc), lds);
    if (threadIdx.x >= WAVE) return;
    // slab gtile space: q rows [0, E/16) then k, v rows [.., +Ekv/16)
    const int gbase = (mat == 0) ? 0
                      : (mat == 1) ? (E >> 4)
                                   : (E >> 4) + (Ekv >> 4);
#pragma unroll
    for (int rt = 0; rt < RT; ++rt) {
        const size_t gtile = (size_t)gbase + stile * RT + rt;
        float* sl = slab + ((gtile * gridDim.y + blockIdx.y) * 64) * 16;
#pragma unroll
        for (int jt = 0; jt < JT; ++jt) {
            const int j2 = jt * 16 + j;
            float4 v;
            v.x = acc[rt][0][jt][0];
            v.y = acc[rt][0][jt][1];
            v.z = acc[rt][0][jt][2];
            v.w = acc[rt][0][jt][3];
            store_f4_wt(sl + (size_t)j2 * 16 + (lane >> 4) * 4, v);
        }
    }
}

// Sum the qkv slabs, apply RoPE, write q_buf / KV caches. One thread per
// (mat, 8-row chunk, token): vectorized float4 slab reads and uint4/f16x8
// stores (scalar 2-4 B accesses made the first version a 5 us kernel).
__global__ void k_qkv_finish(const float* __restrict__ slab, int ks,
                             float* __restrict__ q_buf,
                             __half* __restrict__ k_cache,
                             __half* __restrict__ v_cache,
                             const int* __restrict__ pos,
                             const int* __restrict__ seq,
                             const float* __restrict__ inv_freq, int E,
                             int Ekv, int D, int n_ctx, int T) {
    const int nchunks = (E + 2 * Ekv) / 8;
    const int idx = blockIdx.x * BLOCK + threadIdx.x;
    if (idx >= nchunks * T) return;
    const int chunk = idx % nchunks;
    const int t = idx / nchunks;
    const int r = 8 * chunk;  // global row in the [E | Ekv | Ekv] space
    const int mat = (r < E) ? 0 : (r < E + Ekv) ? 1 : 2;
    const int e = (mat == 0) ? r : (mat == 1) ? r - E : r - E - Ekv;
    const size_t gt0 = ((size_t)(r >> 4)) * ks;  // 8-aligned within mat
    float v[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) v[i] = 0.f;
    for (int k = 0; k < ks; ++k) {
        const float* sl = slab + ((gt0 + k) * 64 + t) * 16 + (e & 15);
        const float4 a = *reinterpret_cast<const float4*>(sl);
        const float4 b = *reinterpret_cast<const float4*>(sl + 4);
        v[0] += a.x; v[1] += a.y; v[2] += a.z; v[3] += a.w;
        v[4] += b.x; v[5] += b.y; v[6] += b.z; v[7] += b.w;
    }
    const int p = pos[t];
    if (mat == 2) {
        __half* dst = v_cache + ((size_t)seq[t] * n_ctx + p) * Ekv + e;
        uint4 o;
        o.x = pack_f16(v[0], v[1]);
        o.y = pack_f16(v[2], v[3]);
        o.z = pack_f16(v[4], v[5]);
        o.w = pack_f16(v[6], v[7]);
        *reinterpret_cast<uint4*>(dst) = o;
        return;
    }
    float o[8];
#pragma unroll
    for (int q2 = 0; q2 < 4; ++q2) {
        const int d = (e + 2 * q2) % D;
        const float theta = (float)p * inv_freq[d >> 1];
        float sn, cs;
        __sincosf(theta, &sn, &cs);
        o[2 * q2] = v[2 * q2] * cs - v[2 * q2 + 1] * sn;
        o[2 * q2 + 1] = v[2 * q2] * sn + v[2 * q2 + 1] * cs;
    }
    if (mat == 0) {
        float* dst = q_buf + (size_t)t * E + e;
        *reinterpret_cast<float4*>(dst) =
            make_float4(o[0], o[1], o[2], o[3]);
        *reinterpret_cast<float4*>(dst + 4) =
            make_float4(o[4], o[5], o[6], o[7]);
    } else {
        __half* dst = k_cache + ((size_t)seq[t] * n_ctx + p) * Ekv + e;
        uint4 w;
        w.x = pack_f16(o[0], o[1]);
        w.y = pack_f16(o[2], o[3]);
        w.z = pack_f16(o[4], o[5]);
        w.w = pack_f16(o[6], o[7]);
        *reinterpret_cast<uint4*>(dst) = w;
    }
}

template <int WT, int JT, int RT>
__global__ __launch_bounds__(BLOCK) void k_ffn16_slab(
    WMat2 w1, WMat2 w3, const unsigned short* __restrict__ xprep,
    const unsigned short* __restrict__ normprep,
    const float* __restrict__ ss_in, float eps, float* __restrict__ slab,
    int T) {
    const int lane = threadIdx.x & (WAVE - 1);
    const int j = lane & 15;
    float acc[RT][2][JT][4];
    const WMat2* ws[2] = {&w1, &w3};
    const int nbf = (WT == W_F16) ? (w1.cols >> 5)
                                  : (((w1.cols >> 5) + 3) & ~3);
    int per = (nbf + gridDim.y - 1) / gridDim.y;
    per = (per + 3) & ~3;
    const int b0 = min(nbf, (int)blockIdx.y * per);
    const int b1 = min(nbf, b0 + per);
    wave_tile_kloop<WT, true, 2, JT, RT>(ws, blockIdx.x, xprep, normprep,
                                         ss_in, eps, acc, b0, b1);
    __shared__ float lds[3 * 64 * 4 * RT * 2 * JT];
    combine_acc<RT * 2 * JT>(reinterpret_cast<float(*)[4]>(acc), lds);
    if (threadIdx.x >= WAVE) return;
    const int Ftiles = w1.rows >> 4;
#pragma unroll
    for (int rt = 0; rt < RT; ++rt) {
#pragma unroll
        for (int n = 0; n < 2; ++n) {
            const size_t gtile = (size_t)n * Ftiles + blockIdx.x * RT + rt;
            float* sl = slab + ((gtile * gridDim.y + blockIdx.y) * 64) * 16;
#pragma unroll
            for (int jt = 0; jt < JT; ++jt) {
                const int j2 = jt * 16 + j;
                float4 v;
                v.x = acc[rt][n][jt][0];
                v.y = acc[rt][n][jt][1];
                v.z = acc[rt][n][jt][2];
                v.w = acc[rt][n][jt][3];
                store_f4_wt(sl + (size_t)j2 * 16 + (lane >> 4) * 4, v);
            }
        }
    }
}

// Sum the w1/w3 slabs, SwiGLU, emit gprep. One thread per (8-row chunk,
// token): float4 slab reads, one uint4 (f16x8) gprep store.
__global__ void k_ffn_finish(const float* __restrict__ slab, int ks,
                             unsigned short* __restrict__ gprep, int F,
                             int T, int jtw) {
    const int nchunks = F / 8;
    const int idx = blockIdx.x * BLOCK + threadIdx.x;
    if (idx >= nchunks * T) return;
    const int chunk = idx % nchunks;
    const int t = idx / nchunks;
    const int r = chunk * 8;
    const int Ftiles = F >> 4;
    const size_t t1 = ((size_t)(r >> 4)) * ks;
    const size_t t3 = ((size_t)Ftiles + (r >> 4)) * ks;
    float s1[8], s3[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) { s1[i] = 0.f; s3[i] = 0.f; }
    for (int k = 0; k < ks; ++k) {
        const float* p1 = slab + ((t1 + k) * 64 + t) * 16 + (r & 15);
        const float* p3 = slab + ((t3 + k) * 64 + t) * 16 + (r & 15);
        const float4 a1 = *reinterpret_cast<const float4*>(p1);
        const float4 b1 = *reinterpret_cast<const float4*>(p1 + 4);
        const float4 a3 = *reinterpret_cast<const float4*>(p3);
        const float4 b3 = *reinterpret_cast<const float4*>(p3 + 4);
        s1[0] += a1.x; s1[1] += a1.y; s1[2] += a1.z; s1[3] += a1.w;
        s1[4] += b1.x; s1[5] += b1.y; s1[6] += b1.z; s1[7] += b1.w;
        s3[0] += a3.x; s3[1] += a3.y; s3[2] += a3.z; s3[3] += a3.w;
        s3[4] += b3.x; s3[5] += b3.y; s3[6] += b3.z; s3[7] += b3.w;
    }
    float g[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) {
        const float silu = s1[i] / (1.0f + __expf(-s1[i]));
        g[i] = silu * s3[i];
    }
    uint4 o;
    o.x = pack_f16(g[0], g[1]);
    o.y = pack_f16(g[2], g[3]);
    o.z = pack_f16(g[4], g[5]);
    o.w = pack_f16(g[6], g[7]);
    *reinterpret_cast<uint4*>(
        gprep + (((size_t)(r >> 3) * jtw + (t >> 4)) * 16 + (t & 15)) * 8) =
        o;
}

// ================================================== prefill (large-M) path
// Hand-written large-M dequant-GEMM: retires the rocBLAS-over-detiled-f16
// prefill path (q4/byte/f16 tiles read DIRECTLY, no f16 weight copy) and
// the 64-token host tiling. One launch covers all T tokens: the grid is
// (gtile row-tiles) x (MT 64-token groups), flattened with an XCD-aware
// decode so all MT token-groups of one row tile land on the SAME XCD
// back-to-back — the row tile's weight stream is read from HBM once and
// re-served to the other groups from that XCD's L2 (blockIdx -> XCD is
// round-robin on dispatch order, 8 XCDs; MI355X_MICROARCH.md).
//
// Returns (gtile, mtile); gtile may be >= GT (pad block -> caller exits).
__device__ __forceinline__ void xcd_decode(int GT, int MT, int& gtile,
                                           int& mtile) {
    const int xcd = blockIdx.x & 7;
    const int q = blockIdx.x >> 3;
    mtile = q % MT;
    gtile = (q / MT) * 8 + xcd;
    (void)GT;
}

static inline int xcd_grid(int GT, int MT) {
    return ((GT + 7) & ~7) * MT;
}

// QKV projections for token group mtile + RoPE + KV append (the large-M
// analog of k_qkv16; epilogue identical, tokens indexed globally).
// RT row tiles per wave share the B panel (its load + norm-build cost).
template <int WT, int JT = 4, int RT = 1>
__global__ __launch_bounds__(BLOCK) void k_qkv16_mt(
    WMat2 wq, WMat2 wk, WMat2 wv, const unsigned short* __restrict__ xprep,
    const unsigned short* __restrict__ normprep,
    const float* __restrict__ ss_in, float eps, float* __restrict__ q_buf,
    __half* __restrict__ k_cache, __half* __restrict__ v_cache,
    const int* __restrict__ pos, const int* __restrict__ seq,
    const float* __restrict__ inv_freq, int E, int Ekv, int D, int n_ctx,
    int T, int MT, int jtw) {
    int gtile, mtile;
    const int tq = (E >> 4) / RT;
    const int tk = (Ekv >> 4) / RT;
    const int GT = tq + 2 * tk;
    xcd_decode(GT, MT, gtile, mtile);
    if (gtile >= GT) return;
    const int mat = (gtile < tq) ? 0 : (gtile < tq + tk) ? 1 : 2;
    const int tile = gtile - ((mat == 0) ? 0 : (mat == 1) ? tq : tq + tk);
    const WMat2& w = (mat == 0) ? wq : (mat == 1) ? wk : wv;
    const int lane = threadIdx.x & (WAVE - 1);
    const int j = lane & 15;
    float acc[RT][1][JT][4];
    const WMat2* ws[1] = {&w};
    const int nbe = (WT == W_F16) ? (E >> 5) : (((E >> 5) + 3) & ~3);
    wave_tile_kloop<WT, true, 1, JT, RT>(ws, tile, xprep, normprep, ss_in,
                                         eps, acc, 0, nbe, jtw,
                                         mtile * JT);
    __shared__ float lds[3 * 64 * 4 * JT * RT];
    combine_acc<RT * JT>(reinterpret_cast<float(*)[4]>(acc), lds);
    if (threadIdx.x >= WAVE) return;
#pragma unroll
    for (int rt = 0; rt < RT; ++rt) {
    const int r0 = (tile * RT + rt) * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int jt = 0; jt < JT; ++jt) {
        const int j2 = (mtile * JT + jt) * 16 + j;
        if (j2 >= T) continue;
        const int p = pos[j2];
        if (mat == 2) {
            __half* dst =
                v_cache + ((size_t)seq[j2] * n_ctx + p) * Ekv + r0;
#pragma unroll
            for (int jj = 0; jj < 4; ++jj)
                dst[jj] = __float2half(acc[rt][0][jt][jj]);
            continue;
        }
#pragma unroll
        for (int q2 = 0; q2 < 2; ++q2) {
            const int e = r0 + 2 * q2;
            const int d = e % D;
            const float theta = (float)p * inv_freq[d >> 1];
            float sn, cs;
            __sincosf(theta, &sn, &cs);
            const float x0 = acc[rt][0][jt][2 * q2];
            const float x1 = acc[rt][0][jt][2 * q2 + 1];
            const float o0 = x0 * cs - x1 * sn;
            const float o1 = x0 * sn + x1 * cs;
            if (mat == 0) {
                q_buf[(size_t)j2 * E + e] = o0;
                q_buf[(size_t)j2 * E + e + 1] = o1;
            } else {
                __half* dst =
                    k_cache + ((size_t)seq[j2] * n_ctx + p) * Ekv + e;
                dst[0] = __float2half(o0);
                dst[1] = __float2half(o1);
            }
        }
    }
    }
}

// Large-M GEMM with the fused residual + sumsq + xprep epilogue (wo / w2
// consumers) or plain store. RES_SQ semantics match k_gemm16<GM_RES_SQ>.
template <int WT, bool RES_SQ, int JT = 4, int RT = 1>
__global__ __launch_bounds__(BLOCK) void k_gemm16_mt(
    WMat2 w, const unsigned short* __restrict__ bprep,
    float* __restrict__ y, unsigned short* __restrict__ xprep_out,
    float* __restrict__ ss_out, int T, int MT, int jtw) {
    int gtile, mtile;
    const int GT = (w.rows >> 4) / RT;
    xcd_decode(GT, MT, gtile, mtile);
    if (gtile >= GT) return;
    const int lane = threadIdx.x & (WAVE - 1);
    const int j = lane & 15;
    float acc[RT][1][JT][4];
    const WMat2* ws[1] = {&w};
    const int nbt = w.cols >> 5;
    const int nbk = (WT == W_F16) ? nbt : ((nbt + 3) & ~3);
    wave_tile_kloop<WT, false, 1, JT, RT>(ws, gtile, bprep, nullptr,
                                          nullptr, 0.f, acc, 0, nbk, jtw,
                                          mtile * JT);
    __shared__ float lds[3 * 64 * 4 * JT * RT];
    combine_acc<RT * JT>(reinterpret_cast<float(*)[4]>(acc), lds);
    if (threadIdx.x >= WAVE) return;
#pragma unroll
    for (int rt = 0; rt < RT; ++rt) {
    const int r0 = (gtile * RT + rt) * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int jt = 0; jt < JT; ++jt) {
        const int j2 = (mtile * JT + jt) * 16 + j;
        if (j2 >= T) continue;
        float sq = 0.0f;
#pragma unroll
        for (int jj = 0; jj < 4; ++jj) {
            float v = acc[rt][0][jt][jj];
            if (RES_SQ) {
                v += y[(size_t)j2 * w.rows + r0 + jj];
                sq += v * v;
            }
            y[(size_t)j2 * w.rows + r0 + jj] = v;
            acc[rt][0][jt][jj] = v;
        }
        if (RES_SQ && xprep_out != nullptr) {
            uint2 o;
            o.x = pack_f16(acc[rt][0][jt][0], acc[rt][0][jt][1]);
            o.y = pack_f16(acc[rt][0][jt][2], acc[rt][0][jt][3]);
            *reinterpret_cast<uint2*>(
                xprep_out +
                (((size_t)(r0 >> 3) * jtw + mtile * JT + jt) * 16 + j) * 8 +
                (r0 & 7)) = o;
        }
        if (RES_SQ && ss_out != nullptr) {
            float s2 = sq;
            s2 += __shfl_xor(s2, 16);
            s2 += __shfl_xor(s2, 32);
            if (lane < 16) atomicAdd(ss_out + j2, s2);
        }
    }
    }
}

// Large-M w1/w3 + fused RMSNorm + SwiGLU -> gprep (analog of k_ffn16).
template <int WT, int JT = 4, int RT = 1>
__global__ __launch_bounds__(BLOCK) void k_ffn16_mt(
    WMat2 w1, WMat2 w3, const unsigned short* __restrict__ xprep,
    const unsigned short* __restrict__ normprep,
    const float* __restrict__ ss_in, float eps,
    unsigned short* __restrict__ gprep, int T, int MT, int jtw) {
    int gtile, mtile;
    const int GT = (w1.rows >> 4) / RT;
    xcd_decode(GT, MT, gtile, mtile);
    if (gtile >= GT) return;
    const int lane = threadIdx.x & (WAVE - 1);
    const int j = lane & 15;
    float acc[RT][2][JT][4];
    const WMat2* ws[2] = {&w1, &w3};
    const int nbf = (WT == W_F16) ? (w1.cols >> 5)
                                  : (((w1.cols >> 5) + 3) & ~3);
    wave_tile_kloop<WT, true, 2, JT, RT>(ws, gtile, xprep, normprep, ss_in,
                                         eps, acc, 0, nbf, jtw,
                                         mtile * JT);
    __shared__ float lds[3 * 64 * 4 * 2 * JT * RT];
    combine_acc<RT * 2 * JT>(reinterpret_cast<float(*)[4]>(acc), lds);
    if (threadIdx.x >= WAVE) return;
#pragma unroll
    for (int rt = 0; rt < RT; ++rt) {
    const int r0 = (gtile * RT + rt) * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int jt = 0; jt < JT; ++jt) {
        const int j2 = (mtile * JT + jt) * 16 + j;
        if (j2 >= T) continue;
        float g[4];
#pragma unroll
        for (int jj = 0; jj < 4; ++jj) {
            const float v1 = acc[rt][0][jt][jj];
            const float silu = v1 / (1.0f + __expf(-v1));
            g[jj] = silu * acc[rt][1][jt][jj];
        }
        uint2 o;
        o.x = pack_f16(g[0], g[1]);
        o.y = pack_f16(g[2], g[3]);
        *reinterpret_cast<uint2*>(
            gprep + (((size_t)(r0 >> 3) * jtw + mtile * JT + jt) * 16 + j)
                        * 8 + (r0 & 7)) = o;
    }
    }
}

// ============================================================== launchers

// column-tile count for a token batch T (1, 2 or 4 tiles of 16)
static inline int pick_jt(int T) {
    if (T <= 16) return 1;
    if (T <= 32) return 2;
    return 4;
}

// xprep token-panel width in 16-token tiles: the decode path packs up to
// 4 tiles; the prefill path packs ceil(T/16) padded to whole 4-tile
// groups (JT=4 kernels read whole groups; pad tiles are dead columns)
int jt_width(int T) {
    if (T <= 64) return pick_jt(T);
    return (((T + 15) >> 4) + 3) & ~3;
}

static inline int pick_tmax(int T) {
    if (T <= 1) return 1;
    if (T <= 2) return 2;
    if (T <= 4) return 4;
    if (T <= 8) return 8;
    return 16;
}

void launch_rmsnorm(hipStream_t s, const float* x, const float* w, float* y,
                    int T, int E, float eps) {
    hipLaunchKernelGGL(k_rmsnorm, dim3(T), dim3(BLOCK), 0, s, x, w, y, E, eps);
}

#define DISPATCH_WT(WTV, ...)                    \
    switch (WTV) {                               \
        case W_Q4_0: {                           \
            constexpr int WTc = W_Q4_0;          \
            __VA_ARGS__;                         \
            break;                               \
        }                                        \
        case W_Q4_1: {                           \
            constexpr int WTc = W_Q4_1;          \
            __VA_ARGS__;                         \
            break;                               \
        }                                        \
        case W_F16: {                            \
            constexpr int WTc = W_F16;           \
            __VA_ARGS__;                         \
            break;                               \
        }                                        \
        default: {                               \
            constexpr int WTc = W_F32;           \
            __VA_ARGS__;                         \
            break;                               \
        }                                        \
    }

#define DISPATCH_TMAX(TMAXV, ...)        \
    switch (TMAXV) {                     \
        case 1: {                        \
            constexpr int TMc = 1;       \
            __VA_ARGS__;                 \
            break;                       \
        }                                \
        case 2: {                        \
            constexpr int TMc = 2;       \
            __VA_ARGS__;                 \
            break;                       \
        }                                \
        case 4: {                        \
            constexpr int TMc = 4;       \
            __VA_ARGS__;                 \
            break;                       \
        }                                \
        case 8: {                        \
            constexpr int TMc = 8;       \
            __VA_ARGS__;                 \
            break;                       \
        }                                \
        default: {                       \
            constexpr int TMc = 16;      \
            __VA_ARGS__;                 \
            break;                       \
        }                                \
    }

void launch_qkv_rope_append(hipStream_t s, const WMat& wq, const WMat& wk,
                            const WMat& wv, const float* xn, float* q_buf,
                            __half* k_cache_layer, __half* v_cache_layer,
                            const int* pos, const int* seq,
                            const float* inv_freq, int E, int D, int n_ctx,
                            int T) {
    const int tmax = pick_tmax(T);
    const dim3 grid(3 * E / 4);
    DISPATCH_WT(wq.wtype, DISPATCH_TMAX(
        tmax, hipLaunchKernelGGL((k_qkv_rope_append<WTc, TMc>), grid,
                                 dim3(BLOCK), 0, s, wq, wk, wv, xn, q_buf,
                                 k_cache_layer, v_cache_layer, pos, seq,
                                 inv_freq, E, D, n_ctx, T)));
}

void launch_attention(hipStream_t s, const float* q_buf,
                      const __half* k_cache_layer, __half* v_cache_layer,
                      float* out, unsigned short* out_prep, const int* pos,
                      const int* seq, int T, int H, int E, int Ekv, int D,
                      int n_ctx, const float* qkv_slab, int ks,
                      const float* inv_freq) {
    const dim3 grid(T, H);
    // lds_q[D] + prologue stage[BLOCK] + lds_m/lds_l[2*NWAVES] +
    // merge area [16][128]
    const size_t lds =
        (D + BLOCK + 2 * NWAVES + 16 * 128) * sizeof(float);
    if (qkv_slab != nullptr) {
        hipLaunchKernelGGL(k_attention<true>, grid, dim3(BLOCK), lds, s,
                           q_buf, k_cache_layer, v_cache_layer, out,
                           out_prep, pos, seq, E, Ekv, D, n_ctx,
                           jt_width(T), qkv_slab, ks, inv_freq);
        return;
    }
    hipLaunchKernelGGL(k_attention<false>, grid, dim3(BLOCK), lds, s, q_buf,
                       k_cache_layer, v_cache_layer, out, out_prep, pos, seq,
                       E, Ekv, D, n_ctx, jt_width(T), qkv_slab, ks,
                       inv_freq);
}

// ------------------------------------------------- MFMA-path launchers

void launch_prep_x(hipStream_t s, const float* x, unsigned short* xprep,
                   float* ss, int cols, int T) {
    // chunk columns so the launch spreads over ~256 CUs even at T=16
    // (a T-block launch measured 4.6 us on 16 CUs); ss zeroed upstream
    // ~128 blocks total: a microbenchmark of this kernel shape (see
    // profiles/microbench_small_kernel.hip) shows larger grids LOSE —
    // launch/ramp overhead outweighs the parallelism for ~1 MB of work
    const int want = (128 + T - 1) / max(T, 1);
    const int chunks = max(1, min(want, (cols >> 3) / 64));
    hipLaunchKernelGGL(k_prep_x, dim3(T, chunks), dim3(BLOCK), 0, s, x,
                       xprep, ss, cols, jt_width(T));
}

#define DISPATCH_JT(JTV, ...)            \
    switch (JTV) {                       \
        case 1: {                        \
            constexpr int JTc = 1;       \
            __VA_ARGS__;                 \
            break;                       \
        }                                \
        case 2: {                        \
            constexpr int JTc = 2;       \
            __VA_ARGS__;                 \
            break;                       \
        }                                \
        default: {                       \
            constexpr int JTc = 4;       \
            __VA_ARGS__;                 \
            break;                       \
        }                                \
    }

#define DISPATCH_WT2(WTV, ...)                   \
    switch (WTV) {                               \
        case W_Q4_0: {                           \
            constexpr int WTc = W_Q4_0;          \
            __VA_ARGS__;                         \
            break;                               \
        }                                        \
        case W_Q4_1: {                           \
            constexpr int WTc = W_Q4_1;          \
            __VA_ARGS__;                         \
            break;                               \
        }                                        \
        case W_Q8B: {                            \
            constexpr int WTc = W_Q8B;           \
            __VA_ARGS__;                         \
            break;                               \
        }                                        \
        case W_Q8B16: {                          \
            constexpr int WTc = W_Q8B16;         \
            __VA_ARGS__;                         \
            break;                               \
        }                                        \
        default: {                               \
            constexpr int WTc = W_F16;           \
            __VA_ARGS__;                         \
            break;                               \
        }                                        \
    }

int qkv16_ks(int E, int Ekv) {
    const int tiles3 = (E + 2 * Ekv) >> 4;
    int ks = 1;
    while ((tiles3 / 2) * ks < 512 && ks < 8) ks <<= 1;
    return ks;
}

int gemm16_ks(int rows) {
    const int R = rows / 16;
    int ks = 1;
    while (R * ks < 512 && ks < 8) ks <<= 1;
    // the slab path runs RT=2 (half the blocks) — keep at least ks=2 so
    // big-E models (65B: R=512) still land >=512 blocks
    if (ks < 2) ks = 2;
    return ks;
}

void launch_reduce_prep(hipStream_t s, float* y, const float* slab, int ks,
                        unsigned short* xprep, float* ss, int cols, int T) {
    const int want = (128 + T - 1) / max(T, 1);  // see microbench note
    const int chunks = max(1, min(want, (cols >> 3) / 64));
    hipLaunchKernelGGL(k_reduce_prep, dim3(T, chunks), dim3(BLOCK), 0, s, y,
                       slab, ks, xprep, ss, cols, pick_jt(T));
}

void launch_gemm16(hipStream_t s, const WMat2& w,
                   const unsigned short* bprep,
                   const unsigned short* normprep, const float* ss_in,
                   float eps, float* y, unsigned short* xprep_out,
                   float* ss_out, int T, int mode) {
    const int R = w.rows / 16;
    if (mode == GM_ATOMIC || mode == GM_SLAB) {
        // split K so R*KS lands near 2-3 blocks/CU (256 CUs)
        const int ks = gemm16_ks(w.rows);
        if (mode == GM_SLAB) {
            if (R % 2 == 0) {  // RT=2: halve the B-panel re-read
                const dim3 grid(R / 2, ks);
                DISPATCH_WT2(w.wtype, DISPATCH_JT(pick_jt(T),
                    hipLaunchKernelGGL(
                        (k_gemm16<WTc, GM_SLAB, JTc, 2>), grid, dim3(BLOCK),
                        0, s, w, bprep, normprep, ss_in, eps, y, xprep_out,
                        ss_out, T)));
            } else {
                const dim3 grid(R, ks);
                DISPATCH_WT2(w.wtype, DISPATCH_JT(pick_jt(T),
                    hipLaunchKernelGGL(
                        (k_gemm16<WTc, GM_SLAB, JTc>), grid, dim3(BLOCK), 0,
                        s, w, bprep, normprep, ss_in, eps, y, xprep_out,
                        ss_out, T)));
            }
        } else {
            const dim3 grid(R, ks);
            DISPATCH_WT2(w.wtype, DISPATCH_JT(pick_jt(T), hipLaunchKernelGGL(
                (k_gemm16<WTc, GM_ATOMIC, JTc>), grid, dim3(BLOCK), 0, s, w,
                bprep, normprep, ss_in, eps, y, xprep_out, ss_out, T)));
        }
        return;
    }
    const dim3 grid(R);
    DISPATCH_WT2(w.wtype, DISPATCH_JT(pick_jt(T), {
        if (mode == GM_RES_SQ)
            hipLaunchKernelGGL((k_gemm16<WTc, GM_RES_SQ, JTc>), grid,
                               dim3(BLOCK), 0, s, w, bprep, normprep, ss_in,
                               eps, y, xprep_out, ss_out, T);
        else if (mode == GM_NORM_PLAIN) {
            // lm_head: thousands of row tiles re-read the same B panel —
            // RT divides that L2 traffic while the grid still fills
            if (R % 4 == 0 && R / 4 >= 448)
                hipLaunchKernelGGL((k_gemm16<WTc, GM_NORM_PLAIN, JTc, 4>),
                                   dim3(R / 4), dim3(BLOCK), 0, s, w, bprep,
                                   normprep, ss_in, eps, y, xprep_out,
                                   ss_out, T);
            else if (R % 2 == 0 && R >= 1024)
                hipLaunchKernelGGL((k_gemm16<WTc, GM_NORM_PLAIN, JTc, 2>),
                                   dim3(R / 2), dim3(BLOCK), 0, s, w, bprep,
                                   normprep, ss_in, eps, y, xprep_out,
                                   ss_out, T);
            else
                hipLaunchKernelGGL((k_gemm16<WTc, GM_NORM_PLAIN, JTc>), grid,
                                   dim3(BLOCK), 0, s, w, bprep, normprep,
                                   ss_in, eps, y, xprep_out, ss_out, T);
        }
        else
            hipLaunchKernelGGL((k_gemm16<WTc, GM_PLAIN, JTc>), grid,
                               dim3(BLOCK), 0, s, w, bprep, normprep, ss_in,
                               eps, y, xprep_out, ss_out, T);
    }));
}

int launch_qkv16(hipStream_t s, const WMat2& wq, const WMat2& wk,
                 const WMat2& wv, const unsigned short* xprep,
                 const unsigned short* normprep, const float* ss_in,
                 float eps, float* q_buf, __half* k_cache_layer,
                 __half* v_cache_layer, const int* pos, const int* seq,
                 const float* inv_freq, int E, int Ekv, int D, int n_ctx,
                 int T, float* slab, int skip_finish) {
    const int tiles3 = (E + 2 * Ekv) >> 4;
    const bool rt2_ok = ((E >> 4) % 2 == 0) && ((Ekv >> 4) % 2 == 0);
    // slab split-K + RT=2 path when the fused grid underfills the chip
    if (slab != nullptr && tiles3 < 1024 && rt2_ok) {
        // RT=2 measured best (RT=4 loses ~3%: fill drops below 2/CU)
        const int ks = qkv16_ks(E, Ekv);
        const dim3 grid(tiles3 / 2, ks);
        DISPATCH_WT2(wq.wtype, DISPATCH_JT(pick_jt(T), hipLaunchKernelGGL(
            (k_qkv16_slab<WTc, JTc, 2>), grid, dim3(BLOCK), 0, s, wq, wk,
            wv, xprep, normprep, ss_in, eps, slab, E, Ekv, T)));
        if (!skip_finish) {
            const int total = ((E + 2 * Ekv) / 8) * T;
            hipLaunchKernelGGL(k_qkv_finish,
                               dim3((total + BLOCK - 1) / BLOCK),
                               dim3(BLOCK), 0, s, slab, ks, q_buf,
                               k_cache_layer, v_cache_layer, pos, seq,
                               inv_freq, E, Ekv, D, n_ctx, T);
        }
        return 1;
    }
    if (rt2_ok && tiles3 / 2 >= 512) {
        // big models: RT=2 fused — halves the B-panel re-read while the
        // halved grid still fills the chip
        const dim3 grid(tiles3 / 2);
        DISPATCH_WT2(wq.wtype, DISPATCH_JT(pick_jt(T), hipLaunchKernelGGL(
            (k_qkv16<WTc, JTc, 2>), grid, dim3(BLOCK), 0, s, wq, wk, wv,
            xprep, normprep, ss_in, eps, q_buf, k_cache_layer,
            v_cache_layer, pos, seq, inv_freq, E, Ekv, D, n_ctx, T)));
        return 0;
    }
    const dim3 grid(tiles3);
    DISPATCH_WT2(wq.wtype, DISPATCH_JT(pick_jt(T), hipLaunchKernelGGL(
        (k_qkv16<WTc, JTc>), grid, dim3(BLOCK), 0, s, wq, wk, wv, xprep,
        normprep, ss_in, eps, q_buf, k_cache_layer, v_cache_layer, pos, seq,
        inv_freq, E, Ekv, D, n_ctx, T)));
    return 0;
}

void launch_ffn16(hipStream_t s, const WMat2& w1, const WMat2& w3,
                  const unsigned short* xprep,
                  const unsigned short* normprep, const float* ss_in,
                  float eps, unsigned short* gprep, int T, float* slab) {
    const int tilesF = w1.rows / 16;
    if (slab != nullptr && tilesF < 1024 && (tilesF % 2) == 0) {
        int ks = 1;
        while ((tilesF / 2) * ks < 512 && ks < 8) ks <<= 1;
        const dim3 grid(tilesF / 2, ks);
        DISPATCH_WT2(w1.wtype, DISPATCH_JT(pick_jt(T), hipLaunchKernelGGL(
            (k_ffn16_slab<WTc, JTc, 2>), grid, dim3(BLOCK), 0, s, w1, w3,
            xprep, normprep, ss_in, eps, slab, T)));
        const int total = (w1.rows / 8) * T;
        hipLaunchKernelGGL(k_ffn_finish,
                           dim3((total + BLOCK - 1) / BLOCK), dim3(BLOCK),
                           0, s, slab, ks, gprep, w1.rows, T, pick_jt(T));
        return;
    }
    if ((tilesF % 2) == 0 && tilesF / 2 >= 512) {
        const dim3 grid(tilesF / 2);
        DISPATCH_WT2(w1.wtype, DISPATCH_JT(pick_jt(T), hipLaunchKernelGGL(
            (k_ffn16<WTc, JTc, 2>), grid, dim3(BLOCK), 0, s, w1, w3, xprep,
            normprep, ss_in, eps, gprep, T)));
        return;
    }

    const dim3 grid(tilesF);
    DISPATCH_WT2(w1.wtype, DISPATCH_JT(pick_jt(T), hipLaunchKernelGGL(
        (k_ffn16<WTc, JTc>), grid, dim3(BLOCK), 0, s, w1, w3, xprep,
        normprep, ss_in, eps, gprep, T)));
}

void launch_gemv(hipStream_t s, const WMat& w, const float* x,
                 const float* res, float* y, int T) {
    const int tmax = pick_tmax(T);
    const dim3 grid(w.rows / 4);
    if (res != nullptr) {
        DISPATCH_WT(w.wtype, DISPATCH_TMAX(
            tmax, hipLaunchKernelGGL((k_gemv<WTc, TMc, true>), grid,
                                     dim3(BLOCK), 0, s, w, x, res, y, T)));
    } else {
        DISPATCH_WT(w.wtype, DISPATCH_TMAX(
            tmax, hipLaunchKernelGGL((k_gemv<WTc, TMc, false>), grid,
                                     dim3(BLOCK), 0, s, w, x, res, y, T)));
    }
}

void launch_ffn_gate(hipStream_t s, const WMat& w1, const WMat& w3,
                     const float* xn, float* g, int T) {
    const int tmax = pick_tmax(T);
    const dim3 grid(w1.rows / 4);
    DISPATCH_WT(w1.wtype, DISPATCH_TMAX(
        tmax, hipLaunchKernelGGL((k_ffn_gate<WTc, TMc>), grid, dim3(BLOCK),
                                 0, s, w1, w3, xn, g, T)));
}

void launch_embed(hipStream_t s, const WMat& tab, const int* tokens,
                  float* out, int T, int E) {
    switch (tab.wtype) {
        case W_Q4_0:
            hipLaunchKernelGGL((k_embed<W_Q4_0>), dim3(T), dim3(BLOCK), 0, s,
                               tab, tokens, out, E);
            break;
        case W_Q4_1:
            hipLaunchKernelGGL((k_embed<W_Q4_1>), dim3(T), dim3(BLOCK), 0, s,
                               tab, tokens, out, E);
            break;
        case W_F16:
            hipLaunchKernelGGL((k_embed<W_F16>), dim3(T), dim3(BLOCK), 0, s,
                               tab, tokens, out, E);
            break;
        default:
            hipLaunchKernelGGL((k_embed<W_F32>), dim3(T), dim3(BLOCK), 0, s,
                               tab, tokens, out, E);
            break;
    }
}

// ------------------------------------------------- prefill-path launchers

void launch_attn_prefill(hipStream_t s, const float* q_buf,
                         const __half* k_cache_layer,
                         const __half* v_cache_layer, float* out,
                         unsigned short* out_prep, const int* pos,
                         const int* seq, int T, int H, int E, int Ekv,
                         int D, int n_ctx) {
    const dim3 grid((T + 15) / 16, H);
    if (D <= 128) {  // every LLaMA head dim; matrix-core path
        hipLaunchKernelGGL(k_attn_prefill_mfma, grid, dim3(BLOCK), 0, s,
                           q_buf, k_cache_layer, v_cache_layer, out,
                           out_prep, pos, seq, E, Ekv, D, n_ctx,
                           jt_width(T), T);
        return;
    }
    const size_t lds = (size_t)(16 * D + 16 * BLOCK + 32) * sizeof(float);
    hipLaunchKernelGGL(k_attn_prefill, grid, dim3(BLOCK), lds, s, q_buf,
                       k_cache_layer, v_cache_layer, out, out_prep, pos,
                       seq, E, Ekv, D, n_ctx, jt_width(T), T);
}

void launch_qkv16_mt(hipStream_t s, const WMat2& wq, const WMat2& wk,
                     const WMat2& wv, const unsigned short* xprep,
                     const unsigned short* normprep, const float* ss_in,
                     float eps, float* q_buf, __half* k_cache_layer,
                     __half* v_cache_layer, const int* pos, const int* seq,
                     const float* inv_freq, int E, int Ekv, int D,
                     int n_ctx, int T) {
    const int MT = (T + 63) >> 6;
    const int jtw = jt_width(T);
    const int tiles3 = (E + 2 * Ekv) >> 4;
    const bool rt2_ok = ((E >> 4) % 2 == 0) && ((Ekv >> 4) % 2 == 0);
    // RT=2 halves the B-panel reads + norm-build VALU per MFMA as long
    // as the halved grid still fills the chip
    if (rt2_ok && (tiles3 / 2) * MT >= 512) {
        const dim3 grid(xcd_grid(tiles3 / 2, MT));
        DISPATCH_WT2(wq.wtype, hipLaunchKernelGGL(
            (k_qkv16_mt<WTc, 4, 2>), grid, dim3(BLOCK), 0, s, wq, wk, wv,
            xprep, normprep, ss_in, eps, q_buf, k_cache_layer,
            v_cache_layer, pos, seq, inv_freq, E, Ekv, D, n_ctx, T, MT,
            jtw));
        return;
    }
    const dim3 grid(xcd_grid(tiles3, MT));
    DISPATCH_WT2(wq.wtype, hipLaunchKernelGGL(
        (k_qkv16_mt<WTc>), grid, dim3(BLOCK), 0, s, wq, wk, wv, xprep,
        normprep, ss_in, eps, q_buf, k_cache_layer, v_cache_layer, pos,
        seq, inv_freq, E, Ekv, D, n_ctx, T, MT, jtw));
}

void launch_gemm16_mt(hipStream_t s, const WMat2& w,
                      const unsigned short* bprep, float* y,
                      unsigned short* xprep_out, float* ss_out, int T,
                      int res_sq) {
    const int MT = (T + 63) >> 6;
    const int jtw = jt_width(T);
    const int R = w.rows >> 4;
    if (R % 2 == 0 && (R / 2) * MT >= 512) {
        const dim3 grid(xcd_grid(R / 2, MT));
        if (res_sq) {
            DISPATCH_WT2(w.wtype, hipLaunchKernelGGL(
                (k_gemm16_mt<WTc, true, 4, 2>), grid, dim3(BLOCK), 0, s, w,
                bprep, y, xprep_out, ss_out, T, MT, jtw));
        } else {
            DISPATCH_WT2(w.wtype, hipLaunchKernelGGL(
                (k_gemm16_mt<WTc, false, 4, 2>), grid, dim3(BLOCK), 0, s,
                w, bprep, y, xprep_out, ss_out, T, MT, jtw));
        }
        return;
    }
    const dim3 grid(xcd_grid(R, MT));
    if (res_sq) {
        DISPATCH_WT2(w.wtype, hipLaunchKernelGGL(
            (k_gemm16_mt<WTc, true>), grid, dim3(BLOCK), 0, s, w, bprep, y,
            xprep_out, ss_out, T, MT, jtw));
    } else {
        DISPATCH_WT2(w.wtype, hipLaunchKernelGGL(
            (k_gemm16_mt<WTc, false>), grid, dim3(BLOCK), 0, s, w, bprep, y,
            xprep_out, ss_out, T, MT, jtw));
    }
}

void launch_ffn16_mt(hipStream_t s, const WMat2& w1, const WMat2& w3,
                     const unsigned short* xprep,
                     const unsigned short* normprep, const float* ss_in,
                     float eps, unsigned short* gprep, int T) {
    const int MT = (T + 63) >> 6;
    const int jtw = jt_width(T);
    const int R = w1.rows >> 4;
    if (R % 2 == 0 && (R / 2) * MT >= 512) {
        const dim3 grid(xcd_grid(R / 2, MT));
        DISPATCH_WT2(w1.wtype, hipLaunchKernelGGL(
            (k_ffn16_mt<WTc, 4, 2>), grid, dim3(BLOCK), 0, s, w1, w3,
            xprep, normprep, ss_in, eps, gprep, T, MT, jtw));
        return;
    }
    const dim3 grid(xcd_grid(R, MT));
    DISPATCH_WT2(w1.wtype, hipLaunchKernelGGL(
        (k_ffn16_mt<WTc>), grid, dim3(BLOCK), 0, s, w1, w3, xprep, normprep,
        ss_in, eps, gprep, T, MT, jtw));
}

void launch_argmax(hipStream_t s, const float* logits,
                   unsigned long long* keys, int* out, int T, int V) {
    (void)hipMemsetAsync(keys, 0, sizeof(unsigned long long) * T, s);
    const int npart = min(32, (V + BLOCK - 1) / BLOCK);
    hipLaunchKernelGGL(k_argmax_part, dim3(T, npart), dim3(BLOCK), 0, s,
                       logits, keys, V);
    hipLaunchKernelGGL(k_argmax_finish, dim3((T + 63) / 64), dim3(64), 0,
                       s, keys, out, T);
}
