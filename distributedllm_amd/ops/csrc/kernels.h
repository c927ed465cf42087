// Launch API of the CDNA4 kernel library (kernels.hip).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <stdint.h>

// W_Q8B is the BYTE-stream quant path: q8_0 weights re-biased to u8 at
// repack, and q5_0/q5_1/q4_K/q5_K expanded to re-biased bytes — all
// share one kernel form (w = alpha*((1024+u) - 1152) + beta in packed
// f16, (alpha, beta) per 32-weight block). W_Q8B16 is the same byte
// stream with per-16-weight (alpha, beta) planes — the k-quant formats
// whose sub-block scales have 16-weight granularity (q2_K/q3_K/q6_K).
enum WType { W_F32 = 0, W_F16 = 1, W_Q4_0 = 2, W_Q4_1 = 3, W_Q8B = 8,
             W_Q8B16 = 9 };

// One weight matrix resident in HBM, repacked SoA (see kernels.hip header).
struct WMat {
    const void* data;    // f16/f32 values, or u8 nibbles for q4_*
    const void* scales;  // q4_0: f16[rows][nb]; q4_1: f16[rows][nb*2] (d,m)
    int rows;
    int cols;
    int wtype;  // WType
};

// MFMA-tiled weight matrix (the production decode path). Layouts, with
// R = rows/16 row-tiles and nb = cols/32 q4 blocks:
//   q4_0/q4_1: data   = u32[R][nb][4][16]   (nibble words, see kernels.hip)
//              scales = f32[R][nb][16]      (q4_1: float2 (d, m))
//   W_F16    : data   = bf16[R][cols/8][16][8]  (weights converted to bf16)
struct WMat2 {
    const void* data;
    const void* scales;
    int rows;
    int cols;
    int wtype;  // WType (W_F32 never appears here — legacy path)
};

// k_gemm16 fused-epilogue modes. GM_ATOMIC = grid-level split-K partials
// atomicAdd'ed into y (residual pre-loaded); follow with launch_prep_x.
enum GemmMode { GM_PLAIN = 0, GM_RES_SQ = 1, GM_NORM_PLAIN = 2,
                GM_ATOMIC = 3, GM_SLAB = 4 };

// split-K factor used by GM_ATOMIC/GM_SLAB (shared with the reducer)
int gemm16_ks(int rows);

// y[t] += sum of ks slab partials, then sumsq + f16 xprep of the result
void launch_reduce_prep(hipStream_t s, float* y, const float* slab, int ks,
                        unsigned short* xprep, float* ss, int cols, int T);

void launch_prep_x(hipStream_t s, const float* x, unsigned short* xprep,
                   float* ss, int cols, int T);

void launch_gemm16(hipStream_t s, const WMat2& w,
                   const unsigned short* bprep,
                   const unsigned short* normprep, const float* ss_in,
                   float eps, float* y, unsigned short* xprep_out,
                   float* ss_out, int T, int mode);

// slab != nullptr enables the RT=2 + split-K slab path for small models;
// returns 1 when the slab path ran. skip_finish=1 leaves the slabs un-
// combined for a fused-attention consumer (decode only).
int launch_qkv16(hipStream_t s, const WMat2& wq, const WMat2& wk,
                 const WMat2& wv, const unsigned short* xprep,
                 const unsigned short* normprep, const float* ss_in,
                 float eps, float* q_buf, __half* k_cache_layer,
                 __half* v_cache_layer, const int* pos, const int* seq,
                 const float* inv_freq, int E, int Ekv, int D, int n_ctx,
                 int T, float* slab, int skip_finish);

// the qkv slab split factor (shared with the fused-attention consumer)
int qkv16_ks(int E, int Ekv);

void launch_ffn16(hipStream_t s, const WMat2& w1, const WMat2& w3,
                  const unsigned short* xprep,
                  const unsigned short* normprep, const float* ss_in,
                  float eps, unsigned short* gprep, int T, float* slab);

void launch_rmsnorm(hipStream_t s, const float* x, const float* w, float* y,
                    int T, int E, float eps);

void launch_qkv_rope_append(hipStream_t s, const WMat& wq, const WMat& wk,
                            const WMat& wv, const float* xn, float* q_buf,
                            __half* k_cache_layer, __half* v_cache_layer,
                            const int* pos, const int* seq,
                            const float* inv_freq, int E, int D, int n_ctx,
                            int T);

// qkv_slab != nullptr: decode-fused variant — the attention prologue
// sums the un-combined qkv slabs, applies RoPE and appends the KV row
// (valid only when every token is its own sequence)
void launch_attention(hipStream_t s, const float* q_buf,
                      const __half* k_cache_layer, __half* v_cache_layer,
                      float* out, unsigned short* out_prep, const int* pos,
                      const int* seq, int T, int H, int E, int Ekv, int D,
                      int n_ctx, const float* qkv_slab, int ks,
                      const float* inv_freq);

void launch_gemv(hipStream_t s, const WMat& w, const float* x,
                 const float* res, float* y, int T);

void launch_ffn_gate(hipStream_t s, const WMat& w1, const WMat& w3,
                     const float* xn, float* g, int T);

void launch_embed(hipStream_t s, const WMat& tab, const int* tokens,
                  float* out, int T, int E);

void launch_argmax(hipStream_t s, const float* logits,
                   unsigned long long* keys, int* out, int T, int V);

// ---------------------------------------------------------- prefill path
// Large-M (T > 64) layer kernels: one launch covers all T tokens with an
// XCD-aware (row-tile x 64-token-group) grid. Side channels use the
// jt_width(T) token-panel stride.
int jt_width(int T);

// Q-tiled (16 queries/block) streaming attention for prefill: each K/V
// row is read once per 16 queries; mixed-seq tiles segment internally
void launch_attn_prefill(hipStream_t s, const float* q_buf,
                         const __half* k_cache_layer,
                         const __half* v_cache_layer, float* out,
                         unsigned short* out_prep, const int* pos,
                         const int* seq, int T, int H, int E, int Ekv,
                         int D, int n_ctx);

void launch_qkv16_mt(hipStream_t s, const WMat2& wq, const WMat2& wk,
                     const WMat2& wv, const unsigned short* xprep,
                     const unsigned short* normprep, const float* ss_in,
                     float eps, float* q_buf, __half* k_cache_layer,
                     __half* v_cache_layer, const int* pos, const int* seq,
                     const float* inv_freq, int E, int Ekv, int D,
                     int n_ctx, int T);

void launch_gemm16_mt(hipStream_t s, const WMat2& w,
                      const unsigned short* bprep, float* y,
                      unsigned short* xprep_out, float* ss_out, int T,
                      int res_sq);

void launch_ffn16_mt(hipStream_t s, const WMat2& w1, const WMat2& w3,
                     const unsigned short* xprep,
                     const unsigned short* normprep, const float* ss_in,
                     float eps, unsigned short* gprep, int T);
