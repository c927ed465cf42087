// kquants.hpp — k-quant super-block codecs (q2_K .. q6_K, QK_K=256)
// for the native tools. Byte-identical to the Python codecs in
// formats/kquants.py (same float32 math, nearbyintf == np.rint RNE,
// f32_to_f16 RNE); round-trip parity is asserted by
// tests/test_native_tools.py. Block layouts documented in
// formats/kquants.py:1-24 (clean-room per the reference loader's
// recognized ftypes, /root/reference/distllm/tensor_processor.cpp:846).
#pragma once

#include <algorithm>
#include <cfloat>
#include <cmath>
#include <cstdint>

#include "ggmlio.hpp"

namespace kq {

constexpr int kQK_K = 256;
constexpr int kQ2KBytes = 84;
constexpr int kQ3KBytes = 110;
constexpr int kQ4KBytes = 144;
constexpr int kQ5KBytes = 176;
constexpr int kQ6KBytes = 210;

// float32 round trip through f16 (numpy _f16(x).astype(float32))
inline float f16rt(float x) {
    return ggmlio::f16_to_f32(ggmlio::f32_to_f16(x));
}

// numpy _safe_inv: 1/d where d >= float32 tiny (smallest normal), else 0
inline float safe_inv(float d) { return d >= FLT_MIN ? 1.0f / d : 0.0f; }

inline float clipf(float v, float lo, float hi) {
    return v < lo ? lo : (v > hi ? hi : v);
}

// np.maximum semantics: ties (incl. +0 vs -0) return the SECOND arg —
// matters only for the sign of a zero scale in all-zero groups, but the
// codecs are byte-parity-tested against the numpy reference
inline float npmax(float a, float b) { return a > b ? a : b; }

inline void put_f16(uint8_t* p, float v) {
    const uint16_t h = ggmlio::f32_to_f16(v);
    p[0] = (uint8_t)(h & 0xFF);
    p[1] = (uint8_t)(h >> 8);
}

inline float get_f16(const uint8_t* p) {
    return ggmlio::f16_to_f32((uint16_t)(p[0] | (p[1] << 8)));
}

// per-group affine params (formats/kquants.py _affine_group_params):
// w ~= scale*q - negmin, q in [0, maxq]
inline void affine_params(const float* g, int n, int maxq,
                          float* scale, float* negmin) {
    float gmax = g[0], gmin = g[0];
    for (int i = 1; i < n; ++i) {
        gmax = std::max(gmax, g[i]);
        gmin = std::min(gmin, g[i]);
    }
    gmin = std::min(gmin, 0.0f);
    gmax = std::max(gmax, gmin);
    *scale = (gmax - gmin) / (float)maxq;
    *negmin = -gmin;
}

// ------------------------------------------------------------ q4_K/q5_K
// 6-bit (scale, min) pairs for 8 groups packed in 12 bytes
// (get_scale_min_k4 layout — formats/kquants.py:60).

inline void pack_scales_k4(const uint8_t* sc, const uint8_t* mn,
                           uint8_t* out) {
    for (int i = 0; i < 4; ++i) {
        out[i] = (uint8_t)((sc[i] & 63) | ((sc[i + 4] >> 4) << 6));
        out[4 + i] = (uint8_t)((mn[i] & 63) | ((mn[i + 4] >> 4) << 6));
        out[8 + i] = (uint8_t)((sc[i + 4] & 0xF) | ((mn[i + 4] & 0xF) << 4));
    }
}

inline void unpack_scales_k4(const uint8_t* p, uint8_t* sc, uint8_t* mn) {
    for (int i = 0; i < 4; ++i) {
        sc[i] = (uint8_t)(p[i] & 63);
        mn[i] = (uint8_t)(p[4 + i] & 63);
        sc[i + 4] = (uint8_t)((p[8 + i] & 0xF) | ((p[i] >> 6) << 4));
        mn[i + 4] = (uint8_t)((p[8 + i] >> 4) | ((p[4 + i] >> 6) << 4));
    }
}

// shared q4_K/q5_K super-scale + per-group q computation
inline void k4_groups(const float* x, int maxq, float* d_out, float* dm_out,
                      uint8_t sc[8], uint8_t mn[8], uint8_t q[8][32]) {
    float scale[8], negmin[8];
    for (int g = 0; g < 8; ++g)
        affine_params(x + g * 32, 32, maxq, &scale[g], &negmin[g]);
    float smax = scale[0], nmax = negmin[0];
    for (int g = 1; g < 8; ++g) {
        smax = std::max(smax, scale[g]);
        nmax = std::max(nmax, negmin[g]);
    }
    const float d = f16rt(smax / 63.0f);
    const float dmin = f16rt(nmax / 63.0f);
    const float id = safe_inv(d), idm = safe_inv(dmin);
    for (int g = 0; g < 8; ++g) {
        sc[g] = (uint8_t)clipf(nearbyintf(scale[g] * id), 0.0f, 63.0f);
        mn[g] = (uint8_t)clipf(nearbyintf(negmin[g] * idm), 0.0f, 63.0f);
        const float dg = d * (float)sc[g];
        const float mg = dmin * (float)mn[g];
        const float idg = safe_inv(dg);
        for (int i = 0; i < 32; ++i)
            q[g][i] = (uint8_t)clipf(nearbyintf((x[g * 32 + i] + mg) * idg),
                                     0.0f, (float)maxq);
    }
    *d_out = d;
    *dm_out = dmin;
}

inline void quantize_block_q4_K(const float* x, uint8_t* out) {
    float d, dmin;
    uint8_t sc[8], mn[8], q[8][32];
    k4_groups(x, 15, &d, &dmin, sc, mn, q);
    put_f16(out, d);
    put_f16(out + 2, dmin);
    pack_scales_k4(sc, mn, out + 4);
    for (int p = 0; p < 4; ++p)
        for (int l = 0; l < 32; ++l)
            out[16 + p * 32 + l] =
                (uint8_t)(q[2 * p][l] | (q[2 * p + 1][l] << 4));
}

inline void dequantize_block_q4_K(const uint8_t* in, float* x) {
    const float d = get_f16(in), dmin = get_f16(in + 2);
    uint8_t sc[8], mn[8];
    unpack_scales_k4(in + 4, sc, mn);
    for (int p = 0; p < 4; ++p) {
        const float dg0 = d * (float)sc[2 * p], mg0 = dmin * (float)mn[2 * p];
        const float dg1 = d * (float)sc[2 * p + 1];
        const float mg1 = dmin * (float)mn[2 * p + 1];
        for (int l = 0; l < 32; ++l) {
            const uint8_t b = in[16 + p * 32 + l];
            x[(2 * p) * 32 + l] = dg0 * (float)(b & 0xF) - mg0;
            x[(2 * p + 1) * 32 + l] = dg1 * (float)(b >> 4) - mg1;
        }
    }
}

inline void quantize_block_q5_K(const float* x, uint8_t* out) {
    float d, dmin;
    uint8_t sc[8], mn[8], q[8][32];
    k4_groups(x, 31, &d, &dmin, sc, mn, q);
    put_f16(out, d);
    put_f16(out + 2, dmin);
    pack_scales_k4(sc, mn, out + 4);
    for (int l = 0; l < 32; ++l) {
        uint8_t h = 0;
        for (int j = 0; j < 4; ++j)
            h |= (uint8_t)((((q[2 * j][l] >> 4) & 1) << (2 * j)) |
                           (((q[2 * j + 1][l] >> 4) & 1) << (2 * j + 1)));
        out[16 + l] = h;
    }
    for (int p = 0; p < 4; ++p)
        for (int l = 0; l < 32; ++l)
            out[48 + p * 32 + l] = (uint8_t)((q[2 * p][l] & 0xF) |
                                             ((q[2 * p + 1][l] & 0xF) << 4));
}

inline void dequantize_block_q5_K(const uint8_t* in, float* x) {
    const float d = get_f16(in), dmin = get_f16(in + 2);
    uint8_t sc[8], mn[8];
    unpack_scales_k4(in + 4, sc, mn);
    for (int j = 0; j < 4; ++j) {
        const float dg0 = d * (float)sc[2 * j], mg0 = dmin * (float)mn[2 * j];
        const float dg1 = d * (float)sc[2 * j + 1];
        const float mg1 = dmin * (float)mn[2 * j + 1];
        for (int l = 0; l < 32; ++l) {
            const uint8_t qh = in[16 + l], ql = in[48 + j * 32 + l];
            const float q0 = (float)((ql & 0xF) |
                                     (((qh >> (2 * j)) & 1) << 4));
            const float q1 = (float)((ql >> 4) |
                                     (((qh >> (2 * j + 1)) & 1) << 4));
            x[(2 * j) * 32 + l] = dg0 * q0 - mg0;
            x[(2 * j + 1) * 32 + l] = dg1 * q1 - mg1;
        }
    }
}

// ----------------------------------------------------------------- q6_K

inline void quantize_block_q6_K(const float* x, uint8_t* out) {
    float gscale[16];
    for (int g = 0; g < 16; ++g) {
        const float* b = x + g * 16;
        float gmax = b[0], gmin = b[0];
        for (int i = 1; i < 16; ++i) {
            gmax = std::max(gmax, b[i]);
            gmin = std::min(gmin, b[i]);
        }
        // asymmetric range: q-32 in [-32, 31], neither side clips
        gscale[g] = npmax(gmax / 31.0f, gmin / -32.0f);
    }
    float smax = gscale[0];
    for (int g = 1; g < 16; ++g) smax = npmax(smax, gscale[g]);
    const float d = f16rt(smax / 127.0f);
    const float id = safe_inv(d);
    int8_t sc[16];
    uint8_t q[256];
    for (int g = 0; g < 16; ++g) {
        sc[g] = (int8_t)clipf(nearbyintf(gscale[g] * id), -128.0f, 127.0f);
        const float dg = d * (float)sc[g];
        const float sgn = dg > 0.0f ? 1.0f : (dg < 0.0f ? -1.0f : 0.0f);
        const float idg = safe_inv(std::fabs(dg)) * sgn;
        for (int i = 0; i < 16; ++i)
            q[g * 16 + i] = (uint8_t)(clipf(nearbyintf(x[g * 16 + i] * idg),
                                            -32.0f, 31.0f) + 32.0f);
    }
    for (int h = 0; h < 2; ++h) {
        const uint8_t* w = q + h * 128;
        for (int l = 0; l < 32; ++l) {
            out[h * 64 + l] = (uint8_t)((w[l] & 0xF) | ((w[l + 64] & 0xF) << 4));
            out[h * 64 + 32 + l] =
                (uint8_t)((w[l + 32] & 0xF) | ((w[l + 96] & 0xF) << 4));
            out[128 + h * 32 + l] =
                (uint8_t)((w[l] >> 4) | ((w[l + 32] >> 4) << 2) |
                          ((w[l + 64] >> 4) << 4) | ((w[l + 96] >> 4) << 6));
        }
    }
    for (int g = 0; g < 16; ++g) out[192 + g] = (uint8_t)sc[g];
    put_f16(out + 208, d);
}

inline void dequantize_block_q6_K(const uint8_t* in, float* x) {
    const float d = get_f16(in + 208);
    for (int h = 0; h < 2; ++h) {
        const uint8_t* ql = in + h * 64;
        const uint8_t* qh = in + 128 + h * 32;
        for (int k = 0; k < 4; ++k) {
            for (int l = 0; l < 32; ++l) {
                const uint8_t lo = (k & 1) ? ql[32 + l] : ql[l];
                const int nib = (k < 2) ? (lo & 0xF) : (lo >> 4);
                const int hi = (qh[l] >> (2 * k)) & 3;
                const float q = (float)(nib | (hi << 4)) - 32.0f;
                const int g = h * 8 + 2 * k + l / 16;
                x[h * 128 + k * 32 + l] =
                    d * (float)(int8_t)in[192 + g] * q;
            }
        }
    }
}

// ----------------------------------------------------------------- q2_K

inline void quantize_block_q2_K(const float* x, uint8_t* out) {
    float scale[16], negmin[16];
    for (int g = 0; g < 16; ++g)
        affine_params(x + g * 16, 16, 3, &scale[g], &negmin[g]);
    float smax = scale[0], nmax = negmin[0];
    for (int g = 1; g < 16; ++g) {
        smax = std::max(smax, scale[g]);
        nmax = std::max(nmax, negmin[g]);
    }
    const float d = f16rt(smax / 15.0f);
    const float dmin = f16rt(nmax / 15.0f);
    const float id = safe_inv(d), idm = safe_inv(dmin);
    uint8_t q[256];
    for (int g = 0; g < 16; ++g) {
        const uint8_t sc =
            (uint8_t)clipf(nearbyintf(scale[g] * id), 0.0f, 15.0f);
        const uint8_t mn =
            (uint8_t)clipf(nearbyintf(negmin[g] * idm), 0.0f, 15.0f);
        out[g] = (uint8_t)(sc | (mn << 4));
        const float dg = d * (float)sc, mg = dmin * (float)mn;
        const float idg = safe_inv(dg);
        for (int i = 0; i < 16; ++i)
            q[g * 16 + i] = (uint8_t)clipf(
                nearbyintf((x[g * 16 + i] + mg) * idg), 0.0f, 3.0f);
    }
    for (int h = 0; h < 2; ++h)
        for (int l = 0; l < 32; ++l) {
            uint8_t b = 0;
            for (int j = 0; j < 4; ++j)
                b |= (uint8_t)(q[h * 128 + j * 32 + l] << (2 * j));
            out[16 + h * 32 + l] = b;
        }
    put_f16(out + 80, d);
    put_f16(out + 82, dmin);
}

inline void dequantize_block_q2_K(const uint8_t* in, float* x) {
    const float d = get_f16(in + 80), dmin = get_f16(in + 82);
    for (int h = 0; h < 2; ++h)
        for (int j = 0; j < 4; ++j) {
            for (int l = 0; l < 32; ++l) {
                const int g = h * 8 + j * 2 + l / 16;
                const float sc = (float)(in[g] & 0xF);
                const float mn = (float)(in[g] >> 4);
                const float q =
                    (float)((in[16 + h * 32 + l] >> (2 * j)) & 3);
                x[h * 128 + j * 32 + l] = d * sc * q - dmin * mn;
            }
        }
}

// ----------------------------------------------------------------- q3_K

inline void pack_scales_q3(const int8_t* sc, uint8_t* out) {
    // two-plane packing (formats/kquants.py _pack_scales_q3): value
    // v[w][i] = sc[4w+i]+32; a-planes carry the low nibbles, t the high
    // 2 bits
    for (int i = 0; i < 4; ++i) {
        const uint8_t v0 = (uint8_t)(sc[i] + 32);
        const uint8_t v1 = (uint8_t)(sc[4 + i] + 32);
        const uint8_t v2 = (uint8_t)(sc[8 + i] + 32);
        const uint8_t v3 = (uint8_t)(sc[12 + i] + 32);
        out[i] = (uint8_t)((v0 & 0xF) | ((v2 & 0xF) << 4));
        out[4 + i] = (uint8_t)((v1 & 0xF) | ((v3 & 0xF) << 4));
        out[8 + i] = (uint8_t)((v0 >> 4) | ((v1 >> 4) << 2) |
                               ((v2 >> 4) << 4) | ((v3 >> 4) << 6));
    }
}

inline void unpack_scales_q3(const uint8_t* p, int8_t* sc) {
    for (int i = 0; i < 4; ++i) {
        const uint8_t a0 = p[i], a1 = p[4 + i], t = p[8 + i];
        sc[i] = (int8_t)(((a0 & 0xF) | (((t >> 0) & 3) << 4)) - 32);
        sc[4 + i] = (int8_t)(((a1 & 0xF) | (((t >> 2) & 3) << 4)) - 32);
        sc[8 + i] = (int8_t)(((a0 >> 4) | (((t >> 4) & 3) << 4)) - 32);
        sc[12 + i] = (int8_t)(((a1 >> 4) | (((t >> 6) & 3) << 4)) - 32);
    }
}

inline void quantize_block_q3_K(const float* x, uint8_t* out) {
    float gscale[16];
    for (int g = 0; g < 16; ++g) {
        const float* b = x + g * 16;
        float gmax = b[0], gmin = b[0];
        for (int i = 1; i < 16; ++i) {
            gmax = std::max(gmax, b[i]);
            gmin = std::min(gmin, b[i]);
        }
        // asymmetric range: q-4 in [-4, 3]
        gscale[g] = npmax(gmax / 3.0f, gmin / -4.0f);
    }
    float smax = gscale[0];
    for (int g = 1; g < 16; ++g) smax = npmax(smax, gscale[g]);
    const float d = f16rt(smax / 31.0f);
    const float id = safe_inv(d);
    int8_t sc[16];
    uint8_t q[256];
    for (int g = 0; g < 16; ++g) {
        sc[g] = (int8_t)clipf(nearbyintf(gscale[g] * id), -32.0f, 31.0f);
        const float dg = d * (float)sc[g];
        const float sgn = dg > 0.0f ? 1.0f : (dg < 0.0f ? -1.0f : 0.0f);
        const float idg = safe_inv(std::fabs(dg)) * sgn;
        for (int i = 0; i < 16; ++i)
            q[g * 16 + i] = (uint8_t)(clipf(nearbyintf(x[g * 16 + i] * idg),
                                            -4.0f, 3.0f) + 4.0f);
    }
    for (int i = 0; i < 32; ++i) out[i] = 0;          // hmask
    for (int i = 0; i < 64; ++i) out[32 + i] = 0;     // qs
    for (int h = 0; h < 2; ++h)
        for (int j = 0; j < 4; ++j)
            for (int l = 0; l < 32; ++l) {
                const uint8_t w = q[h * 128 + j * 32 + l];
                out[32 + h * 32 + l] |= (uint8_t)((w & 3) << (2 * j));
                out[l] |= (uint8_t)(((w >> 2) & 1) << (h * 4 + j));
            }
    pack_scales_q3(sc, out + 96);
    put_f16(out + 108, d);
}

inline void dequantize_block_q3_K(const uint8_t* in, float* x) {
    const float d = get_f16(in + 108);
    int8_t sc[16];
    unpack_scales_q3(in + 96, sc);
    for (int h = 0; h < 2; ++h)
        for (int j = 0; j < 4; ++j)
            for (int l = 0; l < 32; ++l) {
                const int low = (in[32 + h * 32 + l] >> (2 * j)) & 3;
                const int hi = (in[l] >> (h * 4 + j)) & 1;
                const float q = (float)(low - (hi ? 0 : 4));
                const int g = h * 8 + 2 * j + l / 16;
                x[h * 128 + j * 32 + l] = d * (float)sc[g] * q;
            }
}

}  // namespace kq
