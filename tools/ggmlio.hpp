// GGJT-v3 model / slice file IO + q4 block codecs (C++17, header-only).
//
// Native counterpart of formats/ggml.py and formats/q4.py — the same
// clean-room byte layout:
//   * model files: 'ggjt' magic + version 3, 7-field hparams (n_vocab,
//     n_embd, n_mult, n_head, n_layer, n_rot, ftype), vocab
//     (len:u32 ‖ bytes ‖ score:f32), 32-byte-aligned tensor records
//     (reference reads this header at slice_model.cpp:167-175);
//   * slice/extra-layers files: extended 8-field header with first_layer
//     inserted between n_rot and ftype (written by the reference at
//     slice_model.cpp:253-263, read at tensor_processor.cpp:179-188);
//   * tensor record: n_dims:u32 ‖ name_len:u32 ‖ type:u32 ‖ ne[n_dims]:u32
//     ‖ name ‖ pad-to-32B ‖ data; ne[0] is the contiguous dimension.
//
// Used by the native CLIs tools/slice_model.cpp and tools/quantize.cpp
// (the reference vendors these as slice_model.cpp / the llama.cpp
// `quantize` binary — SURVEY §2.2 N2/N4). Byte-compatibility with the
// Python implementation is asserted by tests/test_native_tools.py.
#pragma once

#include <algorithm>
#include <cfloat>
#include <cmath>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

namespace ggmlio {

constexpr uint32_t kMagic = 0x67676A74;  // "ggjt" little-endian
constexpr uint32_t kVersion = 3;
constexpr uint32_t kVersionGqa = 4;  // framework GQA extension
constexpr uint32_t kExtraLayersFirstLayer = 0xFFFFFFFFu;

enum GType : uint32_t { F32 = 0, F16 = 1, Q4_0 = 2, Q4_1 = 3,
                        Q5_0 = 6, Q5_1 = 7, Q8_0 = 8,
                        // k-quant super-block types (QK_K = 256); the
                        // tools pass their bytes through (codec:
                        // formats/kquants.py)
                        Q2_K = 10, Q3_K = 11, Q4_K = 12, Q5_K = 13,
                        Q6_K = 14 };
constexpr uint32_t kQK_K = 256;
constexpr int kQK = 32;               // weights per quant block
constexpr int kQ4_0Bytes = 18;        // f16 d + 16 nibble bytes
constexpr int kQ4_1Bytes = 20;        // f16 d + f16 m + 16 nibble bytes
constexpr int kQ5_0Bytes = 22;        // f16 d + u32 qh + 16 nibble bytes
constexpr int kQ5_1Bytes = 24;        // f16 d + f16 m + u32 qh + 16 B
constexpr int kQ8_0Bytes = 34;        // f16 d + 32 int8

// ------------------------------------------------------------- f16 codec
// Round-to-nearest-even f32 -> f16 (bit-exact with numpy astype(float16)).
inline uint16_t f32_to_f16(float f) {
    uint32_t x;
    std::memcpy(&x, &f, 4);
    const uint32_t sign = (x >> 16) & 0x8000u;
    const int32_t exp = (int32_t)((x >> 23) & 0xFF) - 127 + 15;
    uint32_t mant = x & 0x7FFFFFu;
    if (((x >> 23) & 0xFF) == 0xFF) {  // inf/nan
        return (uint16_t)(sign | 0x7C00u | (mant ? 0x200u : 0));
    }
    if (exp >= 0x1F) return (uint16_t)(sign | 0x7C00u);  // overflow -> inf
    if (exp <= 0) {                                      // subnormal/zero
        if (exp < -10) return (uint16_t)sign;
        mant |= 0x800000u;
        const int shift = 14 - exp;
        const uint32_t q = mant >> shift;
        const uint32_t rem = mant & ((1u << shift) - 1);
        const uint32_t half = 1u << (shift - 1);
        uint32_t r = q;
        if (rem > half || (rem == half && (q & 1))) ++r;
        return (uint16_t)(sign | r);
    }
    uint32_t q = mant >> 13;
    const uint32_t rem = mant & 0x1FFFu;
    uint32_t r = ((uint32_t)exp << 10) | q;
    if (rem > 0x1000u || (rem == 0x1000u && (r & 1))) ++r;  // RNE
    return (uint16_t)(sign | r);
}

inline float f16_to_f32(uint16_t h) {
    const uint32_t sign = (uint32_t)(h & 0x8000u) << 16;
    const uint32_t exp = (h >> 10) & 0x1F;
    const uint32_t mant = h & 0x3FFu;
    uint32_t x;
    if (exp == 0) {
        if (mant == 0) {
            x = sign;
        } else {  // subnormal
            int e = -1;
            uint32_t m = mant;
            while (!(m & 0x400u)) {
                m <<= 1;
                ++e;
            }
            x = sign | ((uint32_t)(127 - 15 - e) << 23) |
                ((m & 0x3FFu) << 13);
        }
    } else if (exp == 0x1F) {
        x = sign | 0x7F800000u | (mant << 13);
    } else {
        x = sign | ((exp - 15 + 127) << 23) | (mant << 13);
    }
    float f;
    std::memcpy(&f, &x, 4);
    return f;
}

// ------------------------------------------------------------ q4 codecs

inline size_t row_bytes(GType t, uint32_t ne0) {
    switch (t) {
        case F32: return (size_t)ne0 * 4;
        case F16: return (size_t)ne0 * 2;
        case Q4_0:
            if (ne0 % kQK) throw std::runtime_error("q4_0 row not /32");
            return (size_t)(ne0 / kQK) * kQ4_0Bytes;
        case Q4_1:
            if (ne0 % kQK) throw std::runtime_error("q4_1 row not /32");
            return (size_t)(ne0 / kQK) * kQ4_1Bytes;
        case Q5_0:
            if (ne0 % kQK) throw std::runtime_error("q5_0 row not /32");
            return (size_t)(ne0 / kQK) * kQ5_0Bytes;
        case Q5_1:
            if (ne0 % kQK) throw std::runtime_error("q5_1 row not /32");
            return (size_t)(ne0 / kQK) * kQ5_1Bytes;
        case Q8_0:
            if (ne0 % kQK) throw std::runtime_error("q8_0 row not /32");
            return (size_t)(ne0 / kQK) * kQ8_0Bytes;
        case Q2_K:
            if (ne0 % kQK_K) throw std::runtime_error("q2_K row not /256");
            return (size_t)(ne0 / kQK_K) * 84;
        case Q3_K:
            if (ne0 % kQK_K) throw std::runtime_error("q3_K row not /256");
            return (size_t)(ne0 / kQK_K) * 110;
        case Q4_K:
            if (ne0 % kQK_K) throw std::runtime_error("q4_K row not /256");
            return (size_t)(ne0 / kQK_K) * 144;
        case Q5_K:
            if (ne0 % kQK_K) throw std::runtime_error("q5_K row not /256");
            return (size_t)(ne0 / kQK_K) * 176;
        case Q6_K:
            if (ne0 % kQK_K) throw std::runtime_error("q6_K row not /256");
            return (size_t)(ne0 / kQK_K) * 210;
    }
    throw std::runtime_error("unknown ggml type");
}

// One q4_0 block from 32 floats (signed-amax scaling, q4.py semantics).
inline void quantize_block_q4_0(const float* x, uint8_t* out) {
    float amax = 0.0f, m = 0.0f;
    for (int j = 0; j < kQK; ++j) {
        if (std::fabs(x[j]) > amax) {
            amax = std::fabs(x[j]);
            m = x[j];
        }
    }
    const float d = m / -8.0f;  // llama.cpp: invert the UNROUNDED scale
    const uint16_t dh = f32_to_f16(d);
    const float inv = (std::fabs(d) >= FLT_MIN) ? 1.0f / d : 0.0f;
    std::memcpy(out, &dh, 2);
    for (int j = 0; j < 16; ++j) {
        const float v0 = x[j] * inv, v1 = x[j + 16] * inv;
        const int q0 = std::clamp((int)std::nearbyintf(v0) + 8, 0, 15);
        const int q1 = std::clamp((int)std::nearbyintf(v1) + 8, 0, 15);
        out[2 + j] = (uint8_t)(q0 | (q1 << 4));
    }
}

inline void quantize_block_q4_1(const float* x, uint8_t* out) {
    float mn = x[0], mx = x[0];
    for (int j = 1; j < kQK; ++j) {
        mn = std::min(mn, x[j]);
        mx = std::max(mx, x[j]);
    }
    const float d = (mx - mn) / 15.0f;
    const uint16_t dh = f32_to_f16(d);
    const uint16_t mh = f32_to_f16(mn);
    const float m = mn;
    const float inv = (std::fabs(d) >= FLT_MIN) ? 1.0f / d : 0.0f;
    std::memcpy(out, &dh, 2);
    std::memcpy(out + 2, &mh, 2);
    for (int j = 0; j < 16; ++j) {
        const int q0 = std::clamp(
            (int)std::nearbyintf((x[j] - m) * inv), 0, 15);
        const int q1 = std::clamp(
            (int)std::nearbyintf((x[j + 16] - m) * inv), 0, 15);
        out[4 + j] = (uint8_t)(q0 | (q1 << 4));
    }
}

inline void dequantize_block_q4_0(const uint8_t* in, float* x) {
    uint16_t dh;
    std::memcpy(&dh, in, 2);
    const float d = f16_to_f32(dh);
    for (int j = 0; j < 16; ++j) {
        x[j] = d * (float)((int)(in[2 + j] & 0x0F) - 8);
        x[j + 16] = d * (float)((int)(in[2 + j] >> 4) - 8);
    }
}

inline void dequantize_block_q4_1(const uint8_t* in, float* x) {
    uint16_t dh, mh;
    std::memcpy(&dh, in, 2);
    std::memcpy(&mh, in + 2, 2);
    const float d = f16_to_f32(dh), m = f16_to_f32(mh);
    for (int j = 0; j < 16; ++j) {
        x[j] = d * (float)(in[2 + j] & 0x0F) + m;
        x[j + 16] = d * (float)(in[2 + j] >> 4) + m;
    }
}

inline void quantize_block_q5_0(const float* x, uint8_t* out) {
    float amax = 0.0f, m = 0.0f;
    for (int j = 0; j < kQK; ++j) {
        if (std::fabs(x[j]) > amax) { amax = std::fabs(x[j]); m = x[j]; }
    }
    const float d = m / -16.0f;
    const uint16_t dh = f32_to_f16(d);
    const float inv = (std::fabs(d) >= FLT_MIN) ? 1.0f / d : 0.0f;
    std::memcpy(out, &dh, 2);
    uint32_t qh = 0;
    for (int j = 0; j < 16; ++j) {
        const int q0 = std::clamp((int)std::nearbyintf(x[j] * inv) + 16,
                                  0, 31);
        const int q1 = std::clamp(
            (int)std::nearbyintf(x[j + 16] * inv) + 16, 0, 31);
        out[6 + j] = (uint8_t)((q0 & 0xF) | ((q1 & 0xF) << 4));
        qh |= (uint32_t)((q0 >> 4) & 1) << j;
        qh |= (uint32_t)((q1 >> 4) & 1) << (j + 16);
    }
    std::memcpy(out + 2, &qh, 4);
}

inline void dequantize_block_q5_0(const uint8_t* in, float* x) {
    uint16_t dh;
    std::memcpy(&dh, in, 2);
    uint32_t qh;
    std::memcpy(&qh, in + 2, 4);
    const float d = f16_to_f32(dh);
    for (int j = 0; j < 16; ++j) {
        const int q0 = (in[6 + j] & 0x0F) | (int)(((qh >> j) & 1) << 4);
        const int q1 = (in[6 + j] >> 4) | (int)(((qh >> (j + 16)) & 1) << 4);
        x[j] = d * (float)(q0 - 16);
        x[j + 16] = d * (float)(q1 - 16);
    }
}

inline void quantize_block_q5_1(const float* x, uint8_t* out) {
    float mn = x[0], mx = x[0];
    for (int j = 1; j < kQK; ++j) {
        mn = std::min(mn, x[j]);
        mx = std::max(mx, x[j]);
    }
    const float d = (mx - mn) / 31.0f;
    const uint16_t dh = f32_to_f16(d);
    const uint16_t mh = f32_to_f16(mn);
    const float m = mn;
    const float inv = (std::fabs(d) >= FLT_MIN) ? 1.0f / d : 0.0f;
    std::memcpy(out, &dh, 2);
    std::memcpy(out + 2, &mh, 2);
    uint32_t qh = 0;
    for (int j = 0; j < 16; ++j) {
        const int q0 = std::clamp(
            (int)std::nearbyintf((x[j] - m) * inv), 0, 31);
        const int q1 = std::clamp(
            (int)std::nearbyintf((x[j + 16] - m) * inv), 0, 31);
        out[8 + j] = (uint8_t)((q0 & 0xF) | ((q1 & 0xF) << 4));
        qh |= (uint32_t)((q0 >> 4) & 1) << j;
        qh |= (uint32_t)((q1 >> 4) & 1) << (j + 16);
    }
    std::memcpy(out + 4, &qh, 4);
}

inline void dequantize_block_q5_1(const uint8_t* in, float* x) {
    uint16_t dh, mh;
    std::memcpy(&dh, in, 2);
    std::memcpy(&mh, in + 2, 2);
    uint32_t qh;
    std::memcpy(&qh, in + 4, 4);
    const float d = f16_to_f32(dh), m = f16_to_f32(mh);
    for (int j = 0; j < 16; ++j) {
        const int q0 = (in[8 + j] & 0x0F) | (int)(((qh >> j) & 1) << 4);
        const int q1 = (in[8 + j] >> 4) | (int)(((qh >> (j + 16)) & 1) << 4);
        x[j] = d * (float)q0 + m;
        x[j + 16] = d * (float)q1 + m;
    }
}

inline void quantize_block_q8_0(const float* x, uint8_t* out) {
    float amax = 0.0f;
    for (int j = 0; j < kQK; ++j) amax = std::max(amax, std::fabs(x[j]));
    const float d = amax / 127.0f;
    const uint16_t dh = f32_to_f16(d);
    const float inv = (std::fabs(d) >= FLT_MIN) ? 1.0f / d : 0.0f;
    std::memcpy(out, &dh, 2);
    for (int j = 0; j < kQK; ++j)
        out[2 + j] = (uint8_t)(int8_t)std::clamp(
            (int)std::nearbyintf(x[j] * inv), -127, 127);
}

inline void dequantize_block_q8_0(const uint8_t* in, float* x) {
    uint16_t dh;
    std::memcpy(&dh, in, 2);
    const float d = f16_to_f32(dh);
    for (int j = 0; j < kQK; ++j) x[j] = d * (float)(int8_t)in[2 + j];
}

// --------------------------------------------------------------- file IO

struct Hparams {
    uint32_t n_vocab = 0, n_embd = 0, n_mult = 0, n_head = 0, n_layer = 0,
             n_rot = 0, ftype = 0;
    bool extended = false;       // 8-field header
    uint32_t first_layer = 0;    // valid when extended
    // GQA extension (GGJT version 4): n_head_kv right after n_head;
    // written only when gqa (MHA files stay byte-identical v3)
    bool gqa = false;
    uint32_t n_head_kv = 0;      // valid when gqa
};

struct Tensor {
    std::string name;
    std::vector<uint32_t> ne;  // ne[0] = contiguous dim
    GType gtype = F32;
    std::vector<uint8_t> raw;

    size_t rows() const {
        size_t r = 1;
        for (size_t i = 1; i < ne.size(); ++i) r *= ne[i];
        return r;
    }
    size_t nbytes() const { return rows() * row_bytes(gtype, ne[0]); }
};

struct File {
    Hparams hp;
    std::vector<std::pair<std::string, float>> vocab;
    std::vector<Tensor> tensors;
};

class Reader {
 public:
    explicit Reader(const std::string& path) {
        FILE* f = std::fopen(path.c_str(), "rb");
        if (!f) throw std::runtime_error("cannot open " + path);
        std::fseek(f, 0, SEEK_END);
        buf_.resize((size_t)std::ftell(f));
        std::fseek(f, 0, SEEK_SET);
        if (std::fread(buf_.data(), 1, buf_.size(), f) != buf_.size()) {
            std::fclose(f);
            throw std::runtime_error("short read on " + path);
        }
        std::fclose(f);
    }

    File parse(bool extended) {
        off_ = 0;
        File out;
        if (u32() != kMagic) throw std::runtime_error("bad GGJT magic");
        const uint32_t version = u32();
        if (version != kVersion && version != kVersionGqa)
            throw std::runtime_error("bad GGJT version");
        Hparams& hp = out.hp;
        hp.n_vocab = u32();
        hp.n_embd = u32();
        hp.n_mult = u32();
        hp.n_head = u32();
        hp.gqa = (version == kVersionGqa);
        if (hp.gqa) hp.n_head_kv = u32();
        hp.n_layer = u32();
        hp.n_rot = u32();
        hp.extended = extended;
        if (extended) hp.first_layer = u32();
        hp.ftype = u32();
        out.vocab.reserve(hp.n_vocab);
        for (uint32_t i = 0; i < hp.n_vocab; ++i) {
            const uint32_t ln = u32();
            std::string w((const char*)buf_.data() + off_, ln);
            off_ += ln;
            float score;
            std::memcpy(&score, buf_.data() + off_, 4);
            off_ += 4;
            out.vocab.emplace_back(std::move(w), score);
        }
        while (off_ < buf_.size()) {
            Tensor t;
            const uint32_t nd = u32();
            const uint32_t name_len = u32();
            t.gtype = (GType)u32();
            if (nd < 1 || nd > 2)
                throw std::runtime_error("unsupported tensor rank");
            t.ne.resize(nd);
            for (uint32_t i = 0; i < nd; ++i) t.ne[i] = u32();
            t.name.assign((const char*)buf_.data() + off_, name_len);
            off_ += name_len;
            off_ = (off_ + 31) & ~(size_t)31;
            const size_t sz = t.nbytes();
            if (off_ + sz > buf_.size())
                throw std::runtime_error("truncated tensor " + t.name);
            t.raw.assign(buf_.begin() + off_, buf_.begin() + off_ + sz);
            off_ += sz;
            out.tensors.push_back(std::move(t));
        }
        return out;
    }

 private:
    uint32_t u32() {
        if (off_ + 4 > buf_.size()) throw std::runtime_error("truncated");
        uint32_t v;
        std::memcpy(&v, buf_.data() + off_, 4);
        off_ += 4;
        return v;
    }
    std::vector<uint8_t> buf_;
    size_t off_ = 0;
};

inline void write_file(const std::string& path, const File& file) {
    FILE* f = std::fopen(path.c_str(), "wb");
    if (!f) throw std::runtime_error("cannot create " + path);
    size_t pos = 0;
    auto put = [&](const void* p, size_t n) {
        if (std::fwrite(p, 1, n, f) != n) {
            std::fclose(f);
            throw std::runtime_error("short write on " + path);
        }
        pos += n;
    };
    auto put32 = [&](uint32_t v) { put(&v, 4); };
    const Hparams& hp = file.hp;
    put32(kMagic);
    put32(hp.gqa ? kVersionGqa : kVersion);
    put32(hp.n_vocab);
    put32(hp.n_embd);
    put32(hp.n_mult);
    put32(hp.n_head);
    if (hp.gqa) put32(hp.n_head_kv);
    put32(hp.n_layer);
    put32(hp.n_rot);
    if (hp.extended) put32(hp.first_layer);
    put32(hp.ftype);
    if (file.vocab.size() != hp.n_vocab)
        throw std::runtime_error("vocab size != n_vocab");
    for (const auto& [w, score] : file.vocab) {
        put32((uint32_t)w.size());
        put(w.data(), w.size());
        put(&score, 4);
    }
    static const uint8_t zeros[32] = {0};
    for (const Tensor& t : file.tensors) {
        put32((uint32_t)t.ne.size());
        put32((uint32_t)t.name.size());
        put32((uint32_t)t.gtype);
        for (uint32_t d : t.ne) put32(d);
        put(t.name.data(), t.name.size());
        const size_t pad = (-pos) & 31;
        put(zeros, pad);
        if (t.raw.size() != t.nbytes())
            throw std::runtime_error("tensor size mismatch: " + t.name);
        put(t.raw.data(), t.raw.size());
    }
    std::fclose(f);
}

}  // namespace ggmlio
