// quantize — native CLI that requantizes a GGML model to
// q4_0/q4_1/q5_0/q5_1/q8_0.
//
// Native counterpart of the vendored llama.cpp `quantize` binary the
// reference's provisioning shells out to
// (/root/reference/distllm/cli_api/provision.py:213-217), clean-room per
// the block layouts in formats/q4.py: q4_0 blocks are
// bit-identical to the Python codec (signed-amax scale, f16 d, nibble
// pairs j/j+16). 1-D tensors stay f32, matching real checkpoints.
#include <cstdio>
#include <cstring>
#include <string>
#include <vector>

#include "ggmlio.hpp"
#include "kquants.hpp"

namespace {

std::vector<float> to_f32(const ggmlio::Tensor& t) {
    const size_t rows = t.rows(), cols = t.ne[0];
    std::vector<float> out(rows * cols);
    const uint8_t* p = t.raw.data();
    switch (t.gtype) {
        case ggmlio::F32:
            std::memcpy(out.data(), p, out.size() * 4);
            break;
        case ggmlio::F16:
            for (size_t i = 0; i < out.size(); ++i) {
                uint16_t h;
                std::memcpy(&h, p + i * 2, 2);
                out[i] = ggmlio::f16_to_f32(h);
            }
            break;
        case ggmlio::Q4_0:
            for (size_t i = 0; i < out.size() / ggmlio::kQK; ++i)
                ggmlio::dequantize_block_q4_0(p + i * ggmlio::kQ4_0Bytes,
                                              out.data() + i * ggmlio::kQK);
            break;
        case ggmlio::Q4_1:
            for (size_t i = 0; i < out.size() / ggmlio::kQK; ++i)
                ggmlio::dequantize_block_q4_1(p + i * ggmlio::kQ4_1Bytes,
                                              out.data() + i * ggmlio::kQK);
            break;
        case ggmlio::Q5_0:
            for (size_t i = 0; i < out.size() / ggmlio::kQK; ++i)
                ggmlio::dequantize_block_q5_0(p + i * ggmlio::kQ5_0Bytes,
                                              out.data() + i * ggmlio::kQK);
            break;
        case ggmlio::Q5_1:
            for (size_t i = 0; i < out.size() / ggmlio::kQK; ++i)
                ggmlio::dequantize_block_q5_1(p + i * ggmlio::kQ5_1Bytes,
                                              out.data() + i * ggmlio::kQK);
            break;
        case ggmlio::Q8_0:
            for (size_t i = 0; i < out.size() / ggmlio::kQK; ++i)
                ggmlio::dequantize_block_q8_0(p + i * ggmlio::kQ8_0Bytes,
                                              out.data() + i * ggmlio::kQK);
            break;
        case ggmlio::Q2_K:
            for (size_t i = 0; i < out.size() / kq::kQK_K; ++i)
                kq::dequantize_block_q2_K(p + i * kq::kQ2KBytes,
                                          out.data() + i * kq::kQK_K);
            break;
        case ggmlio::Q3_K:
            for (size_t i = 0; i < out.size() / kq::kQK_K; ++i)
                kq::dequantize_block_q3_K(p + i * kq::kQ3KBytes,
                                          out.data() + i * kq::kQK_K);
            break;
        case ggmlio::Q4_K:
            for (size_t i = 0; i < out.size() / kq::kQK_K; ++i)
                kq::dequantize_block_q4_K(p + i * kq::kQ4KBytes,
                                          out.data() + i * kq::kQK_K);
            break;
        case ggmlio::Q5_K:
            for (size_t i = 0; i < out.size() / kq::kQK_K; ++i)
                kq::dequantize_block_q5_K(p + i * kq::kQ5KBytes,
                                          out.data() + i * kq::kQK_K);
            break;
        case ggmlio::Q6_K:
            for (size_t i = 0; i < out.size() / kq::kQK_K; ++i)
                kq::dequantize_block_q6_K(p + i * kq::kQ6KBytes,
                                          out.data() + i * kq::kQK_K);
            break;
    }
    return out;
}

ggmlio::Tensor from_f32(const ggmlio::Tensor& t, const std::vector<float>& x,
                        ggmlio::GType target) {
    ggmlio::Tensor out;
    out.name = t.name;
    out.ne = t.ne;
    out.gtype = target;
    out.raw.resize(out.nbytes());
    uint8_t* p = out.raw.data();
    switch (target) {
        case ggmlio::F32:
            std::memcpy(p, x.data(), x.size() * 4);
            break;
        case ggmlio::F16:
            for (size_t i = 0; i < x.size(); ++i) {
                const uint16_t h = ggmlio::f32_to_f16(x[i]);
                std::memcpy(p + i * 2, &h, 2);
            }
            break;
        case ggmlio::Q4_0:
            for (size_t i = 0; i < x.size() / ggmlio::kQK; ++i)
                ggmlio::quantize_block_q4_0(x.data() + i * ggmlio::kQK,
                                            p + i * ggmlio::kQ4_0Bytes);
            break;
        case ggmlio::Q4_1:
            for (size_t i = 0; i < x.size() / ggmlio::kQK; ++i)
                ggmlio::quantize_block_q4_1(x.data() + i * ggmlio::kQK,
                                            p + i * ggmlio::kQ4_1Bytes);
            break;
        case ggmlio::Q5_0:
            for (size_t i = 0; i < x.size() / ggmlio::kQK; ++i)
                ggmlio::quantize_block_q5_0(x.data() + i * ggmlio::kQK,
                                            p + i * ggmlio::kQ5_0Bytes);
            break;
        case ggmlio::Q5_1:
            for (size_t i = 0; i < x.size() / ggmlio::kQK; ++i)
                ggmlio::quantize_block_q5_1(x.data() + i * ggmlio::kQK,
                                            p + i * ggmlio::kQ5_1Bytes);
            break;
        case ggmlio::Q8_0:
            for (size_t i = 0; i < x.size() / ggmlio::kQK; ++i)
                ggmlio::quantize_block_q8_0(x.data() + i * ggmlio::kQK,
                                            p + i * ggmlio::kQ8_0Bytes);
            break;
        case ggmlio::Q2_K:
            for (size_t i = 0; i < x.size() / kq::kQK_K; ++i)
                kq::quantize_block_q2_K(x.data() + i * kq::kQK_K,
                                        p + i * kq::kQ2KBytes);
            break;
        case ggmlio::Q3_K:
            for (size_t i = 0; i < x.size() / kq::kQK_K; ++i)
                kq::quantize_block_q3_K(x.data() + i * kq::kQK_K,
                                        p + i * kq::kQ3KBytes);
            break;
        case ggmlio::Q4_K:
            for (size_t i = 0; i < x.size() / kq::kQK_K; ++i)
                kq::quantize_block_q4_K(x.data() + i * kq::kQK_K,
                                        p + i * kq::kQ4KBytes);
            break;
        case ggmlio::Q5_K:
            for (size_t i = 0; i < x.size() / kq::kQK_K; ++i)
                kq::quantize_block_q5_K(x.data() + i * kq::kQK_K,
                                        p + i * kq::kQ5KBytes);
            break;
        case ggmlio::Q6_K:
            for (size_t i = 0; i < x.size() / kq::kQK_K; ++i)
                kq::quantize_block_q6_K(x.data() + i * kq::kQK_K,
                                        p + i * kq::kQ6KBytes);
            break;
    }
    return out;
}

// k-quant targets need row length % 256 == 0; mirror the Python
// provisioner's per-tensor fallback (cluster/provision.py
// _KQUANT_FALLBACK: q2..q5_K -> q5_0, q6_K -> q8_0)
ggmlio::GType effective_target(ggmlio::GType target, uint32_t ne0) {
    if (target >= ggmlio::Q2_K && target <= ggmlio::Q6_K &&
        ne0 % kq::kQK_K != 0)
        return target == ggmlio::Q6_K ? ggmlio::Q8_0 : ggmlio::Q5_0;
    return target;
}

}  // namespace

int main(int argc, char** argv) {
    if (argc < 4) {
        std::fprintf(stderr,
                     "usage: quantize <in.bin> <out.bin> "
                     "<q4_0|q4_1|q5_0|q5_1|q8_0|q2_K|q3_K|q4_K|q5_K|q6_K"
                     "|f16>\n");
        return 2;
    }
    ggmlio::GType target;
    uint32_t ftype;  // llama_ftype id (differs from the GType for q5/q8)
    const std::string t = argv[3];
    if (t == "q4_0") { target = ggmlio::Q4_0; ftype = 2; }
    else if (t == "q4_1") { target = ggmlio::Q4_1; ftype = 3; }
    else if (t == "f16") { target = ggmlio::F16; ftype = 1; }
    else if (t == "q8_0") { target = ggmlio::Q8_0; ftype = 7; }
    else if (t == "q5_0") { target = ggmlio::Q5_0; ftype = 8; }
    else if (t == "q5_1") { target = ggmlio::Q5_1; ftype = 9; }
    // k-quant ftypes use the _M variants, matching
    // cluster/provision.py VALID_QUANT
    else if (t == "q2_K") { target = ggmlio::Q2_K; ftype = 10; }
    else if (t == "q3_K") { target = ggmlio::Q3_K; ftype = 12; }
    else if (t == "q4_K") { target = ggmlio::Q4_K; ftype = 15; }
    else if (t == "q5_K") { target = ggmlio::Q5_K; ftype = 17; }
    else if (t == "q6_K") { target = ggmlio::Q6_K; ftype = 18; }
    else {
        std::fprintf(stderr, "unknown target type %s\n", t.c_str());
        return 2;
    }
    try {
        ggmlio::Reader reader(argv[1]);
        ggmlio::File in = reader.parse(/*extended=*/false);
        ggmlio::File out;
        out.hp = in.hp;
        out.hp.ftype = ftype;
        out.vocab = in.vocab;
        size_t quantized = 0;
        for (const auto& ten : in.tensors) {
            const ggmlio::GType tt = effective_target(target, ten.ne[0]);
            if (ten.ne.size() == 1 || ten.gtype == tt) {
                out.tensors.push_back(ten);  // 1-D stays f32, same-type copy
                continue;
            }
            out.tensors.push_back(from_f32(ten, to_f32(ten), tt));
            ++quantized;
        }
        ggmlio::write_file(argv[2], out);
        std::printf("quantized %zu of %zu tensors -> %s\n", quantized,
                    in.tensors.size(), argv[2]);
        return 0;
    } catch (const std::exception& e) {
        std::fprintf(stderr, "error: %s\n", e.what());
        return 1;
    }
}
