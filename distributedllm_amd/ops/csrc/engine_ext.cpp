// Torch-extension binding of the native slice engine.
//
// MI355X-native replacement of the reference's CPython extension `llm.so`
// (/root/reference/distllm/tensor_processor.cpp:2238-2260 exposed 9 functions
// around a global TransformerSlice). This engine instead:
//   * is an explicit object (no global mutable slice, SURVEY.md §5.2 hazard),
//   * keeps weights resident in HBM3E in a repacked SoA layout,
//   * exchanges activations as torch tensors (DLPack-compatible device
//     buffers), never per-element Python lists,
//   * is stateless w.r.t. generation: positions/sequence ids are explicit
//     device tensors, so the decode step is hipGraph-capturable and the KV
//     "clear_context" is a host-side position reset.
//
// The forward pass launches the CDNA4 kernels of kernels.hip per layer on
// the current torch HIP stream.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include <memory>
#include <string>
#include <vector>

#include "kernels.h"

namespace {

struct DevMat {
    torch::Tensor data;    // u8 nibbles (q4_*) or f16/f32 values
    torch::Tensor scales;  // f16 scales (q4_*), undefined otherwise
    WMat w{};
};

struct Layer {
    torch::Tensor attn_norm, ffn_norm;  // f32 [E]
    DevMat wq, wk, wv, wo, w1, w2, w3;
};

constexpr int kMaxTokens = 16;  // per-forward token-tile cap (TMAX ceiling)

DevMat make_devmat(torch::Tensor data, torch::Tensor scales, int64_t wtype,
                   int64_t rows, int64_t cols) {
    TORCH_CHECK(data.is_cuda() && data.is_contiguous(),
                "weight data must be contiguous on device");
    DevMat m;
    m.data = data;
    m.w.rows = (int)rows;
    m.w.cols = (int)cols;
    m.w.wtype = (int)wtype;
    m.w.data = data.data_ptr();
    if (wtype == W_Q4_0 || wtype == W_Q4_1) {
        TORCH_CHECK(scales.defined() && scales.is_cuda() &&
                        scales.is_contiguous() &&
                        scales.scalar_type() == torch::kFloat16,
                    "q4 weights need f16 scales on device");
        const int nb = (int)(cols / 32);
        const int64_t per_block = (wtype == W_Q4_1) ? 2 : 1;
        TORCH_CHECK(scales.numel() == rows * nb * per_block,
                    "scales size mismatch");
        TORCH_CHECK(data.numel() == rows * nb * 16 &&
                        data.scalar_type() == torch::kUInt8,
                    "q4 nibble data size mismatch");
        m.scales = scales;
        m.w.scales = scales.data_ptr();
    } else {
        const auto want =
            (wtype == W_F16) ? torch::kFloat16 : torch::kFloat32;
        TORCH_CHECK(data.scalar_type() == want, "weight dtype mismatch");
        TORCH_CHECK(data.numel() == rows * cols, "weight size mismatch");
        m.w.scales = nullptr;
    }
    return m;
}

torch::Tensor check_f32(const torch::Tensor& t, const char* name) {
    TORCH_CHECK(t.is_cuda() && t.is_contiguous() &&
                    t.scalar_type() == torch::kFloat32,
                name, " must be a contiguous f32 device tensor");
    return t;
}

torch::Tensor check_i32(const torch::Tensor& t, const char* name) {
    TORCH_CHECK(t.is_cuda() && t.is_contiguous() &&
                    t.scalar_type() == torch::kInt32,
                name, " must be a contiguous i32 device tensor");
    return t;
}

class SliceEngine {
 public:
    SliceEngine(int64_t n_embd, int64_t n_head, int64_t n_layers,
                int64_t n_ff, int64_t n_ctx, int64_t max_batch, double eps,
                double rope_base)
        : E_((int)n_embd),
          H_((int)n_head),
          D_((int)(n_embd / n_head)),
          F_((int)n_ff),
          L_((int)n_layers),
          ctx_((int)n_ctx),
          B_((int)max_batch),
          eps_((float)eps) {
        TORCH_CHECK(E_ % H_ == 0, "n_embd not divisible by n_head");
        TORCH_CHECK(D_ % 2 == 0, "head_dim must be even for RoPE pairs");
        TORCH_CHECK(D_ <= 256, "head_dim > 256 unsupported");
        layers_.resize(L_);
        loaded_.assign(L_, false);
        auto dev = torch::TensorOptions().device(torch::kCUDA);
        auto f32 = dev.dtype(torch::kFloat32);
        auto f16 = dev.dtype(torch::kFloat16);
        // KV cache: [L, B, ctx, E] f16 — the HBM3E-resident analog of the
        // reference's kv_cache_init (tensor_processor.cpp:1089-1132).
        k_cache_ = torch::zeros({L_, B_, ctx_, E_}, f16);
        v_cache_ = torch::zeros({L_, B_, ctx_, E_}, f16);
        // RoPE pair frequencies: theta_i = pos * base^(-2i/D)
        auto freqs = torch::pow(
            (float)rope_base,
            torch::arange(0, D_ / 2, dev.dtype(torch::kFloat32)) *
                (-2.0f / (float)D_));
        inv_freq_ = freqs.contiguous();
        xn_ = torch::empty({kMaxTokens, E_}, f32);
        qb_ = torch::empty({kMaxTokens, E_}, f32);
        ab_ = torch::empty({kMaxTokens, E_}, f32);
        ffb_ = torch::empty({kMaxTokens, F_}, f32);
    }

    void set_layer(int64_t li, torch::Tensor attn_norm,
                   torch::Tensor ffn_norm, py::list mats) {
        TORCH_CHECK(li >= 0 && li < L_, "layer index out of range");
        TORCH_CHECK(mats.size() == 7, "expected 7 matrices (q,k,v,o,1,2,3)");
        Layer& l = layers_[li];
        l.attn_norm = check_f32(attn_norm, "attn_norm");
        l.ffn_norm = check_f32(ffn_norm, "ffn_norm");
        DevMat* slots[7] = {&l.wq, &l.wk, &l.wv, &l.wo, &l.w1, &l.w2, &l.w3};
        const int64_t rows[7] = {E_, E_, E_, E_, F_, E_, F_};
        const int64_t cols[7] = {E_, E_, E_, E_, E_, F_, E_};
        for (size_t i = 0; i < 7; ++i) {
            auto tup = mats[i].cast<py::tuple>();
            *slots[i] = make_devmat(tup[0].cast<torch::Tensor>(),
                                    tup[1].cast<torch::Tensor>(),
                                    tup[2].cast<int64_t>(), rows[i], cols[i]);
        }
        loaded_[li] = true;
    }

    void set_extra(torch::Tensor tok_data, torch::Tensor tok_scales,
                   int64_t tok_wtype, torch::Tensor norm_w,
                   torch::Tensor out_data, torch::Tensor out_scales,
                   int64_t out_wtype, int64_t n_vocab) {
        V_ = (int)n_vocab;
        tok_ = make_devmat(tok_data, tok_scales, tok_wtype, V_, E_);
        out_ = make_devmat(out_data, out_scales, out_wtype, V_, E_);
        final_norm_ = check_f32(norm_w, "norm_w");
        has_extra_ = true;
    }

    // x: [T, E] f32 (modified in place and returned), pos/seq: [T] i32.
    torch::Tensor forward(torch::Tensor x, torch::Tensor pos,
                          torch::Tensor seq) {
        check_f32(x, "x");
        check_i32(pos, "pos");
        check_i32(seq, "seq");
        const int T = (int)x.size(0);
        TORCH_CHECK(x.dim() == 2 && x.size(1) == E_, "x must be [T, E]");
        TORCH_CHECK(T >= 1 && T <= kMaxTokens,
                    "forward handles at most ", kMaxTokens,
                    " tokens per call; tile larger batches host-side");
        TORCH_CHECK(pos.numel() == T && seq.numel() == T, "pos/seq size");
        for (int li = 0; li < L_; ++li)
            TORCH_CHECK(loaded_[li], "layer ", li, " not loaded");
        hipStream_t s = c10::hip::getCurrentHIPStream().stream();
        float* xp = x.data_ptr<float>();
        const int* pp = pos.data_ptr<int>();
        const int* sp = seq.data_ptr<int>();
        const size_t layer_stride = (size_t)B_ * ctx_ * E_;
        __half* kbase = reinterpret_cast<__half*>(k_cache_.data_ptr());
        __half* vbase = reinterpret_cast<__half*>(v_cache_.data_ptr());
        float* xn = xn_.data_ptr<float>();
        float* qb = qb_.data_ptr<float>();
        float* ab = ab_.data_ptr<float>();
        float* ffb = ffb_.data_ptr<float>();
        const float* ifr = inv_freq_.data_ptr<float>();
        for (int li = 0; li < L_; ++li) {
            Layer& l = layers_[li];
            __half* kc = kbase + (size_t)li * layer_stride;
            __half* vc = vbase + (size_t)li * layer_stride;
            launch_rmsnorm(s, xp, l.attn_norm.data_ptr<float>(), xn, T, E_,
                           eps_);
            launch_qkv_rope_append(s, l.wq.w, l.wk.w, l.wv.w, xn, qb, kc, vc,
                                   pp, sp, ifr, E_, D_, ctx_, T);
            launch_attention(s, qb, kc, vc, ab, pp, sp, T, H_, E_, D_, ctx_);
            launch_gemv(s, l.wo.w, ab, /*res=*/xp, xp, T);
            launch_rmsnorm(s, xp, l.ffn_norm.data_ptr<float>(), xn, T, E_,
                           eps_);
            launch_ffn_gate(s, l.w1.w, l.w3.w, xn, ffb, T);
            launch_gemv(s, l.w2.w, ffb, /*res=*/xp, xp, T);
        }
        return x;
    }

    torch::Tensor embed(torch::Tensor tokens) {
        TORCH_CHECK(has_extra_, "extra layers not loaded");
        check_i32(tokens, "tokens");
        const int T = (int)tokens.numel();
        auto out = torch::empty(
            {T, E_},
            torch::TensorOptions().device(torch::kCUDA).dtype(torch::kFloat32));
        hipStream_t s = c10::hip::getCurrentHIPStream().stream();
        launch_embed(s, tok_.w, tokens.data_ptr<int>(),
                     out.data_ptr<float>(), T, E_);
        return out;
    }

    // Final RMSNorm + lm_head (reference: get_llm_output,
    // tensor_processor.cpp:1787-1892). all_logits=false returns only the
    // last row's logits.
    torch::Tensor logits(torch::Tensor x, bool all_logits) {
        TORCH_CHECK(has_extra_, "extra layers not loaded");
        check_f32(x, "x");
        TORCH_CHECK(x.dim() == 2 && x.size(1) == E_, "x must be [T, E]");
        torch::Tensor xin = all_logits ? x : x.slice(0, x.size(0) - 1);
        xin = xin.contiguous();
        const int T = (int)xin.size(0);
        TORCH_CHECK(T <= kMaxTokens, "logits: too many rows per call");
        hipStream_t s = c10::hip::getCurrentHIPStream().stream();
        launch_rmsnorm(s, xin.data_ptr<float>(),
                       final_norm_.data_ptr<float>(), xn_.data_ptr<float>(),
                       T, E_, eps_);
        auto lg = torch::empty(
            {T, V_},
            torch::TensorOptions().device(torch::kCUDA).dtype(torch::kFloat32));
        launch_gemv(s, out_.w, xn_.data_ptr<float>(), nullptr,
                    lg.data_ptr<float>(), T);
        return lg;
    }

    torch::Tensor argmax(torch::Tensor lg) {
        check_f32(lg, "logits");
        const int T = (int)lg.size(0);
        auto out = torch::empty(
            {T},
            torch::TensorOptions().device(torch::kCUDA).dtype(torch::kInt32));
        hipStream_t s = c10::hip::getCurrentHIPStream().stream();
        launch_argmax(s, lg.data_ptr<float>(), out.data_ptr<int>(), T,
                      (int)lg.size(1));
        return out;
    }

    int64_t max_tokens() const { return kMaxTokens; }
    int64_t n_ctx() const { return ctx_; }
    int64_t max_batch() const { return B_; }

 private:
    int E_, H_, D_, F_, L_, ctx_, B_;
    int V_ = 0;
    float eps_;
    std::vector<Layer> layers_;
    std::vector<bool> loaded_;
    torch::Tensor k_cache_, v_cache_, inv_freq_;
    torch::Tensor xn_, qb_, ab_, ffb_;
    bool has_extra_ = false;
    DevMat tok_, out_;
    torch::Tensor final_norm_;
};

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.doc() = "MI355X-native layer-slice inference engine (CDNA4 HIP kernels)";
    py::class_<SliceEngine, std::shared_ptr<SliceEngine>>(m, "SliceEngine")
        .def(py::init<int64_t, int64_t, int64_t, int64_t, int64_t, int64_t,
                      double, double>(),
             py::arg("n_embd"), py::arg("n_head"), py::arg("n_layers"),
             py::arg("n_ff"), py::arg("n_ctx"), py::arg("max_batch"),
             py::arg("eps"), py::arg("rope_base"))
        .def("set_layer", &SliceEngine::set_layer)
        .def("set_extra", &SliceEngine::set_extra)
        .def("forward", &SliceEngine::forward)
        .def("embed", &SliceEngine::embed)
        .def("logits", &SliceEngine::logits, py::arg("x"),
             py::arg("all_logits") = false)
        .def("argmax", &SliceEngine::argmax)
        .def_property_readonly("max_tokens", &SliceEngine::max_tokens)
        .def_property_readonly("n_ctx", &SliceEngine::n_ctx)
        .def_property_readonly("max_batch", &SliceEngine::max_batch);
}
