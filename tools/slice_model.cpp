// slice_model — native CLI that cuts a GGML model into layer-range slice
// files and extracts the extra (embedding/norm/output) layers.
//
// Native counterpart of the reference's in-tree C++ tool
// (/root/reference/distllm/slice_model.cpp:361-446: `slice_model slice
// <model> <from> <to> [out]` and `slice_model extra_layers <model> [out]`),
// re-implemented clean-room against the byte layout documented in
// formats/ggml.py. Output files are byte-identical to the Python slicer
// (asserted by tests/test_native_tools.py):
//   * slices: extended 8-field header with first_layer; tensor names keep
//     their ORIGINAL layer indices (loaders re-base with first_layer);
//   * extra_layers.bin: n_layer=0, first_layer=0xFFFFFFFF, holds
//     tok_embeddings.weight, norm.weight, output.weight.
#include <cstdio>
#include <cstring>
#include <string>

#include "ggmlio.hpp"

namespace {

bool starts_with(const std::string& s, const std::string& p) {
    return s.rfind(p, 0) == 0;
}

// tensors "layers.<i>.*" with a <= i <= b
bool in_layer_range(const std::string& name, uint32_t a, uint32_t b) {
    if (!starts_with(name, "layers.")) return false;
    const size_t dot = name.find('.', 7);
    if (dot == std::string::npos) return false;
    const std::string idx = name.substr(7, dot - 7);
    if (idx.empty() ||
        idx.find_first_not_of("0123456789") != std::string::npos)
        return false;
    const uint32_t i = (uint32_t)std::stoul(idx);
    return a <= i && i <= b;
}

bool is_extra(const std::string& name) {
    return name == "tok_embeddings.weight" || name == "norm.weight" ||
           name == "output.weight";
}

int usage() {
    std::fprintf(stderr,
                 "usage:\n"
                 "  slice_model slice <model.bin> <from> <to> [out.bin]\n"
                 "  slice_model extra_layers <model.bin> [out.bin]\n");
    return 2;
}

}  // namespace

int main(int argc, char** argv) {
    if (argc < 3) return usage();
    const std::string cmd = argv[1];
    const std::string model_path = argv[2];
    try {
        ggmlio::Reader reader(model_path);
        ggmlio::File in = reader.parse(/*extended=*/false);

        ggmlio::File out;
        out.hp = in.hp;
        out.hp.extended = true;
        out.vocab = in.vocab;

        std::string out_path;
        if (cmd == "slice") {
            if (argc < 5) return usage();
            const uint32_t a = (uint32_t)std::stoul(argv[3]);
            const uint32_t b = (uint32_t)std::stoul(argv[4]);
            if (b < a || b >= in.hp.n_layer) {
                std::fprintf(stderr, "bad layer range [%u, %u] for %u\n", a,
                             b, in.hp.n_layer);
                return 2;
            }
            out.hp.first_layer = a;
            out.hp.n_layer = b - a + 1;
            out_path = (argc > 5) ? argv[5]
                                  : ("slice_" + std::to_string(a) + "_" +
                                     std::to_string(b) + ".bin");
            for (auto& t : in.tensors)
                if (in_layer_range(t.name, a, b))
                    out.tensors.push_back(std::move(t));
        } else if (cmd == "extra_layers") {
            out.hp.first_layer = ggmlio::kExtraLayersFirstLayer;
            out.hp.n_layer = 0;
            out_path = (argc > 3) ? argv[3] : "extra_layers.bin";
            for (auto& t : in.tensors)
                if (is_extra(t.name)) out.tensors.push_back(std::move(t));
        } else {
            return usage();
        }
        if (out.tensors.empty()) {
            std::fprintf(stderr, "no tensors selected\n");
            return 1;
        }
        ggmlio::write_file(out_path, out);
        std::printf("%s: %zu tensors -> %s\n", cmd.c_str(),
                    out.tensors.size(), out_path.c_str());
        return 0;
    } catch (const std::exception& e) {
        std::fprintf(stderr, "error: %s\n", e.what());
        return 1;
    }
}
