#!/usr/bin/env python3
"""Big-checkpoint load path: streaming-write a synthetic GGML file of a
real model's architecture, then mmap-load + repack it into the HIP
engine, reporting wall time and peak host RSS (VERDICT r1 missing #5:
the loader must stay RAM-bounded at 100 GB-class checkpoints like the
reference's mmap load, tensor_processor.cpp:996-1074)."""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import argparse
import json
import resource
import time

import numpy as np

from distributedllm_amd.formats import ggml, synthetic
from distributedllm_amd.models.llama import (
    PRESETS, layer_tensor_names)

FTYPES = {"q4_0": ggml.FTYPE_MOSTLY_Q4_0, "q4_1": ggml.FTYPE_MOSTLY_Q4_1,
          "q5_0": ggml.FTYPE_MOSTLY_Q5_0, "q5_1": ggml.FTYPE_MOSTLY_Q5_1,
          "q8_0": ggml.FTYPE_MOSTLY_Q8_0, "f16": ggml.FTYPE_MOSTLY_F16,
          "q2_K": ggml.FTYPE_MOSTLY_Q2_K, "q3_K": ggml.FTYPE_MOSTLY_Q3_K_M,
          "q4_K": ggml.FTYPE_MOSTLY_Q4_K_M,
          "q5_K": ggml.FTYPE_MOSTLY_Q5_K_M,
          "q6_K": ggml.FTYPE_MOSTLY_Q6_K}


def build_streaming(path: str, preset, ftype: int, seed: int = 0,
                    fast: bool = False) -> float:
    """Write the synthetic model file tensor by tensor (bounded RAM).

    fast=True writes RANDOM RAW BYTES for the 2-D tensors (any bit
    pattern is a valid quantized block; f16 scale bytes may be inf/nan
    patterns, fine for a LOAD benchmark) — the f32->quantize path costs
    minutes of CPU at 30B+ scale and measures the codec, not the
    loader."""
    t0 = time.perf_counter()
    hp = preset.hparams(ftype)
    wt = ggml._FTYPE_TO_GGML[ftype]
    rng = np.random.default_rng(seed)
    E, F, V = preset.n_embd, preset.n_ff, preset.n_vocab
    Ekv = preset.n_embd_kv

    def mat(name, rows, cols):
        if fast:
            nb = ggml.tensor_nbytes(wt, (cols, rows))
            return ggml.GGMLTensor(name=name, ne=(cols, rows), gtype=wt,
                                   raw=rng.bytes(nb))
        a = rng.standard_normal((rows, cols), dtype=np.float32) * 0.02
        return ggml.GGMLTensor.from_f32(name, a, wt)

    def norm(name, n):
        a = 1.0 + rng.standard_normal(n).astype(np.float32) * 0.01
        return ggml.GGMLTensor.from_f32(name, a, ggml.GGML_TYPE_F32)

    with ggml.GGMLWriter(path, hp,
                         synthetic.synthetic_vocab(V)) as w:
        w.add(mat("tok_embeddings.weight", V, E))
        w.add(norm("norm.weight", E))
        w.add(mat("output.weight", V, E))
        for i in range(preset.n_layer):
            pre = f"layers.{i}."
            w.add(norm(pre + "attention_norm.weight", E))
            w.add(mat(pre + "attention.wq.weight", E, E))
            w.add(mat(pre + "attention.wk.weight", Ekv, E))
            w.add(mat(pre + "attention.wv.weight", Ekv, E))
            w.add(mat(pre + "attention.wo.weight", E, E))
            w.add(norm(pre + "ffn_norm.weight", E))
            w.add(mat(pre + "feed_forward.w1.weight", F, E))
            w.add(mat(pre + "feed_forward.w2.weight", E, F))
            w.add(mat(pre + "feed_forward.w3.weight", F, E))
    return time.perf_counter() - t0


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama_13b")
    ap.add_argument("--ftype", default="f16", choices=list(FTYPES))
    ap.add_argument("--path", default="/tmp/load_bench_model.bin")
    ap.add_argument("--ctx", type=int, default=512)
    ap.add_argument("--keep", action="store_true")
    ap.add_argument("--fast-build", action="store_true",
                    help="random raw bytes instead of f32+quantize")
    args = ap.parse_args()
    p = PRESETS[args.model]
    ftype = FTYPES[args.ftype]

    t_build = None
    if not os.path.exists(args.path):
        t_build = build_streaming(args.path, p, ftype,
                                  fast=args.fast_build)
    fsize = os.path.getsize(args.path)

    rss0 = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss * 1024
    t0 = time.perf_counter()
    f = ggml.GGMLFile.load(args.path, extended=False, use_mmap=True)
    t_parse = time.perf_counter() - t0

    import torch
    assert torch.cuda.is_available(), "load bench needs the GPU"
    from distributedllm_amd.engine import HIPSliceEngine
    t0 = time.perf_counter()
    eng = HIPSliceEngine.from_ggml(f, n_ctx=args.ctx, max_batch=1)
    torch.cuda.synchronize()
    t_load = time.perf_counter() - t0
    rss1 = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss * 1024
    free, total = torch.cuda.mem_get_info()
    print(json.dumps({
        "model": p.name, "ftype": args.ftype,
        "file_gb": round(fsize / 1e9, 2),
        "build_s": round(t_build, 1) if t_build else None,
        "parse_s": round(t_parse, 2),
        "load_repack_s": round(t_load, 1),
        "gbps": round(fsize / 1e9 / t_load, 2),
        "peak_rss_gb": round(rss1 / 1e9, 2),
        "rss_delta_gb": round((rss1 - rss0) / 1e9, 2),
        "hbm_used_gb": round((total - free) / 1e9, 2),
        "n_layers": eng.n_layers}), flush=True)
    if not args.keep:
        os.remove(args.path)


if __name__ == "__main__":
    main()
