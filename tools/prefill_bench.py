#!/usr/bin/env python3
"""Prompt-prefill throughput: whole-model forward over a span of S
tokens (the native large-M kernel path for S > 64). Round-1 baselines
to beat (3B q4_0, rocBLAS-over-detiled-f16): 32.9k tok/s @ S=512,
37.8k @ 1024; 64-token tile path ~20-25k."""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import argparse
import json
import time

import torch

from distributedllm_amd.engine import HIPSliceEngine
from distributedllm_amd.formats import ggml
from distributedllm_amd.models.llama import PRESETS

FTYPES = {"q4_0": ggml.FTYPE_MOSTLY_Q4_0, "q8_0": ggml.FTYPE_MOSTLY_Q8_0,
          "f16": ggml.FTYPE_MOSTLY_F16}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="open_llama_3b")
    ap.add_argument("--ftype", default="q4_0", choices=list(FTYPES))
    ap.add_argument("--spans", default="128,256,512,1024,2048")
    ap.add_argument("--iters", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--streams", type=int, default=1,
                    help="concatenated same-length prompts of distinct "
                         "sequences in ONE call (mixed admission stream)")
    args = ap.parse_args()
    p = PRESETS[args.model]
    hp = p.hparams(FTYPES[args.ftype])
    spans = [int(s) for s in args.spans.split(",")]
    ctx = max(spans)
    eng = HIPSliceEngine.random(hp, n_layers=p.n_layer, n_ctx=ctx,
                                max_batch=max(args.streams, 1), seed=0,
                                with_extra=True, max_prefill=ctx)
    g = torch.Generator(device="cuda")
    g.manual_seed(1)
    for S in spans:
        toks = torch.randint(3, hp.n_vocab, (S * args.streams,),
                             dtype=torch.int32, device="cuda", generator=g)
        x0 = eng.embed(toks)
        pos = torch.arange(S, dtype=torch.int32,
                           device="cuda").repeat(args.streams)
        seq = torch.arange(args.streams, dtype=torch.int32,
                           device="cuda").repeat_interleave(S)
        for _ in range(args.warmup):
            eng.forward(x0.clone(), pos, seq)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            eng.forward(x0.clone(), pos, seq)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / args.iters
        print(json.dumps({
            "span": S, "streams": args.streams,
            "prefill_tok_s": round(S * args.streams / dt, 1),
            "ms": round(dt * 1e3, 2), "ftype": args.ftype,
            "model": p.name}), flush=True)


if __name__ == "__main__":
    main()
