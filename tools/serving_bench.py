#!/usr/bin/env python3
"""Continuous-batching serving benchmark (GPU): N requests with random
prompts served concurrently through serving.ContinuousBatcher on one
engine — measures end-to-end serving tokens/s including scheduling,
logits and sampling (bench.py measures the raw decode pipeline)."""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch

from distributedllm_amd.engine import HIPSliceEngine
from distributedllm_amd.formats import ggml
from distributedllm_amd.models.llama import PRESETS
from distributedllm_amd.serving import ContinuousBatcher


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="open_llama_3b")
    ap.add_argument("--requests", type=int, default=256)
    ap.add_argument("--prompt-len", type=int, default=32)
    ap.add_argument("--num-tokens", type=int, default=64)
    ap.add_argument("--slots", type=int, default=64,
                    help="KV slots per lane")
    ap.add_argument("--lanes", type=int, default=5,
                    help="concurrent stream lanes (weight-sharing "
                         "clones); total slots = slots x lanes")
    ap.add_argument("--ctx", type=int, default=2048)
    args = ap.parse_args()

    hp = PRESETS[args.model].hparams(ggml.FTYPE_MOSTLY_Q4_0)
    eng = HIPSliceEngine.random(hp, n_layers=hp.n_layer, n_ctx=args.ctx,
                                max_batch=args.slots, with_extra=True)
    lanes = None
    if args.lanes > 1:
        lanes = [eng] + [eng.clone_shared()
                         for _ in range(args.lanes - 1)]
    g = torch.Generator().manual_seed(1)
    bat = ContinuousBatcher(eng, engines=lanes)
    reqs = [bat.submit(torch.randint(3, hp.n_vocab, (args.prompt_len,),
                                     generator=g).tolist(),
                       args.num_tokens)
            for _ in range(args.requests)]
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    steps = 0
    done_at = {}
    while bat.pending:
        fin = bat.step()
        if fin:
            torch.cuda.synchronize()
            now = time.perf_counter() - t0
            for r in fin:
                done_at[r.rid] = now
        steps += 1
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    total = sum(len(r.out) for r in reqs)
    lat = sorted(done_at.values())
    p = lambda q: lat[min(len(lat) - 1, int(q * len(lat)))]
    print(f"{args.requests} requests x {args.num_tokens} new tokens "
          f"(prompt {args.prompt_len}, {bat.n_slots} slots on "
          f"{len(bat.lanes)} lanes): "
          f"{total} tokens in {dt:.2f}s = {total/dt:.0f} tok/s, "
          f"{steps} decode steps; completion p50={p(0.5):.2f}s "
          f"p95={p(0.95):.2f}s max={lat[-1]:.2f}s")


if __name__ == "__main__":
    main()
