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
    ap.add_argument("--lanes", type=int, default=1,
                    help="concurrent stream lanes (weight-sharing "
                         "clones); total slots = slots x lanes")
    ap.add_argument("--ctx", type=int, default=2048)
    ap.add_argument("--prefill-chunk", type=int, default=0,
                    help="chunked-prefill bound (0 = unbounded)")
    ap.add_argument("--long-requests", type=int, default=0,
                    help="long-prompt requests arriving mid-run (QoS "
                         "experiment: how much do they stall in-flight "
                         "decodes?)")
    ap.add_argument("--long-prompt", type=int, default=512)
    ap.add_argument("--speculate", type=int, default=0,
                    help="in-batcher prompt-lookup speculation (greedy "
                         "requests, K drafts; token-exact). NOTE: "
                         "random-init synthetic text repeats "
                         "unrealistically often — acceptance rates here "
                         "overstate real-text gains")
    args = ap.parse_args()

    hp = PRESETS[args.model].hparams(ggml.FTYPE_MOSTLY_Q4_0)
    eng = HIPSliceEngine.random(hp, n_layers=hp.n_layer, n_ctx=args.ctx,
                                max_batch=args.slots, with_extra=True)
    lanes = None
    if args.lanes > 1:
        lanes = [eng] + [eng.clone_shared()
                         for _ in range(args.lanes - 1)]
    g = torch.Generator().manual_seed(1)
    bat = ContinuousBatcher(eng, engines=lanes,
                            prefill_chunk=args.prefill_chunk or None,
                            spec_ngram=3 if args.speculate else 0,
                            spec_k=args.speculate)
    reqs = [bat.submit(torch.randint(3, hp.n_vocab, (args.prompt_len,),
                                     generator=g).tolist(),
                       args.num_tokens)
            for _ in range(args.requests)]
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    steps = 0
    done_at = {}
    long_at = max(1, args.num_tokens // 4)
    stalls = []  # per-step wall time after the long arrivals
    while bat.pending:
        ts = time.perf_counter()
        fin = bat.step()
        if fin:
            torch.cuda.synchronize()
            now = time.perf_counter() - t0
            for r in fin:
                done_at[r.rid] = now
        steps += 1
        if args.long_requests and steps >= long_at:
            torch.cuda.synchronize()
            stalls.append(time.perf_counter() - ts)
        if args.long_requests and steps == long_at:
            for _ in range(args.long_requests):
                reqs.append(bat.submit(
                    torch.randint(3, hp.n_vocab, (args.long_prompt,),
                                  generator=g).tolist(), args.num_tokens))
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
    if stalls:
        ss = sorted(stalls)
        print(f"per-step wall after long arrivals (stall = decode "
              f"starvation): p50={ss[len(ss)//2]*1e3:.0f}ms "
              f"max={ss[-1]*1e3:.0f}ms over {len(ss)} steps "
              f"(chunk={args.prefill_chunk or 'unbounded'})")


if __name__ == "__main__":
    main()
