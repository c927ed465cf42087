"""All CLI commands.

Command-for-command parity with the reference CLI
(/root/reference/distllm/cli_api/__init__.py:9-24 and the per-command
files): provision, run_node, status, push_slice, load_slice, list_slices,
generate_text, perplexity, run_proxy.
"""
from __future__ import annotations

import argparse
import json
import sys
import threading

from .base import Command


def _random_dataset_prompt(path: str, seed=None) -> str:
    """Random prompt from a local HF dataset directory (the reference
    samples from a hub dataset, perplexity.py:35-51; this environment has
    no network, so the dataset must already be on disk)."""
    import random
    from datasets import load_from_disk
    ds = load_from_disk(path)
    if hasattr(ds, "keys") and not hasattr(ds, "features"):  # DatasetDict
        ds = ds[sorted(ds.keys())[0]]
    rng = random.Random(seed)
    row = ds[rng.randrange(len(ds))]
    for key in ("text", "content", "sentence", "document"):
        if key in row and isinstance(row[key], str) and row[key].strip():
            return row[key]
    raise ValueError(f"no text-like column found in dataset at {path}")


def _progress(label: str):
    state = {"last": -1.0}

    def cb(sent: int, total: int) -> None:
        pct = 100.0 * sent / max(total, 1)
        if pct - state["last"] < 1.0 and sent < total:
            return  # throttle to ~1% steps (piped output stays readable)
        state["last"] = pct
        print(f"\r[{label}] {sent}/{total} bytes ({pct:5.1f}%)",
              end="", flush=True)
        if sent >= total:
            print()
    return cb


class ProvisionCommand(Command):
    name = "provision"
    help = ("Prepare a model (convert/requantize), slice it per the "
            "config's nodes_map, record it in the registry, push slices "
            "to the nodes")

    def configure(self, p: argparse.ArgumentParser) -> None:
        p.add_argument("config", help="cluster+model JSON config")
        p.add_argument("--root", default=".",
                       help="working root (models/ + models_registry/)")
        p.add_argument("--no-push", action="store_true",
                       help="slice + register only, do not upload to nodes")

    def __call__(self, args) -> int:
        from ..cluster.provision import provision
        entry = provision(args.config, root=args.root, push=not args.no_push,
                          progress=_progress("push"))
        print(f"provisioned {entry.model_id}: "
              f"{len(entry.slices)} slice(s) in {entry.model_dir}")
        return 0


class RunNodeCommand(Command):
    name = "run_node"
    help = "Start a compute node (TCP server, or reverse-connect to a proxy)"

    def configure(self, p: argparse.ArgumentParser) -> None:
        p.add_argument("--host", default="0.0.0.0")
        p.add_argument("--port", type=int, default=9998)
        p.add_argument("--uploads_dir", default="uploads")
        p.add_argument("--device", default=None,
                       help="force engine device (cuda/cpu); default auto")
        p.add_argument("--n-ctx", type=int, default=2048)
        p.add_argument("--reverse", action="store_true",
                       help="dial out to a proxy instead of listening")
        p.add_argument("--proxy-host", default="127.0.0.1")
        p.add_argument("--proxy-port", type=int, default=9999)
        p.add_argument("--name", default="node")

    def __call__(self, args) -> int:
        from ..cluster.node import NodeState, run_server
        if args.reverse:
            from ..cluster.proxy import connect_then_serve
            state = NodeState(args.uploads_dir, device=args.device,
                              n_ctx=args.n_ctx)
            print(f"[node] reverse-connecting to "
                  f"{args.proxy_host}:{args.proxy_port}")
            connect_then_serve(args.proxy_host, args.proxy_port, state,
                               name=args.name)
            return 0
        run_server(args.host, args.port, args.uploads_dir,
                   device=args.device, n_ctx=args.n_ctx)
        return 0


class _NodeCommand(Command):
    """Base for commands addressing one node as host:port."""

    def configure(self, p: argparse.ArgumentParser) -> None:
        p.add_argument("address", help="node address host:port")

    def _conn(self, args):
        from ..cluster.client import Connection, parse_address
        host, port = parse_address(args.address)
        return Connection(host, port)


class StatusCommand(Command):
    name = "status"
    help = ("Report a node's status (host:port), or a whole cluster's "
            "pipeline readiness (--config)")

    def configure(self, p: argparse.ArgumentParser) -> None:
        p.add_argument("address", nargs="?", default=None,
                       help="node address host:port")
        p.add_argument("--config", default=None,
                       help="cluster config: poll every nodes_map node")
        p.add_argument("--n-layer", type=int, default=None,
                       help="with --config: also check the loaded slices "
                            "tile this many layers")

    def __call__(self, args) -> int:
        if (args.address is None) == (args.config is None):
            print("provide exactly one of <address> / --config",
                  file=sys.stderr)
            return 2
        if args.config:
            from ..cluster.control import ControlCenter
            with open(args.config) as f:
                nodes_map = json.load(f)["nodes_map"]
            cc = ControlCenter(nodes_map)
            status = cc.get_status()
            for addr, st in sorted(status.items()):
                print(json.dumps({"node": addr,
                                  "connectivity": st.connectivity,
                                  "model": st.model,
                                  "first_layer": st.first_layer,
                                  "n_layers": st.n_layers,
                                  "device": st.device,
                                  "error": st.error}))
            if args.n_layer:
                ready, why = cc.pipeline_ready(args.n_layer)
                print(f"pipeline_ready: {ready} ({why})")
                return 0 if ready else 1
            return 0
        from ..cluster.client import Connection, parse_address
        host, port = parse_address(args.address)
        conn = Connection(host, port)
        s = conn.get_status()
        print(json.dumps({"status": s.status, "model": s.model,
                          "first_layer": s.first_layer,
                          "n_layers": s.n_layers, "device": s.device},
                         indent=1))
        conn.close()
        return 0


class PushSliceCommand(_NodeCommand):
    name = "push_slice"
    help = "Upload a slice file to a node (chunked, checksummed)"

    def configure(self, p: argparse.ArgumentParser) -> None:
        super().configure(p)
        p.add_argument("path", help="slice file to upload")
        p.add_argument("--metadata", default="{}",
                       help="JSON metadata (e.g. '{\"format\": \"test\"}')")

    def __call__(self, args) -> int:
        conn = self._conn(args)
        resp = conn.push_slice(args.path, json.loads(args.metadata),
                               progress=_progress("push"))
        print(f"uploaded {resp.name} ({resp.total_size} bytes)")
        conn.close()
        return 0


class LoadSliceCommand(_NodeCommand):
    name = "load_slice"
    help = "Load an uploaded slice into the node's engine"

    def configure(self, p: argparse.ArgumentParser) -> None:
        super().configure(p)
        p.add_argument("slice_name")

    def __call__(self, args) -> int:
        conn = self._conn(args)
        resp = conn.load_slice(args.slice_name)
        print(f"loaded {resp.name}: layers "
              f"[{resp.first_layer}, {resp.first_layer + resp.n_layers - 1}]")
        conn.close()
        return 0


class ListSlicesCommand(_NodeCommand):
    name = "list_slices"
    help = "List slices uploaded to a node"

    def __call__(self, args) -> int:
        conn = self._conn(args)
        for s in conn.list_slices():
            print(json.dumps(s))
        conn.close()
        return 0


class GenerateTextCommand(Command):
    name = "generate_text"
    help = "Generate text with a provisioned distributed model"

    def configure(self, p: argparse.ArgumentParser) -> None:
        p.add_argument("config", help="cluster+model JSON config")
        p.add_argument("--prompt", required=True)
        p.add_argument("--num-tokens", type=int, default=50)
        p.add_argument("--temp", type=float, default=0.7)
        p.add_argument("--rp", type=float, default=1.1,
                       help="repetition penalty")
        p.add_argument("--greedy", action="store_true")
        p.add_argument("--top-k", type=int, default=0,
                       help="sample from the k most likely tokens (0=off)")
        p.add_argument("--top-p", type=float, default=1.0,
                       help="nucleus sampling mass (1.0=off)")
        p.add_argument("--seed", type=int, default=None)
        p.add_argument("--speculate", type=int, default=0, metavar="K",
                       help="with --greedy: verify up to K prompt-lookup "
                            "draft tokens per pipeline hop (token-exact; "
                            "amortizes the per-token TCP round-trip)")
        p.add_argument("--root", default=".")

    def __call__(self, args) -> int:
        from ..cluster.llm_client import get_llm
        llm = get_llm(args.config, root=args.root)
        print(args.prompt, end="", flush=True)
        for piece in llm.generate(args.prompt, max_steps=args.num_tokens,
                                  temperature=args.temp,
                                  repeat_penalty=args.rp,
                                  greedy=args.greedy, seed=args.seed,
                                  top_k=args.top_k, top_p=args.top_p,
                                  speculative=getattr(args, "speculate",
                                                      0)):
            print(piece, end="", flush=True)
        print()
        r = llm.throughput.report()
        print(f"[{r['count']:.0f} tokens in {r['seconds']:.2f}s = "
              f"{r['per_second']:.2f} tok/s]", file=sys.stderr)
        return 0


class PerplexityCommand(Command):
    name = "perplexity"
    help = "Perplexity of a text under a provisioned distributed model"

    def configure(self, p: argparse.ArgumentParser) -> None:
        p.add_argument("config")
        p.add_argument("--prompt", default=None)
        p.add_argument("--file", default=None,
                       help="read the text from a file instead")
        p.add_argument("--dataset", default=None,
                       help="local HF dataset directory (datasets.load_"
                            "from_disk); a random text row is sampled, "
                            "like the reference's --dataset")
        p.add_argument("--seed", type=int, default=None)
        p.add_argument("--root", default=".")

    def __call__(self, args) -> int:
        sources = [x for x in (args.prompt, args.file, args.dataset)
                   if x is not None]
        if len(sources) != 1:
            print("provide exactly one of --prompt / --file / --dataset",
                  file=sys.stderr)
            return 2
        text = args.prompt
        if args.file:
            with open(args.file) as f:
                text = f.read()
        if args.dataset:
            text = _random_dataset_prompt(args.dataset, args.seed)
        if args.config.endswith(".bin"):
            # local mode: evaluate directly on a GGML model file (no
            # cluster needed; same NLL math as the distributed client)
            ppl = _local_perplexity(args.config, text)
        else:
            from ..cluster.llm_client import get_llm
            llm = get_llm(args.config, root=args.root)
            ppl = llm.perplexity(text)
        print(f"perplexity: {ppl:.4f}")
        return 0


class BatchGenerateCommand(Command):
    name = "batch_generate"
    help = ("Serve many prompts concurrently on a LOCAL model file via "
            "continuous batching (shared decode steps, KV-slot reuse)")

    def configure(self, p: argparse.ArgumentParser) -> None:
        p.add_argument("model", help="GGML model file (full, not a slice)")
        p.add_argument("--prompt", action="append", default=None,
                       help="repeatable; or use --prompts-file")
        p.add_argument("--prompts-file", default=None,
                       help="one prompt per line")
        p.add_argument("--num-tokens", type=int, default=50)
        p.add_argument("--slots", type=int, default=None,
                       help="max concurrent requests (default: engine "
                            "max_batch)")
        p.add_argument("--ctx", type=int, default=2048)
        p.add_argument("--temp", type=float, default=0.7)
        p.add_argument("--rp", type=float, default=1.1)
        p.add_argument("--greedy", action="store_true")
        p.add_argument("--top-k", type=int, default=0)
        p.add_argument("--top-p", type=float, default=1.0)
        p.add_argument("--seed", type=int, default=None)
        p.add_argument("--speculative", action="store_true",
                       help="prompt-lookup speculative decoding (greedy "
                            "only): multiple tokens per forward when "
                            "the continuation repeats earlier text; "
                            "token-exact. One prompt uses the single-"
                            "stream fast path, several enable in-"
                            "batcher speculation (spec_k=8)")

    def __call__(self, args) -> int:
        import time

        from ..engine import engine_for_slice
        from ..engine.sampler import Sampler
        from ..engine.tokenizer import Tokenizer
        from ..formats import ggml, slicer
        from ..serving import ContinuousBatcher

        prompts = list(args.prompt or [])
        if args.prompts_file:
            with open(args.prompts_file) as f:
                prompts += [ln.rstrip("\n") for ln in f if ln.strip()]
        if not prompts:
            print("no prompts (use --prompt/--prompts-file)",
                  file=sys.stderr)
            return 2

        f = ggml.GGMLFile.load(args.model,
                               extended=ggml.sniff_extended(args.model))
        n_slots = min(len(prompts), args.slots or 64, 64)
        eng = engine_for_slice(f, n_ctx=args.ctx, max_batch=n_slots)
        eng.attach_extra(slicer.make_extra_layers(f))
        tok = Tokenizer(f.vocab)

        if args.speculative:
            if not args.greedy:
                print("--speculative needs --greedy", file=sys.stderr)
                return 2
        if args.speculative and len(prompts) == 1:
            from ..serving.speculative import SpecStats, pld_generate
            st = SpecStats()
            t0 = time.perf_counter()
            out = pld_generate(eng, tok.encode(prompts[0], bos=True),
                               args.num_tokens, stats=st)
            dt = time.perf_counter() - t0
            print(f"[0] {prompts[0]!r} -> {tok.decode(out)!r}")
            print(f"[{len(out)} tokens in {dt:.2f}s = "
                  f"{len(out) / max(dt, 1e-9):.1f} tok/s, "
                  f"{st.forwards} forwards = "
                  f"{len(out) / max(st.forwards, 1):.2f} tok/forward]",
                  file=sys.stderr)
            return 0

        spec = bool(args.speculative)
        bat = ContinuousBatcher(eng, max_slots=n_slots,
                                spec_ngram=3 if spec else 0,
                                spec_k=8 if spec else 0)
        reqs = []
        for text in prompts:
            sampler = None if args.greedy else \
                Sampler(args.temp, args.rp, seed=args.seed,
                        top_k=args.top_k, top_p=args.top_p)
            reqs.append((text, bat.submit(tok.encode(text, bos=True),
                                          args.num_tokens,
                                          sampler=sampler)))
        t0 = time.perf_counter()
        steps = 0
        while bat.pending:
            bat.step()
            steps += 1
        dt = time.perf_counter() - t0
        total = sum(len(r.out) for _, r in reqs)
        for text, r in reqs:
            print(f"[{r.rid}] {text!r} -> {tok.decode(r.out)!r}")
        print(f"[{len(reqs)} requests, {total} tokens in {dt:.2f}s = "
              f"{total / max(dt, 1e-9):.1f} tok/s, {steps} decode steps]",
              file=sys.stderr)
        return 0


class ServeHttpCommand(Command):
    name = "serve_http"
    help = ("HTTP serving front (FastAPI/uvicorn) over the continuous "
            "batcher: POST /generate, GET /health")

    def configure(self, p: argparse.ArgumentParser) -> None:
        p.add_argument("model", help="GGML model file (full, not a slice)")
        p.add_argument("--host", default="127.0.0.1")
        p.add_argument("--port", type=int, default=8080)
        p.add_argument("--slots", type=int, default=64)
        p.add_argument("--lanes", type=int, default=1,
                       help="stream lanes (weight-sharing clones; GPU). "
                            "1 (default) serves all slots in ONE wide "
                            "decode batch — weights/dequant paid once "
                            "per step (26.2k vs 18.4k tok/s measured at "
                            "256 requests on 3B); 0 = auto lane fit "
                            "(the round-1 multi-stream mode)")
        p.add_argument("--ctx", type=int, default=2048)
        p.add_argument("--prefill-chunk", type=int, default=64,
                       help="max prompt tokens prefilled per lane per "
                            "decode step (latency fairness; 0 = "
                            "unbounded)")
        p.add_argument("--speculate", type=int, default=0, metavar="K",
                       help="prompt-lookup speculation inside the "
                            "shared decode step: up to K draft tokens "
                            "per greedy request verified per forward "
                            "(token-exact; 0 = off)")
        p.add_argument("--pipeline", action="store_true",
                       help="serve across torch.distributed pipeline "
                            "ranks (launch with torchrun, one rank per "
                            "GPU; rank 0 runs the HTTP front + batcher, "
                            "followers execute their layer slice)")
        p.add_argument("--backend", default="auto",
                       choices=["auto", "nccl", "gloo"],
                       help="pipeline backend: auto = nccl(RCCL) on "
                            "GPU; gloo supports CPU and one-GPU "
                            "multi-rank shakeout")

    def __call__(self, args) -> int:
        import torch
        import uvicorn

        from ..engine import engine_for_slice
        from ..engine.tokenizer import Tokenizer
        from ..formats import ggml, slicer
        from ..serving import ContinuousBatcher, build_http_app

        if args.pipeline:
            return self._pipeline_main(args)
        f = ggml.GGMLFile.load(args.model,
                               extended=ggml.sniff_extended(args.model))
        n_lanes = args.lanes
        if n_lanes == 0:
            n_lanes = 1
            if torch.cuda.is_available():
                hp = f.hparams
                kv = hp.n_layer * args.slots * args.ctx * hp.n_embd * 4
                w_bytes = sum(len(t.raw) for t in f.tensors) * 2.2
                fit = max(1, int((260e9 - w_bytes) // max(kv, 1)))
                n_lanes = min(5, fit) if fit >= 3 else 1
        eng = engine_for_slice(f, n_ctx=args.ctx, max_batch=args.slots)
        eng.attach_extra(slicer.make_extra_layers(f))
        lanes = None
        if n_lanes > 1 and hasattr(eng, "clone_shared"):
            lanes = [eng] + [eng.clone_shared()
                             for _ in range(n_lanes - 1)]
        tok = Tokenizer(f.vocab)
        bat = ContinuousBatcher(eng, engines=lanes,
                                prefill_chunk=args.prefill_chunk or None,
                                spec_ngram=3 if args.speculate else 0,
                                spec_k=args.speculate)
        app, worker = build_http_app(bat, tok)
        try:
            uvicorn.run(app, host=args.host, port=args.port,
                        log_level="warning")
        finally:
            worker.stop()
        return 0

    def _pipeline_main(self, args) -> int:
        """Multi-GPU serving: rank 0 = HTTP + batcher over the
        PipelineEngine facade; follower ranks execute their slice."""
        import os

        import torch
        import torch.distributed as dist
        import uvicorn

        from ..engine import engine_for_slice
        from ..engine.tokenizer import Tokenizer
        from ..formats import ggml, slicer
        from ..parallel.pipeline import partition_layers
        from ..serving import ContinuousBatcher, build_http_app
        from ..serving.pipeline_server import PipelineEngine, serve_forever

        world = int(os.environ.get("WORLD_SIZE", "1"))
        rank = int(os.environ.get("RANK", "0"))
        device = "cuda" if torch.cuda.is_available() else "cpu"
        backend = ("nccl" if device == "cuda" else "gloo") \
            if args.backend == "auto" else args.backend
        if world > 1:
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            dist.init_process_group(backend, rank=rank, world_size=world)
            if device == "cuda":
                torch.cuda.set_device(
                    int(os.environ.get("LOCAL_RANK", "0"))
                    % torch.cuda.device_count())
        f = ggml.GGMLFile.load(args.model,
                               extended=ggml.sniff_extended(args.model))
        parts = partition_layers(f.hparams.n_layer, world)
        first, count = parts[rank]
        sl = slicer.make_slice(f, first, first + count - 1) \
            if world > 1 else f
        ex = slicer.make_extra_layers(f)
        eng = engine_for_slice(sl, n_ctx=args.ctx, max_batch=args.slots)
        eng.attach_extra(ex)  # rank 0: embed; last rank: lm head
        if rank > 0:
            serve_forever(eng, rank, world)
            dist.destroy_process_group()
            return 0
        facade = PipelineEngine(eng, rank, world) if world > 1 else eng
        tok = Tokenizer(f.vocab)
        if getattr(args, "speculate", 0) and world > 1:
            # the pipeline facade returns logits only on decode steps;
            # draft verification needs logits for non-decode rows
            print("[serve] --speculate is single-engine only; "
                  "ignored under --pipeline", file=sys.stderr)
        spec = getattr(args, "speculate", 0) if world == 1 else 0
        bat = ContinuousBatcher(facade,
                                prefill_chunk=args.prefill_chunk or None,
                                spec_ngram=3 if spec else 0, spec_k=spec)
        app, worker = build_http_app(bat, tok)
        try:
            uvicorn.run(app, host=args.host, port=args.port,
                        log_level="warning")
        finally:
            worker.stop()
            if world > 1:
                facade.shutdown()
                dist.destroy_process_group()
        return 0


def _local_perplexity(model_path: str, text: str) -> float:
    """exp(mean NLL) on a local engine (reference semantics,
    common.py:113-141 — identical math to DistributedLLM.perplexity)."""
    import numpy as np
    import torch

    from ..engine import engine_for_slice
    from ..engine.tokenizer import Tokenizer
    from ..formats import ggml, slicer

    f = ggml.GGMLFile.load(model_path,
                           extended=ggml.sniff_extended(model_path))
    tok = Tokenizer(f.vocab)
    tokens = tok.encode(text, bos=True)
    if len(tokens) < 2:
        raise ValueError("perplexity needs at least 2 tokens")
    eng = engine_for_slice(f, n_ctx=max(len(tokens) + 1, 16), max_batch=1)
    eng.attach_extra(slicer.make_extra_layers(f))
    dev = getattr(eng, "device", "cpu")
    ids = torch.tensor(tokens[:-1], dtype=torch.int32, device=dev)
    pos = torch.arange(len(tokens) - 1, dtype=torch.int32, device=dev)
    seq = torch.zeros(len(tokens) - 1, dtype=torch.int32, device=dev)
    y = eng.forward(eng.embed(ids), pos, seq)
    lg = eng.logits(y, all_logits=True)
    logp = torch.log_softmax(lg.float().cpu(), dim=-1).numpy()
    nll = [-logp[i, tokens[i + 1]] for i in range(len(tokens) - 1)]
    return float(np.exp(np.mean(nll)))


class RunProxyCommand(Command):
    name = "run_proxy"
    help = "Run the NAT-traversal proxy (bridges clients to a reverse node)"

    def configure(self, p: argparse.ArgumentParser) -> None:
        p.add_argument("--host", default="0.0.0.0")
        p.add_argument("--client-port", type=int, default=9997)
        p.add_argument("--node-port", type=int, default=9999)

    def __call__(self, args) -> int:
        from ..cluster.proxy import run_proxy
        run_proxy(args.host, args.client_port, args.node_port)
        threading.Event().wait()  # serve until killed
        return 0
