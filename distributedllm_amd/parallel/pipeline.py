"""RCCL pipeline runner: layer-sliced decode across GPUs over xGMI.

The MI355X-native replacement of the reference's activation hop
(client-mediated TCP round-trip per node per token,
/root/reference/distllm/cli_api/common.py:148-154 +
control_center.py:224-244): here each pipeline stage is one process per GPU
and the hop is a direct ``torch.distributed`` P2P send/recv (backend "nccl"
== RCCL on ROCm) of the [mbs, E] activation block between adjacent ranks;
the sampled token ids ride a tiny P2P message from the last stage back to
rank 0. Micro-batches keep every stage busy (the reference pipeline is
fully serialized — SURVEY §2.3).

Also runs on CPU with the gloo backend (world_size > 1) for tests.
"""
from __future__ import annotations

import time
from dataclasses import dataclass
from typing import List, Tuple

import torch
import torch.distributed as dist


def partition_layers(n_layers: int, world: int) -> List[Tuple[int, int]]:
    """Contiguous (first_layer, count) per rank; remainder to early ranks."""
    base = n_layers // world
    rem = n_layers % world
    out = []
    first = 0
    for r in range(world):
        cnt = base + (1 if r < rem else 0)
        out.append((first, cnt))
        first += cnt
    return out


@dataclass
class PipelineConfig:
    mbs: int = 4          # sequences per micro-batch
    n_mb: int = 1         # micro-batches in flight
    device: str = "cuda"

    @property
    def global_batch(self) -> int:
        return self.mbs * self.n_mb


class _StagedWork:
    """irecv handle for a host-staged hop: wait() completes the gloo
    receive then uploads into the device-side target."""

    def __init__(self, req, host: torch.Tensor, dev: torch.Tensor):
        self.req, self.host, self.dev = req, host, dev

    def wait(self) -> None:
        self.req.wait()
        self.dev.copy_(self.host)


class DecodePipeline:
    """Synchronized batched decode across pipeline stages.

    Every sequence advances one token per step. Stage r owns a contiguous
    layer range; rank 0 embeds, the last rank samples (greedy argmax on
    device) and feeds token ids back to rank 0.
    """

    def __init__(self, engine, cfg: PipelineConfig,
                 rank: int = 0, world: int = 1, engines=None):
        """engines: optional list of k <= n_mb weight-sharing engine
        clones ("stream lanes"): micro-batch m runs on engines[m % k]
        and, on CUDA, on that lane's own HIP stream — concurrent lanes
        overlap the latency-bound kernel chains (single- AND multi-GPU;
        micro-batches of one lane share scratch, so a lane is
        stream-serialized by construction)."""
        self.engine = engine
        self.engines = engines  # None => self.engine for every mb
        self.cfg = cfg
        self.rank = rank
        self.world = world
        self.is_first = rank == 0
        self.is_last = rank == world - 1
        dev = cfg.device
        E = engine.hp.n_embd
        M, mbs = cfg.n_mb, cfg.mbs
        self.n_lanes = len(engines) if engines is not None else 1
        self.pos = [torch.zeros(mbs, dtype=torch.int32, device=dev)
                    for _ in range(M)]
        # lane clones have their OWN KV caches: micro-batch m is the
        # (m // k)-th of its lane, so its slot range is local to the
        # clone; a single shared engine partitions slots across all mbs
        if engines is not None:
            k = self.n_lanes
            need = ((M + k - 1) // k) * mbs
            self.seq = [torch.arange((m // k) * mbs, (m // k + 1) * mbs,
                                     dtype=torch.int32, device=dev)
                        for m in range(M)]
        else:
            need = M * mbs
            self.seq = [torch.arange(m * mbs, (m + 1) * mbs,
                                     dtype=torch.int32, device=dev)
                        for m in range(M)]
        # the engine KV cache is indexed by sequence id with no device-
        # side bounds check — catch undersized engines at construction
        for e in (engines if engines is not None else [engine]):
            cap = int(getattr(e, "max_batch", need))
            if cap < need:
                raise ValueError(
                    f"engine KV holds {cap} sequence slots but the "
                    f"pipeline assigns ids [0, {need}) per lane "
                    f"(n_mb={M} x mbs={mbs}, {self.n_lanes} lanes); "
                    f"build the engine with max_batch >= {need}")
        # zero-init: mid-stage graph-capture warmup runs before any real
        # activations arrive; empty-buffer garbage could seed NaNs into
        # the KV cache rows written during warmup
        self.x_recv = [torch.zeros(mbs, E, dtype=torch.float32, device=dev)
                       for _ in range(M)]
        self.tok = [torch.randint(3, engine.hp.n_vocab, (mbs,),
                                  dtype=torch.int32, device=dev)
                    for _ in range(M)]
        self._graphs = None      # per-mb captured hipGraphs
        self._graph_out = None   # per-mb tensors the graph writes
        self._streams = None     # per-mb streams (single-GPU multi-mb)
        self.temperature = 0.0   # 0 = greedy argmax (graph-capturable)
        # Host-staged hops: gloo cannot move CUDA tensors, but a gloo
        # pipeline with CUDA compute is how the multi-rank path is
        # shaken out on a 1-GPU box (RCCL refuses two ranks on one
        # device — "Duplicate GPU detected"). Device tensors bounce
        # through pinned host buffers around each gloo send/recv.
        self._staged = (dev == "cuda" and world > 1 and
                        dist.is_initialized() and
                        dist.get_backend() == "gloo")
        self._hop_bufs = {}

    # -------------------------------------------------- transport (hops)

    def _hopbuf(self, key, like: torch.Tensor) -> torch.Tensor:
        buf = self._hop_bufs.get(key)
        if (buf is None or buf.shape != like.shape or
                buf.dtype != like.dtype):
            buf = torch.empty(like.shape, dtype=like.dtype, device="cpu",
                              pin_memory=True)
            self._hop_bufs[key] = buf
        return buf

    def _hop_send(self, t: torch.Tensor, dst: int, key) -> None:
        if not self._staged:
            dist.send(t, dst=dst)
            return
        h = self._hopbuf(("s",) + key, t)
        h.copy_(t)  # D2H, synchronous w.r.t. host
        dist.send(h, dst=dst)

    def _hop_recv(self, t: torch.Tensor, src: int, key) -> None:
        if not self._staged:
            dist.recv(t, src=src)
            return
        h = self._hopbuf(("r",) + key, t)
        dist.recv(h, src=src)
        t.copy_(h)

    def _hop_isend(self, t: torch.Tensor, dst: int, key):
        if not self._staged:
            return dist.isend(t, dst=dst)
        h = self._hopbuf(("s",) + key, t)
        h.copy_(t)
        return dist.isend(h, dst=dst)

    def _hop_irecv(self, t: torch.Tensor, src: int, key):
        if not self._staged:
            return dist.irecv(t, src=src)
        h = self._hopbuf(("r",) + key, t)
        return _StagedWork(dist.irecv(h, src=src), h, t)

    # ------------------------------------------------------ hipGraph mode

    def _eng(self, m: int):
        if self.engines is None:
            return self.engine
        return self.engines[m % self.n_lanes]

    def _mb_compute(self, m: int):
        """The capturable (comm-free) compute of micro-batch m.

        Returns the tensor the next stage needs: activations for mid
        stages, sampled token ids for the last stage (None for single-GPU,
        where tok[m] is updated in place)."""
        eng = self._eng(m)
        if self.is_first:
            x = eng.embed(self.tok[m])
        else:
            x = self.x_recv[m]
        # every sequence advances one token per step — the batched-decode
        # contract the engine's fused qkv/attention path requires
        y = eng.forward(x, self.pos[m], self.seq[m], decode=True)
        out = y
        if self.is_last:
            lg = eng.logits(y, all_logits=True)
            if self.temperature > 0.0:
                # device-side temperature sampling (eager path only —
                # capture_graphs pins greedy; RNG state is not
                # graph-stable). Repetition-penalty sampling lives in
                # the TCP client path (engine/sampler.py parity).
                probs = torch.softmax(lg.float() / self.temperature,
                                      dim=-1)
                nxt = torch.multinomial(probs, 1)[:, 0].to(torch.int32)
            else:
                nxt = eng.argmax(lg)
            if self.world == 1:
                self.tok[m].copy_(nxt)
                out = None
            else:
                out = nxt
        self.pos[m] += 1
        return out

    def capture_graphs(self, warmup_steps: int = 2) -> None:
        """Capture each micro-batch's compute into a hipGraph (decode
        steady state). Comm (RCCL send/recv) stays eager between replays.
        The warmup steps run for real (they advance positions/KV)."""
        assert self.cfg.device == "cuda"
        assert self.temperature == 0.0, \
            "sampled decode is eager-only (RNG is not graph-stable)"
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup_steps):
                for m in range(self.cfg.n_mb):
                    self._mb_compute(m)
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        self._graphs = []
        self._graph_out = []
        for m in range(self.cfg.n_mb):
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                out = self._mb_compute(m)
            self._graphs.append(g)
            self._graph_out.append(out)

    def _advance_mb(self, m: int) -> None:
        """Push micro-batch m one token forward through this stage
        (blocking comm; the overlapped path lives in run_steps)."""
        if not self.is_first:
            self._hop_recv(self.x_recv[m], self.rank - 1, ("x", m))
        out = self._compute(m)
        if not self.is_last:
            self._hop_send(out, self.rank + 1, ("x", m))
        elif self.world > 1:
            self._hop_send(out, 0, ("t", m))
        if self.is_first and self.world > 1:
            # token ids for this micro-batch's next step come from the
            # last stage; ordered FIFO per rank pair, so recv here pairs
            # with the send above.
            self._hop_recv(self.tok[m], self.world - 1, ("t", m))

    def _compute(self, m: int):
        if self._graphs is not None:
            self._graphs[m].replay()
            return self._graph_out[m]
        return self._mb_compute(m)

    def _post_recv(self, m: int):
        """Post the async receive of micro-batch m's NEXT input: mid/last
        stages receive activations from the previous stage; rank 0 of a
        multi-stage pipeline receives the sampled token ids from the last
        stage."""
        if not self.is_first:
            return self._hop_irecv(self.x_recv[m], self.rank - 1, ("x", m))
        if self.world > 1:
            return self._hop_irecv(self.tok[m], self.world - 1, ("t", m))
        return None

    def run_steps(self, steps: int) -> None:
        """Overlapped pipeline driver (world > 1): every receive is posted
        as soon as the buffer is free, so the activation hop of micro-batch
        m rides xGMI while micro-batch m+1 computes (the reference pipeline
        is fully serialized — SURVEY §2.3; the north-star design point is
        comm/compute overlap across pipeline stages)."""
        n_mb = self.cfg.n_mb
        use_streams = (self.engines is not None and
                       self.cfg.device == "cuda" and self.n_lanes > 1)
        if use_streams and self._streams is None:
            self._streams = [torch.cuda.Stream()
                             for _ in range(self.n_lanes)]
        if self.world == 1 or steps == 0:
            if use_streams:
                # single-GPU multi-stream: each lane's step chain lives
                # on its own stream; lanes are independent (per-lane
                # engines/KV/tokens), so the whole schedule is enqueued
                # with no cross-stream syncs and the streams fill each
                # other's latency stalls.
                for _ in range(steps):
                    for m in range(n_mb):
                        with torch.cuda.stream(
                                self._streams[m % self.n_lanes]):
                            self._compute(m)
                for st in self._streams:
                    torch.cuda.current_stream().wait_stream(st)
                return
            for _ in range(steps):
                for m in range(n_mb):
                    self._advance_mb(m)
            return

        # multi-GPU overlapped driver. With lanes, micro-batch m's
        # compute AND its NCCL ops run under lane m%k's stream (the
        # NCCL work is ordered against the current stream at call
        # time), so k lanes overlap on the GPU while the host-side
        # posting order keeps every (src, dst) send/recv sequence
        # deterministic — torch P2P has no tags, matching is by order.
        from contextlib import nullcontext

        def lane_ctx(m):
            if use_streams:
                return torch.cuda.stream(self._streams[m % self.n_lanes])
            return nullcontext()

        recvs: List = [None] * n_mb
        sends: List = [None] * n_mb
        # prime: mid/last stages post receives for every micro-batch;
        # rank 0 owns the initial tokens, so its first token receive is
        # posted only after it has sent work downstream.
        if not self.is_first:
            for m in range(n_mb):
                with lane_ctx(m):
                    recvs[m] = self._post_recv(m)
        for s in range(steps):
            for m in range(n_mb):
                with lane_ctx(m):
                    if recvs[m] is not None:
                        recvs[m].wait()
                    if sends[m] is not None:
                        sends[m].wait()  # out buffer is being re-written
                    out = self._compute(m)
                    dst = 0 if self.is_last else self.rank + 1
                    sends[m] = self._hop_isend(
                        out, dst, ("t", m) if self.is_last else ("x", m))
                    # mid/last stages receive exactly `steps` activation
                    # blocks — do not post one past the end (it would
                    # never be matched); rank 0 must still drain the
                    # final token sends.
                    if self.is_first or s + 1 < steps:
                        recvs[m] = self._post_recv(m)
                    else:
                        recvs[m] = None
        for m in range(n_mb):
            if sends[m] is not None:
                sends[m].wait()
            # rank 0: drain the final token receives so the communicator
            # is quiescent between run_steps calls
            if recvs[m] is not None:
                recvs[m].wait()
        if use_streams:
            for st in self._streams:
                torch.cuda.current_stream().wait_stream(st)

    def current_tokens(self) -> torch.Tensor:
        return torch.stack(self.tok)

    # ------------------------------------------------------- real serving

    def prime(self, prompt_ids) -> None:
        """Prefill every sequence with the same prompt (the batched-
        serving demo semantics; per-sequence prompts live in
        serving.ContinuousBatcher). Processes prompt[:-1] in <=64-token
        tiles through the pipeline stages eagerly and leaves tok at the
        LAST prompt token with pos = len(prompt)-1, so the first decode
        step evaluates that token at its true position and samples the
        true continuation (the TCP client's `generate` semantics —
        feeding the last token again at a new position would condition
        generation on a duplicated token)."""
        dev = self.cfg.device
        prompt = torch.as_tensor(prompt_ids, dtype=torch.int32, device=dev)
        Tp = int(prompt.numel())
        assert Tp >= 1, "prompt must be non-empty"
        mbs = self.cfg.mbs
        E = self.engine.hp.n_embd
        tile = max(1, 64 // mbs)  # prompt positions per pipeline hop
        for m in range(self.cfg.n_mb):
            # lane engines hold separate KV caches: micro-batch m's
            # prompt KV must land on ITS lane's engine (and lane-local
            # seq ids, self.seq[m]) or decode reads empty rows
            # (ADVICE r1 — prime() used self.engine for every mb)
            eng = self._eng(m)
            for p0 in range(0, Tp - 1, tile):
                p1 = min(Tp - 1, p0 + tile)
                n = p1 - p0
                # token-major layout: [n positions x mbs sequences]
                pos = (torch.arange(p0, p1, dtype=torch.int32, device=dev)
                       .repeat_interleave(mbs))
                seq = self.seq[m].repeat(n)
                if self.is_first:
                    toks = prompt[p0:p1].repeat_interleave(mbs)
                    x = eng.embed(toks)
                else:
                    x = torch.empty(n * mbs, E, dtype=torch.float32,
                                    device=dev)
                    self._hop_recv(x, self.rank - 1, ("pf", m))
                y = eng.forward(x, pos, seq)
                if not self.is_last:
                    self._hop_send(y, self.rank + 1, ("pf", m))
            self.pos[m].fill_(Tp - 1)
            self.tok[m].fill_(int(prompt[-1].item()))


def pipeline_generate(pipe: DecodePipeline, prompt_ids, max_steps: int,
                      greedy: bool = True, temperature: float = 0.0):
    """Prompt-conditioned generation on the RCCL pipeline: prefill then
    decode. Rank 0 returns the generated token ids
    [global_batch, max_steps]; other ranks return None. Greedy device
    argmax by default (reference `sample_next_token` semantics);
    temperature > 0 switches the last rank to on-device softmax
    sampling (repetition-penalty sampling lives in the TCP client path,
    engine/sampler.py)."""
    assert greedy or temperature > 0.0, \
        "non-greedy decode needs temperature > 0"
    n_prompt = len(prompt_ids)
    n_ctx = int(getattr(pipe.engine, "n_ctx", n_prompt + max_steps))
    if n_prompt + max_steps > n_ctx:
        raise ValueError(
            f"prompt ({n_prompt}) + max_steps ({max_steps}) exceeds the "
            f"engine context {n_ctx} — KV positions would go out of range")
    pipe.temperature = float(temperature)  # > 0 wins over greedy
    pipe.prime(prompt_ids)
    out = []
    for _ in range(max_steps):
        pipe.run_steps(1)
        if pipe.is_first:
            out.append(pipe.current_tokens().clone())
    if pipe.is_first:
        # [steps][n_mb][mbs] -> [n_mb*mbs, steps]
        t = torch.stack(out)
        return t.permute(1, 2, 0).reshape(-1, len(out))
    return None


def timed_decode(pipe: DecodePipeline, steps: int, warmup: int,
                 device: str, use_graphs: bool = True) -> float:
    """Barrier-bracketed timing of exactly `steps` steps; returns seconds
    (this rank's wall time — reduce MAX across ranks for the job time)."""
    if use_graphs and device == "cuda":
        pipe.capture_graphs(warmup_steps=2)
    pipe.run_steps(warmup)
    if dist.is_initialized():
        dist.barrier()
    if device == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    pipe.run_steps(steps)
    if dist.is_initialized():
        dist.barrier()
    if device == "cuda":
        torch.cuda.synchronize()
    return time.perf_counter() - t0
