"""Continuous batching across pipeline ranks (multi-GPU serving).

The round-1 serving stack (`ContinuousBatcher` + HTTP front) ran on ONE
engine; the RCCL pipeline ran fixed synthetic batches. This module
composes them: rank 0 runs the UNCHANGED ContinuousBatcher against a
`PipelineEngine` facade whose embed/forward/logits/argmax calls drive
every pipeline stage — per-request prompts prefill through the ranks
(mixed multi-span admission streams ride the native prefill kernels on
each stage), decode steps advance all in-flight requests, logits come
back from the last rank for sampling on rank 0.

Control flow per engine call: rank 0 broadcasts a small command header
(op, T, decode) + the (tokens, pos, seq) arrays; each rank executes its
slice and hands activations to the next over the same P2P transport the
bench pipeline uses (RCCL on GPUs, host-staged gloo on one GPU / CPU).
Follower ranks sit in `serve_forever()` until a shutdown command.
Failure domain: the facade's calls are collective — if rank 0 dies
mid-broadcast, followers block in the next collective until the
launcher (torchrun) tears the job down; per-request errors never reach
the collectives (the batcher validates at submit() and the worker
catches step() failures before any broadcast is cut short).

The reference has no equivalent — its node serves ONE request at a time
over TCP; this is the scheduling layer the north star's
"sliced across N GPUs" serving implies.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

OP_SHUTDOWN = 0
OP_FORWARD = 1   # prefill: forward only, KV append
OP_DECODE = 2    # forward + logits (+ greedy ids) back to rank 0


def _nccl() -> bool:
    return dist.is_initialized() and dist.get_backend() == "nccl"


class _Staged:
    """Host-staged P2P for gloo-with-CUDA (one-GPU shakeout) — same
    trick as DecodePipeline._hop_*."""

    def __init__(self, device: str):
        self.on = (device == "cuda" and dist.is_initialized()
                   and dist.get_backend() == "gloo")

    def send(self, t: torch.Tensor, dst: int) -> None:
        dist.send(t.cpu() if self.on else t, dst=dst)

    def recv(self, t: torch.Tensor, src: int) -> torch.Tensor:
        if not self.on:
            dist.recv(t, src=src)
            return t
        h = torch.empty(t.shape, dtype=t.dtype, device="cpu")
        dist.recv(h, src=src)
        t.copy_(h)
        return t


class PipelineEngine:
    """Rank-0 facade with the single-engine interface the batcher uses.

    embed() records the token ids (the real embedding happens inside
    forward, after the command broadcast, so follower ranks see one
    message per engine call); forward() runs the whole pipeline hop;
    logits()/argmax() return the values the last rank sent back.
    """

    def __init__(self, engine, rank: int, world: int):
        self.engine = engine
        self.rank = rank
        self.world = world
        self.hp = engine.hp
        self.n_ctx = engine.n_ctx
        self.max_batch = engine.max_batch
        self.device = engine.device
        self._staged = _Staged(self.device)
        # control-message device: RCCL collectives move CUDA tensors,
        # gloo moves CPU tensors
        self._cdev = self.device if _nccl() else "cpu"
        self._tokens: Optional[torch.Tensor] = None
        self._lg: Optional[torch.Tensor] = None
        self._ids: Optional[torch.Tensor] = None

    # -------------------------------------------------- engine interface

    def embed(self, tokens: torch.Tensor):
        self._tokens = tokens.to(self.device, torch.int32)
        return self._tokens  # handle; forward() embeds after broadcast

    def forward(self, x, pos, seq, decode: bool = False):
        toks = self._tokens
        assert toks is not None, "forward() requires a preceding embed()"
        self._tokens = None
        T = int(toks.numel())
        if self.world > 1:
            hdr = torch.tensor(
                [OP_DECODE if decode else OP_FORWARD, T],
                dtype=torch.int64, device=self._cdev)
            dist.broadcast(hdr, src=0)
            body = torch.stack([toks.to(self._cdev, torch.int64),
                                pos.to(self._cdev, torch.int64),
                                seq.to(self._cdev, torch.int64)])
            dist.broadcast(body, src=0)
        y = self.engine.forward(
            self.engine.embed(toks),
            pos.to(self.device, torch.int32),
            seq.to(self.device, torch.int32), decode=decode)
        if self.world > 1:
            self._staged.send(y, dst=self.rank + 1)
        if decode:
            if self.world > 1:
                V = self.hp.n_vocab
                self._lg = torch.empty(T, V, dtype=torch.float32,
                                       device=self.device)
                self._staged.recv(self._lg, src=self.world - 1)
                self._ids = torch.empty(T, dtype=torch.int32,
                                        device=self.device)
                self._staged.recv(self._ids, src=self.world - 1)
            else:
                self._lg = self.engine.logits(y, all_logits=True)
                self._ids = self.engine.argmax(self._lg)
        return y

    def logits(self, y, all_logits: bool = True):
        assert self._lg is not None, "logits() only after a decode step"
        return self._lg

    def argmax(self, lg):
        return self._ids

    def shutdown(self) -> None:
        if self.world > 1:
            dist.broadcast(torch.tensor([OP_SHUTDOWN, 0],
                                        dtype=torch.int64,
                                        device=self._cdev), src=0)


def serve_forever(engine, rank: int, world: int) -> None:
    """Follower-rank loop: execute broadcast commands until shutdown.
    The LAST rank computes logits + greedy ids for decode steps and
    sends both to rank 0 (sampling semantics stay on rank 0)."""
    dev = engine.device
    staged = _Staged(dev)
    cdev = dev if _nccl() else "cpu"
    E = engine.hp.n_embd
    while True:
        hdr = torch.zeros(2, dtype=torch.int64, device=cdev)
        dist.broadcast(hdr, src=0)
        op, T = int(hdr[0]), int(hdr[1])
        if op == OP_SHUTDOWN:
            return
        body = torch.zeros(3, T, dtype=torch.int64, device=cdev)
        dist.broadcast(body, src=0)
        pos = body[1].to(dev, torch.int32)
        seq = body[2].to(dev, torch.int32)
        x = torch.empty(T, E, dtype=torch.float32, device=dev)
        staged.recv(x, src=rank - 1)
        y = engine.forward(x, pos, seq, decode=(op == OP_DECODE))
        if rank < world - 1:
            staged.send(y, dst=rank + 1)
        elif op == OP_DECODE:
            lg = engine.logits(y, all_logits=True)
            ids = engine.argmax(lg).to(dev, torch.int32)
            staged.send(lg.float(), dst=0)
            staged.send(ids, dst=0)
