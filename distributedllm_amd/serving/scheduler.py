"""Continuous batching over one slice engine.

The reference serves exactly one request at a time (its `generate` loop
holds the whole cluster, /root/reference/distllm/cli_api/common.py:94-111,
and node forwards are serialized). This layer is what the MI355X engine's
batched-decode design (JT column tiles, per-sequence KV slots, explicit
pos/seq on every forward — engine_ext.cpp) was built for: many requests
with different prompts and lengths share one decode step, new requests
are admitted into free KV slots the moment one finishes, and every
decode step advances every active request by one token.

Semantics per request match the TCP client's `generate`
(cluster/llm_client.py): prefill `prompt[:-1]`, then decode the last
prompt token at its true position and sample the continuation — greedy
through the engine's device argmax, or per-request host `Sampler`
(reference parity, including repetition penalty over that request's
sampled ids only).
"""
from __future__ import annotations

from collections import deque
from contextlib import nullcontext as _nullctx
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence

import torch

from ..engine.sampler import Sampler
from .speculative import lookup_draft


@dataclass
class Request:
    rid: int
    prompt: List[int]
    max_new: int
    sampler: Optional[Sampler] = None  # None => greedy device argmax
    eos_id: Optional[int] = None
    out: List[int] = field(default_factory=list)
    done: bool = False
    slot: int = -1        # KV slot while active
    _next_tok: int = -1   # token to feed at _pos on the next step
    _pos: int = -1
    _pf_pos: int = 0      # prompt tokens already prefilled (chunked)


class ContinuousBatcher:
    """Slot-based scheduler: submit() any time, step() advances every
    active request one token and admits queued requests into free slots.

    A finished request's KV slot is reused immediately — safe because
    the engine's attention streams exactly rows [0, pos) of a slot, so
    a new sequence starting at pos 0 never sees the old one's rows.
    """

    def __init__(self, engine, max_slots: Optional[int] = None,
                 eos_id: Optional[int] = None, engines=None,
                 prefill_chunk: Optional[int] = None,
                 spec_ngram: int = 0, spec_k: int = 0):
        """engines: optional list of k weight-sharing clones ("lanes");
        global slot s lives on lane s%k as that clone's local slot s//k,
        and on CUDA each lane's forward runs on its own HIP stream —
        the decode chain is latency-bound, so concurrent lanes overlap
        (same mechanics as the pipeline's stream lanes).

        prefill_chunk: max prompt tokens prefilled per lane per step
        (None = unbounded, maximizes throughput). A bound interleaves
        long-prompt admission with decode steps, so in-flight requests
        keep producing tokens while a long prompt loads (latency
        fairness — Sarathi-style chunked prefill).

        spec_ngram/spec_k: optional prompt-lookup speculation INSIDE the
        shared decode step (both > 0 to enable). Greedy requests whose
        trailing spec_ngram tokens recur earlier in their sequence get
        up to spec_k draft tokens verified in the same forward — the
        drafts are just more (token, pos, seq) rows in the mixed stream
        the engine already tiles (serving/speculative.py has the
        single-stream form and the KV-staleness argument; per-slot
        positions stay strictly sequential, so the same exactness
        guarantee holds per request). Sampled requests always advance
        one token. Token-exact with spec off — asserted in
        tests/test_serving.py."""
        self.engine = engine
        self.lanes = engines if engines else [engine]
        k = len(self.lanes)
        per_lane = min(int(getattr(e, "max_batch", 1))
                       for e in self.lanes)
        slots = per_lane * k
        if max_slots is not None:
            slots = min(slots, max_slots)
        self.n_slots = max(1, slots)
        self.eos_id = eos_id
        self.free: List[int] = list(range(self.n_slots))
        self.active: Dict[int, Request] = {}   # slot -> request
        self.queue: deque[Request] = deque()
        self._next_rid = 0
        self._dev = getattr(engine, "device", "cpu")
        self.prefill_chunk = prefill_chunk
        self.spec_ngram = spec_ngram
        self.spec_k = spec_k
        self.prefilling: Dict[int, Request] = {}  # slot -> request
        self.steps_run = 0      # decode steps executed
        self.tokens_out = 0     # tokens emitted (== steps when spec off)
        self._streams = None
        if k > 1 and self._dev == "cuda":
            self._streams = [torch.cuda.Stream() for _ in self.lanes]

    def _lane(self, slot: int):
        k = len(self.lanes)
        return self.lanes[slot % k], slot // k

    # ------------------------------------------------------------- intake

    def submit(self, prompt_ids: Sequence[int], max_new: int,
               sampler: Optional[Sampler] = None,
               eos_id: Optional[int] = None) -> Request:
        """Raises ValueError (before any slot is taken) on an invalid or
        oversized request — the caller can reject it (HTTP 422) without
        the decode loop ever seeing it."""
        if len(prompt_ids) < 1:
            raise ValueError("prompt must contain at least one token")
        if max_new < 1:
            raise ValueError("max_new must be >= 1")
        n_ctx = getattr(self.engine, "n_ctx", None)
        if n_ctx is not None and len(prompt_ids) + max_new > n_ctx:
            raise ValueError(
                f"prompt ({len(prompt_ids)}) + max_new ({max_new}) "
                f"exceeds n_ctx={n_ctx}")
        r = Request(rid=self._next_rid, prompt=list(map(int, prompt_ids)),
                    max_new=max_new, sampler=sampler,
                    eos_id=self.eos_id if eos_id is None else eos_id)
        self._next_rid += 1
        self.queue.append(r)
        return r

    def _admit(self) -> None:
        eng = self.engine  # n_ctx/bounds source; lanes share hparams
        n_ctx = getattr(eng, "n_ctx", 1 << 30)
        k = len(self.lanes)
        # take queued requests into free slots; their prompt body
        # prefills below (possibly across several steps when chunked)
        while self.free and self.queue:
            r = self.queue.popleft()
            # submit() already validated against n_ctx; this is a cheap
            # backstop against an engine swap shrinking n_ctx after
            # submission — drop the request rather than corrupt the KV
            if len(r.prompt) + r.max_new > n_ctx:
                r.done = True
                continue
            r.slot = self.free.pop()
            r._pf_pos = 0
            if len(r.prompt) > 1:
                self.prefilling[r.slot] = r
            else:
                self._activate(r)
        # prefill as ONE concatenated (token, pos, seq) stream per lane
        # — the engine tiles any mixed-sequence stream internally, so
        # several prompts share each prefill launch; with prefill_chunk
        # set, each lane advances at most that many prompt tokens per
        # step and decode of OTHER requests interleaves
        budget_full = self.prefill_chunk
        per_lane: List[List[Request]] = [[] for _ in range(k)]
        for slot in sorted(self.prefilling):
            per_lane[slot % k].append(self.prefilling[slot])
        for j, reqs in enumerate(per_lane):
            if not reqs:
                continue
            budget = budget_full
            toks, pos, seq = [], [], []
            for r in reqs:
                end = len(r.prompt) - 1
                take = end - r._pf_pos
                if budget is not None:
                    take = min(take, budget)
                    budget -= take
                if take <= 0:
                    continue
                toks += r.prompt[r._pf_pos:r._pf_pos + take]
                pos += list(range(r._pf_pos, r._pf_pos + take))
                seq += [r.slot // k] * take
                r._pf_pos += take
            if not toks:
                continue
            lane_eng = self.lanes[j]
            # prefill on the lane's stream: the lane's next decode
            # launch must observe these KV writes, and same-stream
            # ordering gives that without a sync
            ctx = (torch.cuda.stream(self._streams[j])
                   if self._streams is not None else _nullctx())
            with ctx:
                t = torch.tensor(toks, dtype=torch.int32,
                                 device=self._dev)
                p = torch.tensor(pos, dtype=torch.int32,
                                 device=self._dev)
                q = torch.tensor(seq, dtype=torch.int32,
                                 device=self._dev)
                lane_eng.forward(lane_eng.embed(t), p, q)
        for slot in list(self.prefilling):
            r = self.prefilling[slot]
            if r._pf_pos >= len(r.prompt) - 1:
                del self.prefilling[slot]
                self._activate(r)

    def _activate(self, r: Request) -> None:
        r._next_tok = r.prompt[-1]
        r._pos = len(r.prompt) - 1
        self.active[r.slot] = r

    # --------------------------------------------------------------- step

    def step(self) -> List[Request]:
        """Admit what fits, advance every active request one token;
        returns the requests that finished on this step."""
        self._admit()
        if not self.active:
            return []
        self.steps_run += 1
        k = len(self.lanes)
        # one decode launch per lane, each on its own stream (CUDA)
        per_lane: List[List[int]] = [[] for _ in range(k)]
        for s in sorted(self.active):
            per_lane[s % k].append(s)
        work = []  # (reqs, drafts, greedy_ids, lg) per lane
        for j, lane_slots in enumerate(per_lane):
            if not lane_slots:
                continue
            reqs = [self.active[s] for s in lane_slots]
            drafts = [self._draft(r) for r in reqs]
            eng = self.lanes[j]
            ctx = (torch.cuda.stream(self._streams[j])
                   if self._streams is not None else _nullctx())
            with ctx:
                if not any(drafts):
                    toks = [r._next_tok for r in reqs]
                    pos = [r._pos for r in reqs]
                    seq = [s // k for s in lane_slots]
                    decode = True   # one row per request, distinct seqs
                else:
                    toks, pos, seq = [], [], []
                    for r, s_, d in zip(reqs, lane_slots, drafts):
                        toks += [r._next_tok] + d
                        pos += list(range(r._pos, r._pos + 1 + len(d)))
                        seq += [s_ // k] * (1 + len(d))
                    decode = False  # same-seq multi-pos rows (mixed
                    #                 admission stream semantics)
                t = torch.tensor(toks, dtype=torch.int32,
                                 device=self._dev)
                p = torch.tensor(pos, dtype=torch.int32,
                                 device=self._dev)
                q = torch.tensor(seq, dtype=torch.int32,
                                 device=self._dev)
                y = eng.forward(eng.embed(t), p, q, decode=decode)
                lg = eng.logits(y, all_logits=True)
                greedy_ids = None
                if any(r.sampler is None for r in reqs):
                    greedy_ids = eng.argmax(lg)
            work.append((reqs, drafts, greedy_ids, lg))
        if self._streams is not None:
            for st in self._streams:
                torch.cuda.current_stream().wait_stream(st)

        finished: List[Request] = []
        for reqs, drafts, greedy_ids, lg in work:
            if greedy_ids is not None and greedy_ids.device.type != "cpu":
                greedy_ids = greedy_ids.cpu()
            lg_host = None
            if any(r.sampler is not None for r in reqs):
                lg_host = lg.float().cpu().numpy()
            self._advance(reqs, drafts, greedy_ids, lg_host, finished)
        return finished

    def _draft(self, r: Request) -> List[int]:
        """Prompt-lookup draft for a greedy request (possibly empty)."""
        if (self.spec_k <= 0 or self.spec_ngram <= 0 or
                r.sampler is not None):
            return []
        n_ctx = getattr(self.engine, "n_ctx", 1 << 30)
        k = min(self.spec_k, r.max_new - len(r.out) - 1,
                n_ctx - r._pos - 2)
        if k <= 0:
            return []
        return lookup_draft(r.prompt + r.out, self.spec_ngram, k)

    def _advance(self, reqs, drafts, greedy_ids, lg_host,
                 finished) -> None:
        row = 0
        for r, draft in zip(reqs, drafts):
            nrows = 1 + len(draft)
            if r.sampler is not None:
                emit = [r.sampler(lg_host[row])]
            else:
                nxt = [int(greedy_ids[row + i]) for i in range(nrows)]
                acc = 0
                while acc < len(draft) and draft[acc] == nxt[acc]:
                    acc += 1
                emit = nxt[:acc + 1]   # verified greedy continuations
            row += nrows
            for tid in emit:
                self.tokens_out += 1
                r.out.append(tid)
                r._next_tok = tid
                r._pos += 1
                if len(r.out) >= r.max_new or (r.eos_id is not None and
                                               tid == r.eos_id):
                    r.done = True
                    del self.active[r.slot]
                    self.free.append(r.slot)
                    r.slot = -1
                    finished.append(r)
                    break

    def cancel(self, req: Request) -> bool:
        """Abort a request: drop it from the queue, or free its slot if
        active (the slot's KV rows are dead — safe to reuse, attention
        is pos-bounded). Returns False if it already finished."""
        if req.done:
            return False
        if req.slot >= 0 and (req.slot in self.active or
                              req.slot in self.prefilling):
            self.active.pop(req.slot, None)
            self.prefilling.pop(req.slot, None)
            self.free.append(req.slot)
            req.slot = -1
        else:
            try:
                self.queue.remove(req)
            except ValueError:
                return False
        req.done = True
        return True

    # --------------------------------------------------------- convenience

    @property
    def pending(self) -> int:
        return len(self.queue) + len(self.active) + len(self.prefilling)

    def run_all(self, max_steps: int = 1 << 20) -> List[Request]:
        """Drive step() until every submitted request finishes."""
        out: List[Request] = []
        for _ in range(max_steps):
            if not self.pending:
                break
            out.extend(self.step())
        assert not self.pending, "run_all hit max_steps with work pending"
        return out
