"""Client-side distributed model: generation/perplexity over TCP nodes.

Parity with the reference's `DistributedLLM` + `get_llm`
(/root/reference/distllm/cli_api/common.py:9-154): the client tokenizes and
embeds locally from the extra-layers file, hops activations through the
nodes in layer order, computes logits locally, samples, repeats. Unlike
the reference it loads the extra-layers file ONCE (the reference re-parses
it from disk every call — tensor_processor.cpp:1719, 1789) and tracks
n_past explicitly (stateless nodes).

This TCP data plane is the functional/portability path (and the CPU
cluster configuration); the performance path on an MI355X node is the RCCL
pipeline in parallel/pipeline.py.
"""
from __future__ import annotations

import json
from typing import Iterator, List, Optional, Tuple

import numpy as np
import torch

from ..engine.sampler import Sampler
from ..engine.tokenizer import Tokenizer
from ..utils.metrics import StageTimer, ThroughputMeter
from ..formats import ggml
from ..models.llama import LlamaExtraRef, weights_from_ggml
from .client import Connection, parse_address
from .registry import Registry


class DistributedLLM:
    def __init__(self, connections: List[Tuple[Connection, int, int]],
                 extra_layers_path: str):
        """connections: [(conn, first_layer, last_layer)] sorted by range."""
        self.nodes = sorted(connections, key=lambda c: c[1])
        extra = ggml.GGMLFile.load(extra_layers_path, extended=True)
        self.tokenizer = Tokenizer(extra.vocab)
        self.extra = LlamaExtraRef(weights_from_ggml(extra))
        self.n_past = 0
        # built-in observability (the reference never reports
        # its timing fields) — read after generate()/perplexity()
        self.throughput = ThroughputMeter()
        self.stages = StageTimer()

    # ------------------------------------------------------------- plumbing

    def clear_context(self) -> None:
        self.n_past = 0
        for conn, _, _ in self.nodes:
            conn.clear_context()

    def propagate_tensor(self, x: np.ndarray) -> np.ndarray:
        """One hop through every node in layer order (the reference's
        client-mediated pipeline, common.py:148-154)."""
        for i, (conn, _, _) in enumerate(self.nodes):
            self.stages.start(f"hop{i}")
            x = conn.propagate_forward(x, start_pos=self.n_past)
            self.stages.stop(f"hop{i}")
        self.n_past += x.shape[0]
        return x

    def _embed(self, tokens: List[int]) -> np.ndarray:
        return self.extra.embed(tokens).numpy().astype(np.float32)

    def _logits(self, x: np.ndarray, all_logits: bool = False) -> np.ndarray:
        t = torch.from_numpy(np.ascontiguousarray(x))
        return self.extra.logits(t, all_logits=all_logits).numpy()

    # ------------------------------------------------------------------ api

    def generate(self, prompt: str, max_steps: int = 50,
                 temperature: float = 0.7, repeat_penalty: float = 1.1,
                 seed: Optional[int] = None, greedy: bool = False,
                 top_k: int = 0, top_p: float = 1.0,
                 speculative: int = 0) -> Iterator[str]:
        """speculative=K (greedy only): prompt-lookup drafts of up to K
        tokens are verified per pipeline hop — the per-token TCP
        round-trip (THE latency bound of this plane) is paid once per
        accepted run instead of once per token. Token-exact with plain
        greedy (serving/speculative.py has the argument; the nodes are
        stateless over explicit start_pos, so a rejection simply rewinds
        the client's n_past and the stale KV rows are rewritten)."""
        self.clear_context()
        self.throughput.reset()
        tokens = self.tokenizer.encode(prompt, bos=True)
        if speculative > 0 and greedy:
            yield from self._generate_spec(tokens, max_steps, speculative)
            return
        sampler = Sampler(temperature, repeat_penalty, seed=seed,
                          greedy=greedy, top_k=top_k, top_p=top_p)
        cur = tokens
        for _ in range(max_steps):
            x = self._embed(cur)
            y = self.propagate_tensor(x)
            logits = self._logits(y, all_logits=False)[0]
            tid = sampler(logits)
            self.throughput.tick()
            yield self.tokenizer.decode_token(tid)
            cur = [tid]

    def _generate_spec(self, tokens: List[int], max_steps: int,
                       k: int) -> Iterator[str]:
        from ..serving.speculative import lookup_draft
        if len(tokens) > 1:
            self.propagate_tensor(self._embed(tokens[:-1]))
        cur = tokens[-1]
        p = len(tokens) - 1
        emitted = 0
        while emitted < max_steps:
            draft = lookup_draft(tokens, 3,
                                 min(k, max_steps - emitted - 1))
            self.n_past = p   # rewind over any rejected rows
            y = self.propagate_tensor(self._embed([cur] + draft))
            lg = self._logits(np.ascontiguousarray(y), all_logits=True)
            nxt = lg.argmax(axis=-1)
            acc = 0
            while acc < len(draft) and draft[acc] == int(nxt[acc]):
                acc += 1
            for tid in nxt[:acc + 1]:
                tokens.append(int(tid))
                emitted += 1
                self.throughput.tick()
                yield self.tokenizer.decode_token(int(tid))
                if emitted >= max_steps:
                    return
            cur = tokens[-1]
            p += acc + 1

    def perplexity(self, text: str) -> float:
        """exp(mean NLL) of each next token given its prefix
        (reference semantics, common.py:113-141)."""
        self.clear_context()
        tokens = self.tokenizer.encode(text, bos=True)
        if len(tokens) < 2:
            raise ValueError("perplexity needs at least 2 tokens")
        x = self._embed(tokens[:-1])
        y = self.propagate_tensor(x)
        logits = self._logits(y, all_logits=True)
        logp = torch.log_softmax(torch.from_numpy(logits), dim=-1).numpy()
        nll = [-logp[i, tokens[i + 1]] for i in range(len(tokens) - 1)]
        return float(np.exp(np.mean(nll)))


def get_llm(config_path: str, root: str = ".") -> DistributedLLM:
    """Build a client from a cluster config + the models registry
    (reference get_llm, common.py:9-27): ensures each node has its slice
    loaded, then wires the pipeline in layer order."""
    with open(config_path) as f:
        cfg = json.load(f)
    model_id = cfg["model_id"]
    reg = Registry(root)
    entry = reg.get(model_id)
    if entry is None:
        raise KeyError(f"model {model_id!r} not provisioned (run provision)")
    nodes_map = cfg["nodes_map"]
    conns: List[Tuple[Connection, int, int]] = []
    for addr, (a, b) in nodes_map.items():
        host, port = parse_address(addr)
        conn = Connection(host, port)
        load_one_slice(conn, entry, a, b)
        conns.append((conn, a, b))
    return DistributedLLM(conns, entry.extra_layers_file)


def load_one_slice(conn: Connection, entry, a: int, b: int) -> None:
    """Idempotent: skip if the node already reports this slice loaded
    (reference common.py:35-56)."""
    import os
    want = None
    for s in entry.slices:
        if s.a == a and s.b == b:
            want = os.path.basename(s.path)
            break
    if want is None:
        raise KeyError(f"no provisioned slice for layers [{a}, {b}]")
    status = conn.get_status()
    if status.model == want and status.first_layer == a:
        return
    names = {s["name"] for s in conn.list_slices()}
    if want not in names:
        raise FileNotFoundError(
            f"slice {want!r} not uploaded to the node; run provision")
    conn.load_slice(want)
