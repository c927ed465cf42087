"""Token sampler — exact behavioral parity with the reference Sampler
(/root/reference/distllm/cli_api/common.py:64-86):

* every previously sampled id gets its logit DIVIDED by the repetition
  penalty (including the negative-logit quirk: dividing a negative logit
  makes it LESS negative, i.e. the "penalty" boosts negative logits —
  reproduced deliberately for behavioral compatibility),
* all logits divide by (temperature + 1e-5),
* softmax then categorical draw (numpy RNG).

A greedy mode mirrors the engine's device argmax semantics. Optional
``top_k``/``top_p`` (nucleus) filtering extends the reference surface
(off by default — defaults are exact reference behavior).
"""
from __future__ import annotations

from typing import List, Optional

import numpy as np


def softmax(x: np.ndarray) -> np.ndarray:
    x = x - x.max()
    e = np.exp(x)
    return e / e.sum()


class Sampler:
    EPS = 1e-5

    def __init__(self, temperature: float = 0.7, repeat_penalty: float = 1.1,
                 seed: Optional[int] = None, greedy: bool = False,
                 top_k: int = 0, top_p: float = 1.0):
        self.T = temperature
        self.penalty = repeat_penalty
        self.previous_ids: List[int] = []
        self.greedy = greedy
        self.top_k = top_k      # 0 = off
        self.top_p = top_p      # 1.0 = off
        self.rng = np.random.default_rng(seed)

    def __call__(self, logits) -> int:
        logits = np.asarray(logits, dtype=np.float64).reshape(-1)
        if self.greedy:
            tid = int(np.argmax(logits))
            self.previous_ids.append(tid)
            return tid
        size = logits.shape[0]
        mask = np.isin(np.arange(size), self.previous_ids)
        penalties = (mask * self.penalty + ~mask) * (self.T + self.EPS)
        probs = softmax(logits / penalties)
        if self.top_k and self.top_k < size:
            kth = np.partition(probs, -self.top_k)[-self.top_k]
            probs = np.where(probs >= kth, probs, 0.0)
            # renormalize before the nucleus cut: sequential-filter
            # semantics (llama.cpp/HF renormalize between filters) —
            # without this, top_p over the <1 surviving mass keeps a
            # different set, and top_p > surviving mass is a no-op
            probs = probs / probs.sum()
        if self.top_p < 1.0:
            order = np.argsort(-probs)
            csum = np.cumsum(probs[order])
            # keep the smallest prefix whose mass reaches top_p
            cut = int(np.searchsorted(csum, self.top_p) + 1)
            keep = np.zeros(size, dtype=bool)
            keep[order[:cut]] = True
            probs = np.where(keep, probs, 0.0)
        if self.top_k or self.top_p < 1.0:
            probs = probs / probs.sum()
        tid = int(self.rng.choice(size, p=probs))
        self.previous_ids.append(tid)
        return tid
