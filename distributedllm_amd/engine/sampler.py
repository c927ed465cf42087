"""Token sampler — exact behavioral parity with the reference Sampler
(/root/reference/distllm/cli_api/common.py:64-86):

* every previously sampled id gets its logit DIVIDED by the repetition
  penalty (including the negative-logit quirk: dividing a negative logit
  makes it LESS negative, i.e. the "penalty" boosts negative logits —
  reproduced deliberately for behavioral compatibility),
* all logits divide by (temperature + 1e-5),
* softmax then categorical draw (numpy RNG).

A greedy mode mirrors the engine's device argmax semantics.
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
                 seed: Optional[int] = None, greedy: bool = False):
        self.T = temperature
        self.penalty = repeat_penalty
        self.previous_ids: List[int] = []
        self.greedy = greedy
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
        tid = int(self.rng.choice(size, p=probs))
        self.previous_ids.append(tid)
        return tid
