"""Prompt-lookup speculative decoding (single-stream greedy).

Greedy decode advances one token per forward; at batch 1 the step is
launch-latency-bound (docs/roadmap.md #4). This module generates
MULTIPLE tokens per forward with a model-free draft: the last `ngram`
generated tokens are looked up in the sequence so far, and the tokens
that followed their most recent earlier occurrence are proposed as the
continuation ("prompt lookup decoding"). One forward evaluates the
current token plus the k draft tokens at their true positions (the
engine is stateless over explicit pos/seq — the same mixed-admission
prefill semantics the batcher uses), and the longest prefix whose
greedy argmax agrees is accepted.

Output is TOKEN-EXACT with plain greedy decode by construction: row i
of the verification forward computes the argmax AFTER tokens
..., cur, draft[0..i-1], which is exactly what sequential greedy would
compute once draft[0..i-1] are confirmed. Rejected draft rows leave
stale KV at positions >= the accepted point; those rows are rewritten
by the next verification forward before any attention reads them
(attention streams rows [0, pos) only), so no cleanup is needed.

The reference has no equivalent (one token per full TCP round-trip);
this composes with any engine exposing the stateless forward/embed/
logits/argmax interface (HIP or the CPU twin — exactness is asserted
against sequential greedy in tests/test_speculative.py).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional

import torch


def lookup_draft(ids: List[int], ngram: int, k: int) -> List[int]:
    """Most recent earlier occurrence of the trailing `ngram` tokens;
    returns up to k tokens that followed it (empty when no match)."""
    n = len(ids)
    if ngram <= 0 or n <= ngram:
        return []
    tail = ids[n - ngram:]
    # scan right-to-left over candidate start positions of the match,
    # excluding the trailing occurrence itself
    for s in range(n - ngram - 1, -1, -1):
        if ids[s:s + ngram] == tail:
            return ids[s + ngram:s + ngram + k]
    return []


@dataclass
class SpecStats:
    forwards: int = 0
    tokens: int = 0
    accepted_per_forward: List[int] = field(default_factory=list)


def pld_generate(engine, prompt_ids, max_new: int, *, ngram: int = 3,
                 k: int = 8, seq_id: int = 0,
                 eos_id: Optional[int] = None,
                 stats: Optional[SpecStats] = None) -> List[int]:
    """Greedy generation with prompt-lookup speculation.

    Returns the generated token ids (length <= max_new; stops at
    eos_id). Token-exact with sequential greedy decode.
    """
    dev = getattr(engine, "device", "cpu")
    ids = [int(t) for t in prompt_ids]
    if not ids:
        raise ValueError("prompt must be non-empty")
    n_ctx = int(engine.n_ctx)
    if len(ids) + max_new > n_ctx:
        raise ValueError(
            f"prompt ({len(ids)}) + max_new ({max_new}) exceeds n_ctx "
            f"{n_ctx}")
    cap = int(getattr(engine, "max_prefill",
                      getattr(engine, "max_tokens", 64)) or 64)

    def fwd(tokens: List[int], p0: int) -> torch.Tensor:
        t = torch.tensor(tokens, dtype=torch.int32, device=dev)
        pos = torch.arange(p0, p0 + len(tokens), dtype=torch.int32,
                           device=dev)
        seq = torch.full((len(tokens),), seq_id, dtype=torch.int32,
                         device=dev)
        return engine.forward(engine.embed(t), pos, seq)

    # prefill prompt[:-1] (tiled to the engine's per-call cap)
    for t0 in range(0, len(ids) - 1, cap):
        fwd(ids[t0:t0 + cap], t0)

    out: List[int] = []
    cur = ids[-1]
    p = len(ids) - 1
    st = stats if stats is not None else SpecStats()
    while len(out) < max_new:
        draft = lookup_draft(ids, ngram, min(k, cap - 1,
                                             max_new - len(out) - 1,
                                             n_ctx - p - 2))
        toks = [cur] + draft
        y = engine.forward(
            engine.embed(torch.tensor(toks, dtype=torch.int32,
                                      device=dev)),
            torch.arange(p, p + len(toks), dtype=torch.int32, device=dev),
            torch.full((len(toks),), seq_id, dtype=torch.int32,
                       device=dev),
            decode=(len(toks) == 1))
        nxt = engine.argmax(engine.logits(y, all_logits=True))
        nxt = [int(v) for v in nxt.tolist()]
        st.forwards += 1
        accepted = 0
        while accepted < len(draft) and draft[accepted] == nxt[accepted]:
            accepted += 1
        emit = nxt[:accepted + 1]  # verified continuations
        st.accepted_per_forward.append(len(emit))
        for t in emit:
            out.append(t)
            ids.append(t)
            st.tokens += 1
            if (eos_id is not None and t == eos_id) or len(out) >= max_new:
                break
        else:
            cur = out[-1]
            p += len(emit)
            continue
        break
    return out
