"""Prompt-lookup speculative decoding: token-exactness vs sequential
greedy (the whole point — speculation must never change the output),
draft-lookup behavior, and the KV-staleness edge (rejected draft rows
must not leak into later steps)."""
import pytest
import torch

from distributedllm_amd.engine import TorchSliceEngine
from distributedllm_amd.formats import slicer, synthetic
from distributedllm_amd.serving.speculative import (
    SpecStats, lookup_draft, pld_generate)


def _engine(n_ctx=128, max_batch=2):
    f = synthetic.build_model("tiny", seed=0)
    ex = slicer.make_extra_layers(f)
    eng = TorchSliceEngine.from_ggml(f, n_ctx=n_ctx, max_batch=max_batch)
    eng.attach_extra(ex)
    return eng


def _greedy(eng, prompt, max_new, seq_id=0):
    """Plain sequential greedy decode, one token per forward."""
    ids = list(prompt)
    for t0 in range(0, len(ids) - 1, 64):
        toks = torch.tensor(ids[t0:t0 + 64], dtype=torch.int32)
        pos = torch.arange(t0, t0 + toks.numel(), dtype=torch.int32)
        eng.forward(eng.embed(toks), pos,
                    torch.full((toks.numel(),), seq_id,
                               dtype=torch.int32))
    out = []
    cur, p = ids[-1], len(ids) - 1
    for _ in range(max_new):
        y = eng.forward(eng.embed(torch.tensor([cur], dtype=torch.int32)),
                        torch.tensor([p], dtype=torch.int32),
                        torch.tensor([seq_id], dtype=torch.int32),
                        decode=True)
        cur = int(eng.argmax(eng.logits(y, all_logits=True))[0])
        out.append(cur)
        p += 1
    return out


def test_lookup_draft():
    ids = [1, 2, 3, 9, 9, 1, 2, 3]
    assert lookup_draft(ids, 3, 4) == [9, 9, 1, 2]   # follows [1,2,3]
    assert lookup_draft(ids, 3, 2) == [9, 9]
    assert lookup_draft([5, 6, 7], 3, 4) == []        # no earlier match
    assert lookup_draft([1, 1, 1, 1], 2, 8) == [1]  # most recent occ.
    #   (the s=1 match has a single follower — ids[3:])
    assert lookup_draft([1, 2], 3, 4) == []            # shorter than ngram


@pytest.mark.parametrize("prompt", [
    [5, 9, 3],
    [7, 7, 7, 7, 7, 7],                       # max repetition: all accepts
    list(range(3, 30)),                        # no repetition: no drafts
    [4, 8, 2, 4, 8, 2, 4, 8],                 # periodic
])
def test_pld_matches_sequential_greedy(prompt):
    want = _greedy(_engine(), prompt, 24)
    st = SpecStats()
    got = pld_generate(_engine(), prompt, 24, ngram=3, k=6, stats=st)
    assert got == want
    assert st.tokens == len(got)
    assert st.forwards <= 24  # never worse than one forward per token


def test_pld_accepts_on_repetitive_text():
    """Random-init tiny models loop quickly; speculation must exploit it
    (fewer forwards than tokens) while staying exact."""
    eng = _engine()
    st = SpecStats()
    got = pld_generate(eng, [7, 7, 7, 7], 32, ngram=2, k=8, stats=st)
    assert got == _greedy(_engine(), [7, 7, 7, 7], 32)
    assert st.forwards < st.tokens, (st.forwards, st.tokens)


def test_pld_rejected_draft_rows_do_not_leak():
    """Force mispredictions (tiny k, ngram=1 on quasi-random output) and
    check exactness still holds — covers the stale-KV-row argument."""
    prompt = [11, 3, 11, 5, 11]
    want = _greedy(_engine(), prompt, 20)
    got = pld_generate(_engine(), prompt, 20, ngram=1, k=3)
    assert got == want


def test_pld_context_overflow_rejected():
    with pytest.raises(ValueError, match="n_ctx"):
        pld_generate(_engine(n_ctx=16), [1, 2, 3], 20)


def test_pld_eos_stops():
    eng = _engine()
    full = pld_generate(_engine(), [7, 7, 7], 16)
    if len(set(full)) > 1:
        eos = full[len(full) // 2]
        got = pld_generate(eng, [7, 7, 7], 16, eos_id=eos)
        assert got == full[:full.index(eos) + 1]


def test_pld_exact_on_gqa_model():
    """Speculation over a GQA engine (grouped KV heads) stays exact."""
    from distributedllm_amd.formats import ggml

    def eng():
        f = synthetic.build_model("tiny_gqa", seed=2,
                                  ftype=ggml.FTYPE_MOSTLY_F16)
        e = TorchSliceEngine.from_ggml(f, n_ctx=96, max_batch=1)
        e.attach_extra(slicer.make_extra_layers(f))
        return e

    for prompt in ([9, 9, 9, 9], [4, 8, 2, 4, 8, 2]):
        want = _greedy(eng(), prompt, 20)
        got = pld_generate(eng(), prompt, 20, ngram=2, k=5)
        assert got == want, prompt
