"""The fp32 torch reference model: internal consistency checks.

These properties are what make layer-sliced pipeline inference valid at all:
(1) running layers [0..L) as one slice == chaining slices [0..k) and [k..L),
(2) incremental decode with KV cache == recomputing the whole prefix.
"""
import numpy as np
import torch

from distributedllm_amd.formats import synthetic
from distributedllm_amd.models.llama import (
    LlamaExtraRef, LlamaSliceRef, weights_from_ggml, rope_interleaved)


def _setup(preset="tiny", seed=0):
    f = synthetic.build_model(preset, seed=seed)
    w = weights_from_ggml(f)
    return f, w


class TestRope:
    def test_zero_position_identity(self):
        x = torch.randn(3, 4, 16)
        y = rope_interleaved(x, 0)
        assert torch.allclose(y[0], x[0], atol=1e-6)  # pos 0: no rotation
        assert not torch.allclose(y[1], x[1])

    def test_norm_preserved(self):
        x = torch.randn(5, 4, 16)
        y = rope_interleaved(x, 7)
        # rotation preserves the norm of each pair
        xp = x.view(5, 4, 8, 2)
        yp = y.view(5, 4, 8, 2)
        assert torch.allclose(xp.norm(dim=-1), yp.norm(dim=-1), atol=1e-5)


class TestSliceChaining:
    def test_two_slices_equal_one(self):
        f, w = _setup()
        hp = f.hparams
        x = torch.randn(4, hp.n_embd) * 0.5

        whole = LlamaSliceRef(hp, w, first_layer=0, n_layers=3)
        y_whole = whole.forward(x.clone())

        s0 = LlamaSliceRef(hp, w, first_layer=0, n_layers=2)
        s1 = LlamaSliceRef(hp, w, first_layer=2, n_layers=1)
        y_chain = s1.forward(s0.forward(x.clone()))

        assert torch.allclose(y_whole, y_chain, atol=1e-5)

    def test_incremental_decode_matches_full(self):
        f, w = _setup()
        hp = f.hparams
        x = torch.randn(5, hp.n_embd) * 0.5

        full = LlamaSliceRef(hp, w, first_layer=0, n_layers=3)
        y_full = full.forward(x.clone())

        inc = LlamaSliceRef(hp, w, first_layer=0, n_layers=3)
        y0 = inc.forward(x[:3].clone())
        y1 = inc.forward(x[3:4].clone())
        y2 = inc.forward(x[4:5].clone())

        assert torch.allclose(y_full[:3], y0, atol=1e-5)
        assert torch.allclose(y_full[3], y1[0], atol=1e-4)
        assert torch.allclose(y_full[4], y2[0], atol=1e-4)

    def test_clear_context(self):
        f, w = _setup()
        hp = f.hparams
        x = torch.randn(2, hp.n_embd)
        s = LlamaSliceRef(hp, w, first_layer=0, n_layers=3)
        y1 = s.forward(x.clone())
        s.clear_context()
        y2 = s.forward(x.clone())
        assert torch.allclose(y1, y2, atol=1e-6)


class TestExtra:
    def test_embed_and_logits_shapes(self):
        f, w = _setup()
        hp = f.hparams
        ex = LlamaExtraRef(w)
        emb = ex.embed([1, 5, 7])
        assert emb.shape == (3, hp.n_embd)
        lg = ex.logits(torch.randn(3, hp.n_embd))
        assert lg.shape == (1, hp.n_vocab)
        lg_all = ex.logits(torch.randn(3, hp.n_embd), all_logits=True)
        assert lg_all.shape == (3, hp.n_vocab)
