"""CPU torch engine vs. the simpler LlamaSliceRef ground truth."""
import torch

from distributedllm_amd.engine import TorchSliceEngine
from distributedllm_amd.formats import slicer, synthetic
from distributedllm_amd.models.llama import (
    LlamaExtraRef, LlamaSliceRef, weights_from_ggml)


def _mk(preset="tiny", seed=0):
    f = synthetic.build_model(preset, seed=seed)
    return f, weights_from_ggml(f)


class TestTorchEngineParity:
    def test_prefill_matches_ref(self):
        f, w = _mk()
        hp = f.hparams
        x = torch.randn(5, hp.n_embd) * 0.5

        ref = LlamaSliceRef(hp, w, 0, hp.n_layer)
        y_ref = ref.forward(x.clone())

        eng = TorchSliceEngine(hp, dict(w), hp.n_layer, 0, n_ctx=64,
                               max_batch=2)
        pos = torch.arange(5, dtype=torch.int32)
        seq = torch.zeros(5, dtype=torch.int32)
        y = eng.forward(x.clone(), pos, seq)
        assert torch.allclose(y, y_ref, atol=1e-4)

    def test_batched_decode_isolation(self):
        # two sequences decoding in one batch == each decoded alone
        f, w = _mk()
        hp = f.hparams
        xa = torch.randn(3, hp.n_embd) * 0.5
        xb = torch.randn(3, hp.n_embd) * 0.5

        def run_alone(xs):
            eng = TorchSliceEngine(hp, dict(w), hp.n_layer, 0, n_ctx=16,
                                   max_batch=1)
            outs = []
            for t in range(3):
                pos = torch.tensor([t], dtype=torch.int32)
                seq = torch.tensor([0], dtype=torch.int32)
                outs.append(eng.forward(xs[t:t + 1].clone(), pos, seq))
            return torch.cat(outs)

        ya = run_alone(xa)
        yb = run_alone(xb)

        eng = TorchSliceEngine(hp, dict(w), hp.n_layer, 0, n_ctx=16,
                               max_batch=2)
        outs = []
        for t in range(3):
            x = torch.stack([xa[t], xb[t]])
            pos = torch.tensor([t, t], dtype=torch.int32)
            seq = torch.tensor([0, 1], dtype=torch.int32)
            outs.append(eng.forward(x.clone(), pos, seq))
        y = torch.stack(outs)  # [3, 2, E]
        assert torch.allclose(y[:, 0], ya, atol=1e-4)
        assert torch.allclose(y[:, 1], yb, atol=1e-4)

    def test_sliced_chain_with_extra(self):
        f, w = _mk()
        hp = f.hparams
        s0f = slicer.make_slice(f, 0, 1)
        s1f = slicer.make_slice(f, 2, 2)
        ex = slicer.make_extra_layers(f)

        e0 = TorchSliceEngine.from_ggml(s0f, n_ctx=16, max_batch=1)
        e1 = TorchSliceEngine.from_ggml(s1f, n_ctx=16, max_batch=1)
        e1.attach_extra(ex)

        tokens = torch.tensor([1, 5, 9], dtype=torch.int32)
        x = e1.embed(tokens)  # embeddings come from extra weights
        pos = torch.arange(3, dtype=torch.int32)
        seq = torch.zeros(3, dtype=torch.int32)
        y = e1.forward(e0.forward(x.clone(), pos, seq), pos, seq)
        lg = e1.logits(y)

        ref_s = LlamaSliceRef(hp, w, 0, hp.n_layer)
        ref_e = LlamaExtraRef(w)
        y_ref = ref_s.forward(ref_e.embed([1, 5, 9]))
        lg_ref = ref_e.logits(y_ref)
        assert torch.allclose(lg, lg_ref, atol=1e-3)


def test_byte_quant_repack_values_exact():
    """The W_Q8B repack's re-biased bytes must reconstruct the codec's
    dequantized weights exactly: w = alpha*(u - 128) + beta (the GPU
    kernel computes the same quantity in packed f16)."""
    import numpy as np
    from distributedllm_amd.engine.slice_engine import _byte_values
    from distributedllm_amd.formats import ggml, synthetic
    for ft in (ggml.FTYPE_MOSTLY_Q5_0, ggml.FTYPE_MOSTLY_Q5_1,
               ggml.FTYPE_MOSTLY_Q8_0):
        f = synthetic.build_model("tiny", seed=0, ftype=ft)
        t = next(x for x in f.tensors
                 if x.name.endswith("attention.wq.weight"))
        vals, alpha, beta = _byte_values(t)
        w = (alpha[..., None] * (vals.astype(np.float32) - 128.0) +
             beta[..., None])
        ref = t.to_f32().reshape(w.shape[0], -1, 32)
        assert np.abs(w - ref).max() == 0.0, ggml.TYPE_NAMES[t.gtype]


def test_detile_inverts_repack():
    """detile_mfma (the reference inverse of the kernel tile layouts)
    must exactly invert repack_mfma: for every quant format, repack (to
    CPU tensors) then detile reproduces the codec's dequantized weights
    (f16-rounded) — this is the CPU-side proof that the layouts the
    prefill/decode kernels stream carry the exact quantized values."""
    import numpy as np
    import torch
    from distributedllm_amd.engine.slice_engine import (
        detile_mfma, repack_mfma)
    from distributedllm_amd.formats import ggml, synthetic
    for ft in (ggml.FTYPE_MOSTLY_Q4_0, ggml.FTYPE_MOSTLY_Q4_1,
               ggml.FTYPE_MOSTLY_Q8_0, ggml.FTYPE_MOSTLY_Q5_0,
               ggml.FTYPE_MOSTLY_Q5_1, ggml.FTYPE_MOSTLY_F16):
        f = synthetic.build_model("tiny", seed=1, ftype=ft)
        t = next(x for x in f.tensors
                 if x.name.endswith("feed_forward.w1.weight"))
        rows, cols = t.shape_rows_cols
        mat = repack_mfma(t, "cpu")
        got = detile_mfma(mat, rows, cols).float()
        want = torch.from_numpy(t.to_f32())
        # detile emits f16 (one rounding beyond the codec's own f16
        # arithmetic); q4_1/q5_1 fold beta=m (+16d) with one more f16
        # rounding at repack
        tol = 4e-3 * want.abs().max().item() + 1e-6
        err = (got - want).abs().max().item()
        assert err <= tol, (ggml.TYPE_NAMES[ggml._FTYPE_TO_GGML[ft]], err)


def test_torch_q4_repack_matches_numpy():
    """The GPU-side torch q4 repack (one H2D of compressed bytes, bit
    work on device) must produce bit-identical tiles/scales to the
    numpy reference path."""
    import numpy as np
    import torch
    from distributedllm_amd.engine import slice_engine as SE
    from distributedllm_amd.formats import ggml, synthetic
    for ft in (ggml.FTYPE_MOSTLY_Q4_0, ggml.FTYPE_MOSTLY_Q4_1):
        f = synthetic.build_model("small", seed=3, ftype=ft)
        t = next(x for x in f.tensors
                 if x.name.endswith("attention.wq.weight"))
        d_np, s_np, wt_np = SE.repack_mfma(t, "cpu")       # numpy path
        d_th, s_th, wt_th = SE._repack_q4_torch(t, "cpu")  # torch ops
        assert wt_np == wt_th
        assert torch.equal(d_np, d_th)
        assert torch.equal(s_np.view(torch.int16), s_th.view(torch.int16))


def test_torch_byte_repack_matches_numpy():
    """Torch q5_0/q5_1/q8_0 byte-stream repack == numpy path, bit for
    bit (same contract as the q4 port)."""
    import torch
    from distributedllm_amd.engine import slice_engine as SE
    from distributedllm_amd.formats import ggml, synthetic
    for ft in (ggml.FTYPE_MOSTLY_Q5_0, ggml.FTYPE_MOSTLY_Q5_1,
               ggml.FTYPE_MOSTLY_Q8_0):
        f = synthetic.build_model("small", seed=5, ftype=ft)
        t = next(x for x in f.tensors
                 if x.name.endswith("attention.wq.weight"))
        d_np, s_np, wt_np = SE.repack_mfma(t, "cpu")        # numpy path
        d_th, s_th, wt_th = SE._repack_byte_torch(t, "cpu")
        assert wt_np == wt_th
        assert torch.equal(d_np, d_th), ggml.TYPE_NAMES[t.gtype]
        assert torch.equal(s_np.view(torch.int16), s_th.view(torch.int16))


def test_torch_f16_repack_matches_numpy():
    import torch
    from distributedllm_amd.engine import slice_engine as SE
    from distributedllm_amd.formats import ggml, synthetic
    f = synthetic.build_model("small", seed=6, ftype=ggml.FTYPE_MOSTLY_F16)
    t = next(x for x in f.tensors
             if x.name.endswith("feed_forward.w1.weight"))
    d_np, _, wt_np = SE.repack_mfma(t, "cpu")
    d_th, _, wt_th = SE._repack_f16_torch(t, "cpu")
    assert wt_np == wt_th
    assert torch.equal(d_np.view(torch.int16), d_th.view(torch.int16))


def test_torch_kquant_repack_matches_numpy():
    """Torch q2_K..q6_K super-block repack == numpy path, bit for bit."""
    import torch
    from distributedllm_amd.engine import slice_engine as SE
    from distributedllm_amd.formats import ggml, synthetic
    for ft in (ggml.FTYPE_MOSTLY_Q2_K, ggml.FTYPE_MOSTLY_Q3_K_M,
               ggml.FTYPE_MOSTLY_Q4_K_M, ggml.FTYPE_MOSTLY_Q5_K_M,
               ggml.FTYPE_MOSTLY_Q6_K):
        f = synthetic.build_model("small_k", seed=8, ftype=ft)
        for tname in ("attention.wq.weight", "feed_forward.w1.weight"):
            t = next(x for x in f.tensors if x.name.endswith(tname))
            d_np, s_np, wt_np = SE.repack_mfma(t, "cpu")    # numpy path
            d_th, s_th, wt_th = SE._repack_kquant_torch(t, "cpu")
            assert wt_np == wt_th
            assert torch.equal(d_np, d_th), ggml.TYPE_NAMES[t.gtype]
            assert torch.equal(s_np.view(torch.int16),
                               s_th.view(torch.int16)), \
                ggml.TYPE_NAMES[t.gtype]
