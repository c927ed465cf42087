"""GPU numerics: the HIP/CDNA4 engine against the fp32 torch reference.

Every test loads the SAME GGML bytes into both engines, so the q4_0
dequantization, RMSNorm, RoPE, KV cache, attention, SwiGLU and lm_head
paths are compared end-to-end against plain fp32 PyTorch math
("kernel unit tests vs CPU reference").
Tolerances: activations are f32 in both engines; the HIP KV cache is f16,
so per-layer drift is ~1e-3 relative.
"""
import numpy as np
import pytest
import torch

from distributedllm_amd.formats import ggml, slicer, synthetic
from distributedllm_amd.models.llama import PRESETS

pytestmark = pytest.mark.gpu



def _assert_close(a, b, rel_rms=8e-3, rel_max=6e-2, label=""):
    """Layout/indexing bugs give O(1) relative error; bf16 B-operand
    rounding gives ~0.2-0.5% — assert in relative-RMS terms."""
    a = a.float(); b = b.float()
    num = (a - b).norm().item()
    den = max(b.norm().item(), 1e-6)
    rr = num / den
    mx = (a - b).abs().max().item() / max(b.abs().max().item(), 1e-6)
    assert rr <= rel_rms and mx <= rel_max, \
        f"{label}: rel_rms={rr:.2e} rel_max={mx:.2e}"

def _engines(preset="tiny", ftype=ggml.FTYPE_MOSTLY_Q4_0, seed=0,
             n_ctx=64, max_batch=2):
    from distributedllm_amd.engine import HIPSliceEngine, TorchSliceEngine
    f = synthetic.build_model(preset, ftype=ftype, seed=seed)
    ex = slicer.make_extra_layers(f)
    hip = HIPSliceEngine.from_ggml(f, n_ctx=n_ctx, max_batch=max_batch)
    hip.attach_extra(ex)
    cpu = TorchSliceEngine.from_ggml(f, n_ctx=n_ctx, max_batch=max_batch)
    cpu.attach_extra(ex)
    return f, hip, cpu


def test_native_extension_is_loaded():
    """The GPU path must run our HIP kernels — no fallback allowed."""
    from distributedllm_amd import ops
    m = ops.core()
    assert m.__file__.endswith("_core.so")


@pytest.mark.parametrize("ftype", [ggml.FTYPE_MOSTLY_Q4_0,
                                   ggml.FTYPE_MOSTLY_Q4_1,
                                   ggml.FTYPE_MOSTLY_F16,
                                   ggml.FTYPE_ALL_F32,
                                   ggml.FTYPE_MOSTLY_Q8_0,
                                   ggml.FTYPE_MOSTLY_Q5_0,
                                   ggml.FTYPE_MOSTLY_Q5_1])
def test_embed_matches(ftype):
    f, hip, cpu = _engines(ftype=ftype)
    toks = torch.tensor([0, 5, 17, 255], dtype=torch.int32)
    a = hip.embed(toks.cuda()).cpu()
    b = cpu.embed(toks)
    # byte-format tables are dequantized to f16 at load (one rounding)
    atol = 2e-3 if ftype in (ggml.FTYPE_MOSTLY_Q8_0, ggml.FTYPE_MOSTLY_Q5_0,
                             ggml.FTYPE_MOSTLY_Q5_1) else 1e-6
    assert torch.allclose(a, b, atol=atol), (a - b).abs().max()


@pytest.mark.parametrize("ftype", [ggml.FTYPE_MOSTLY_Q4_0,
                                   ggml.FTYPE_MOSTLY_Q4_1,
                                   ggml.FTYPE_MOSTLY_F16,
                                   ggml.FTYPE_MOSTLY_Q8_0,
                                   ggml.FTYPE_MOSTLY_Q5_0,
                                   ggml.FTYPE_MOSTLY_Q5_1])
def test_forward_prefill_matches(ftype):
    f, hip, cpu = _engines(ftype=ftype)
    hp = f.hparams
    T = 5
    x = torch.randn(T, hp.n_embd) * 0.5
    pos = torch.arange(T, dtype=torch.int32)
    seq = torch.zeros(T, dtype=torch.int32)
    y_gpu = hip.forward(x.cuda(), pos.cuda(), seq.cuda()).cpu()
    y_cpu = cpu.forward(x.clone(), pos, seq)
    _assert_close(y_gpu, y_cpu, label="prefill")


def test_decode_steps_match():
    f, hip, cpu = _engines()
    hp = f.hparams
    torch.manual_seed(1)
    for t in range(6):
        x = torch.randn(1, hp.n_embd) * 0.5
        pos = torch.tensor([t], dtype=torch.int32)
        seq = torch.tensor([0], dtype=torch.int32)
        y_gpu = hip.forward(x.cuda(), pos.cuda(), seq.cuda()).cpu()
        y_cpu = cpu.forward(x.clone(), pos, seq)
        _assert_close(y_gpu, y_cpu, label=f"decode step {t}")


def test_batched_decode_matches():
    f, hip, cpu = _engines(max_batch=2)
    hp = f.hparams
    torch.manual_seed(2)
    for t in range(4):
        x = torch.randn(2, hp.n_embd) * 0.5
        pos = torch.tensor([t, t], dtype=torch.int32)
        seq = torch.tensor([0, 1], dtype=torch.int32)
        y_gpu = hip.forward(x.cuda(), pos.cuda(), seq.cuda()).cpu()
        y_cpu = cpu.forward(x.clone(), pos, seq)
        _assert_close(y_gpu, y_cpu, label=f"batched step {t}")


@pytest.mark.parametrize("T", [20, 40, 64])
def test_multi_tile_token_batches(T):
    """T > 16 runs the JT=2/4 multi-column-tile MFMA path (with a partial
    last tile at T=20/40), including its logits/sampling stages."""
    f, hip, cpu = _engines(n_ctx=96, max_batch=1)
    hp = f.hparams
    x = torch.randn(T, hp.n_embd) * 0.5
    pos = torch.arange(T, dtype=torch.int32)
    seq = torch.zeros(T, dtype=torch.int32)
    y_gpu = hip.forward(x.cuda(), pos.cuda(), seq.cuda()).cpu()
    y_cpu = cpu.forward(x.clone(), pos, seq)
    _assert_close(y_gpu, y_cpu, label=f"multi-tile T={T}")
    lg_gpu = hip.logits(y_gpu.cuda().contiguous(), all_logits=True).cpu()
    lg_cpu = cpu.logits(y_cpu, all_logits=True)
    _assert_close(lg_gpu, lg_cpu, label=f"multi-tile logits T={T}")


def test_prefill_tiling_over_max_tokens():
    """T > kMaxTokens (64) goes through host-side token tiling."""
    f, hip, cpu = _engines(n_ctx=96)
    hp = f.hparams
    T = 70
    x = torch.randn(T, hp.n_embd) * 0.5
    pos = torch.arange(T, dtype=torch.int32)
    seq = torch.zeros(T, dtype=torch.int32)
    y_gpu = hip.forward(x.cuda(), pos.cuda(), seq.cuda()).cpu()
    y_cpu = cpu.forward(x.clone(), pos, seq)
    _assert_close(y_gpu, y_cpu, label="tiling")


def test_logits_and_argmax_match():
    f, hip, cpu = _engines()
    hp = f.hparams
    T = 3
    x = torch.randn(T, hp.n_embd) * 0.5
    lg_gpu = hip.logits(x.cuda(), all_logits=True).cpu()
    lg_cpu = cpu.logits(x.clone(), all_logits=True)
    assert lg_gpu.shape == (T, hp.n_vocab)
    _assert_close(lg_gpu, lg_cpu, label="logits")
    # last-only path is the all-logits path's last row
    last = hip.logits(x.cuda(), all_logits=False).cpu()
    assert torch.allclose(last, lg_gpu[-1:], atol=1e-5)
    # argmax is exact w.r.t. the GPU's own logits
    am = hip.argmax(hip.logits(x.cuda(), all_logits=True)).cpu()
    assert torch.equal(am, lg_gpu.argmax(dim=-1).to(torch.int32))


def test_oddhead_dim():
    """head_dim not a power of two (the OpenLLaMA-3B D=100 class)."""
    f, hip, cpu = _engines(preset="tiny_oddhead")
    hp = f.hparams
    assert hp.head_dim == 24
    T = 4
    x = torch.randn(T, hp.n_embd) * 0.5
    pos = torch.arange(T, dtype=torch.int32)
    seq = torch.zeros(T, dtype=torch.int32)
    y_gpu = hip.forward(x.cuda(), pos.cuda(), seq.cuda()).cpu()
    y_cpu = cpu.forward(x.clone(), pos, seq)
    _assert_close(y_gpu, y_cpu, label="tiling")


def test_random_init_engine_runs():
    from distributedllm_amd.engine import HIPSliceEngine
    p = PRESETS["tiny"]
    hp = p.hparams(ggml.FTYPE_MOSTLY_Q4_0)
    eng = HIPSliceEngine.random(hp, n_layers=2, n_ctx=32, max_batch=2)
    x = torch.randn(2, hp.n_embd, device="cuda")
    pos = torch.zeros(2, dtype=torch.int32, device="cuda")
    seq = torch.arange(2, dtype=torch.int32, device="cuda")
    y = eng.forward(x, pos, seq)
    assert torch.isfinite(y).all()
    lg = eng.logits(y, all_logits=True)
    assert torch.isfinite(lg).all()


def test_cli_end_to_end_on_gpu(tmp_path):
    """The reference workflow (node + provision + generate + perplexity)
    with the node's engine on the MI355X — proves the GGML slice path
    loads into the HIP engine, not just the synthetic one."""
    import json
    import threading
    from distributedllm_amd.cli import execute_command
    from distributedllm_amd.cluster.node import NodeServer

    srv = NodeServer("127.0.0.1", 0, str(tmp_path / "uploads"),
                     device="cuda", n_ctx=128)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    try:
        addr = f"127.0.0.1:{srv.port}"
        root = tmp_path / "root"
        root.mkdir()
        cfg = {"model_id": "tiny_gpu", "location": "synthetic:tiny",
               "nodes_map": {addr: [0, PRESETS["tiny"].n_layer - 1]},
               "quantization": "q4_0",
               "metadata": {"name": "tinygpu", "family": "llama_v1"}}
        cfg_path = tmp_path / "cfg.json"
        cfg_path.write_text(json.dumps(cfg))
        assert execute_command(["provision", str(cfg_path),
                                "--root", str(root)]) == 0
        assert execute_command(["generate_text", str(cfg_path),
                                "--prompt", "hello", "--num-tokens", "4",
                                "--greedy", "--root", str(root)]) == 0
        assert execute_command(["perplexity", str(cfg_path),
                                "--prompt", "hello world words",
                                "--root", str(root)]) == 0
        # the node must actually be running the HIP engine
        from distributedllm_amd.cluster.client import Connection
        c = Connection("127.0.0.1", srv.port)
        assert c.get_status().device == "cuda"
        c.close()
    finally:
        srv.shutdown()
        srv.server_close()


@pytest.mark.parametrize("T", [2, 40])
def test_decode_fused_attention_matches(T):
    """decode=True (distinct sequences) runs the qkv-slab + fused-
    attention path; results must match the fp32 reference like the
    unfused path does."""
    f, hip, cpu = _engines(n_ctx=16, max_batch=64)
    hp = f.hparams
    torch.manual_seed(5)
    for step in range(3):
        x = torch.randn(T, hp.n_embd) * 0.5
        pos = torch.full((T,), step, dtype=torch.int32)
        seq = torch.arange(T, dtype=torch.int32)
        y_gpu = hip.forward(x.cuda(), pos.cuda(), seq.cuda(),
                            decode=True).cpu()
        y_cpu = cpu.forward(x.clone(), pos, seq)
        _assert_close(y_gpu, y_cpu, label=f"fused decode T={T} step {step}")


def test_pipeline_generate_on_gpu():
    """Prompt-conditioned prefill + greedy decode through the HIP engine
    (single stage) produces finite, deterministic continuations."""
    from distributedllm_amd.engine import HIPSliceEngine
    from distributedllm_amd.formats import slicer, synthetic
    from distributedllm_amd.parallel.pipeline import (
        DecodePipeline, PipelineConfig, pipeline_generate)
    f = synthetic.build_model("tiny", seed=0)
    ex = slicer.make_extra_layers(f)
    eng = HIPSliceEngine.from_ggml(f, n_ctx=64, max_batch=2)
    eng.attach_extra(ex)
    cfg = PipelineConfig(mbs=2, n_mb=1, device="cuda")
    pipe = DecodePipeline(eng, cfg, rank=0, world=1)
    toks = pipeline_generate(pipe, [5, 9, 3], max_steps=6)
    assert toks.shape == (2, 6)
    assert toks[0].tolist() == toks[1].tolist()  # same prompt -> same greedy
    # a second pipeline over the same weights reproduces the continuation
    eng2 = HIPSliceEngine.from_ggml(f, n_ctx=64, max_batch=2)
    eng2.attach_extra(ex)
    pipe2 = DecodePipeline(eng2, cfg, rank=0, world=1)
    toks2 = pipeline_generate(pipe2, [5, 9, 3], max_steps=6)
    assert toks2.tolist() == toks.tolist()


@pytest.mark.parametrize("ftype", [ggml.FTYPE_MOSTLY_Q4_0,
                                   ggml.FTYPE_MOSTLY_F16,
                                   ggml.FTYPE_MOSTLY_Q8_0])
def test_midsize_parity(ftype):
    """E=512 exercises the slab/RT kernel paths with realistic grids
    (the tiny presets run them at degenerate sizes)."""
    f, hip, cpu = _engines(preset="small", ftype=ftype, n_ctx=96,
                           max_batch=1)
    hp = f.hparams
    torch.manual_seed(7)
    T = 48
    x = torch.randn(T, hp.n_embd) * 0.5
    pos = torch.arange(T, dtype=torch.int32)
    seq = torch.zeros(T, dtype=torch.int32)
    y_gpu = hip.forward(x.cuda(), pos.cuda(), seq.cuda()).cpu()
    y_cpu = cpu.forward(x.clone(), pos, seq)
    _assert_close(y_gpu, y_cpu, label=f"midsize prefill ftype={ftype}")
    lg_gpu = hip.logits(y_gpu.cuda().contiguous(), all_logits=True).cpu()
    lg_cpu = cpu.logits(y_cpu, all_logits=True)
    _assert_close(lg_gpu, lg_cpu, label="midsize logits")
    # decode continuation with distinct sequences (fused-attention path)
    f2, hip2, cpu2 = _engines(preset="small", ftype=ftype, n_ctx=16,
                              max_batch=24)
    x = torch.randn(24, hp.n_embd) * 0.5
    pos = torch.zeros(24, dtype=torch.int32)
    seq = torch.arange(24, dtype=torch.int32)
    y_gpu = hip2.forward(x.cuda(), pos.cuda(), seq.cuda(),
                         decode=True).cpu()
    y_cpu = cpu2.forward(x.clone(), pos, seq)
    _assert_close(y_gpu, y_cpu, label="midsize fused decode")


@pytest.mark.gpu
def test_multi_stream_clones_match_shared_engine_on_gpu():
    """bench.py's single-GPU multi-stream mode: per-micro-batch
    weight-sharing clones, each stepped on its own HIP stream, must
    decode the same tokens as one shared engine holding every
    micro-batch's KV slots (no cross-stream interference)."""
    from distributedllm_amd.engine import HIPSliceEngine
    from distributedllm_amd.models.llama import PRESETS
    from distributedllm_amd.parallel.pipeline import (
        DecodePipeline, PipelineConfig)
    hp = PRESETS["tiny"].hparams(ggml.FTYPE_MOSTLY_Q4_0)
    mbs, n_mb, steps = 2, 3, 4

    def seed_tokens(pipe):
        for m in range(n_mb):
            pipe.tok[m] = torch.tensor([5 + m, 9 + 2 * m],
                                       dtype=torch.int32, device="cuda")

    # reference: one engine, partitioned KV slots, sequential micro-batches
    ref_eng = HIPSliceEngine.random(hp, n_layers=hp.n_layer, n_ctx=32,
                                    max_batch=mbs * n_mb, seed=0,
                                    with_extra=True)
    cfg = PipelineConfig(mbs=mbs, n_mb=n_mb, device="cuda")
    ref = DecodePipeline(ref_eng, cfg, rank=0, world=1)
    seed_tokens(ref)
    ref.run_steps(steps)
    torch.cuda.synchronize()
    want = ref.current_tokens().tolist()

    # same weights (seed=0), clones + concurrent streams
    eng = HIPSliceEngine.random(hp, n_layers=hp.n_layer, n_ctx=32,
                                max_batch=mbs, seed=0, with_extra=True)
    engines = [eng] + [eng.clone_shared() for _ in range(n_mb - 1)]
    pipe = DecodePipeline(eng, cfg, rank=0, world=1, engines=engines)
    seed_tokens(pipe)
    pipe.run_steps(steps)
    torch.cuda.synchronize()
    assert pipe.current_tokens().tolist() == want


@pytest.mark.gpu
def test_clone_shared_from_checkpoint():
    """clone_shared also works for engines loaded from GGML bytes (real
    multi-stream serving, not just synthetic benches): the clone shares
    the weight tensors and reproduces the original's forward exactly."""
    from distributedllm_amd.engine import HIPSliceEngine
    from distributedllm_amd.formats import slicer, synthetic
    f = synthetic.build_model("tiny", seed=3)
    ex = slicer.make_extra_layers(f)
    eng = HIPSliceEngine.from_ggml(f, n_ctx=32, max_batch=2)
    eng.attach_extra(ex)
    twin = eng.clone_shared()
    assert twin.has_extra
    x = torch.randn(2, f.hparams.n_embd, device="cuda") * 0.5
    pos = torch.zeros(2, dtype=torch.int32, device="cuda")
    seq = torch.arange(2, dtype=torch.int32, device="cuda")
    y0 = eng.forward(x.clone(), pos, seq, decode=True)
    y1 = twin.forward(x.clone(), pos, seq, decode=True)
    assert torch.equal(y0, y1)
    assert torch.equal(eng.logits(y0, all_logits=True),
                       twin.logits(y1, all_logits=True))


def test_continuous_batcher_on_gpu():
    """ContinuousBatcher over the HIP engine (fused decode path, slot
    reuse) decodes each request exactly as the canonical
    one-at-a-time loop on the same weights."""
    from distributedllm_amd.engine import HIPSliceEngine
    from distributedllm_amd.formats import slicer, synthetic
    from distributedllm_amd.serving import ContinuousBatcher
    f = synthetic.build_model("tiny", seed=0)
    ex = slicer.make_extra_layers(f)

    def engine(mb):
        e = HIPSliceEngine.from_ggml(f, n_ctx=32, max_batch=mb)
        e.attach_extra(ex)
        return e

    prompts, steps = [[5, 9, 3], [7], [2, 11, 4, 6, 1]], [4, 3, 5]

    def canonical(prompt, n):
        eng = engine(1)
        out, cur, n_past = [], list(prompt), 0
        for _ in range(n):
            toks = torch.tensor(cur, dtype=torch.int32, device="cuda")
            pos = torch.arange(n_past, n_past + len(cur),
                               dtype=torch.int32, device="cuda")
            seq = torch.zeros(len(cur), dtype=torch.int32, device="cuda")
            y = eng.forward(eng.embed(toks), pos, seq)
            lg = eng.logits(y[-1:].contiguous(), all_logits=True)
            out.append(int(torch.argmax(lg[0]).item()))
            n_past += len(cur)
            cur = [out[-1]]
        return out

    bat = ContinuousBatcher(engine(2), max_slots=2)  # forces slot reuse
    reqs = [bat.submit(p, s) for p, s in zip(prompts, steps)]
    bat.run_all(max_steps=50)
    torch.cuda.synchronize()
    for r, p, s in zip(reqs, prompts, steps):
        assert r.done and r.out == canonical(p, s)


def test_slice_chain_matches_monolith_on_gpu():
    """Two chained slice engines (layers [0,0] and [1,2] of the
    3-layer tiny model, activations handed between them exactly like
    pipeline ranks) must reproduce the monolithic engine's logits —
    the single-GPU version of SURVEY §4's slice-vs-monolith parity.
    Not necessarily bit-exact: the boundary regenerates the sumsq
    side-channel with a different reduction order (k_prep_x vs the
    fused k_reduce_prep), a ~1-ulp rsqrt difference; assert at a
    tolerance far below any layout bug."""
    from distributedllm_amd.engine import HIPSliceEngine
    from distributedllm_amd.formats import slicer, synthetic
    f = synthetic.build_model("tiny", seed=0)
    ex = slicer.make_extra_layers(f)

    mono = HIPSliceEngine.from_ggml(f, n_ctx=32, max_batch=2)
    mono.attach_extra(ex)
    s0 = HIPSliceEngine.from_ggml(slicer.make_slice(f, 0, 0), n_ctx=32,
                                  max_batch=2)
    s1 = HIPSliceEngine.from_ggml(
        slicer.make_slice(f, 1, f.hparams.n_layer - 1), n_ctx=32,
        max_batch=2)
    s0.attach_extra(ex)  # embeds on the first stage
    s1.attach_extra(ex)  # lm-head on the last stage

    toks = torch.tensor([5, 9], dtype=torch.int32, device="cuda")
    pos = torch.zeros(2, dtype=torch.int32, device="cuda")
    seq = torch.arange(2, dtype=torch.int32, device="cuda")

    y_mono = mono.forward(mono.embed(toks), pos, seq, decode=True)
    lg_mono = mono.logits(y_mono, all_logits=True)

    x = s0.forward(s0.embed(toks), pos.clone(), seq, decode=True)
    y_chain = s1.forward(x, pos.clone(), seq, decode=True)
    lg_chain = s1.logits(y_chain, all_logits=True)

    _assert_close(lg_chain, lg_mono, rel_rms=1e-3, rel_max=1e-2,
                  label="slice chain vs monolith")


def test_native_prefill_path_matches_cpu():
    """T > 64 takes the native large-M prefill path (k_qkv16_mt /
    k_gemm16_mt / k_ffn16_mt — XCD-grouped token tiles, q4 read
    directly, the layer loop sequenced in C++) — must match the fp32
    CPU reference, and the kernel decode path must attend seamlessly
    over the KV rows it wrote."""
    f, hip, cpu = _engines(preset="small", n_ctx=512, max_batch=2)
    hp = f.hparams
    assert hip._mfma_path()
    torch.manual_seed(11)
    T = 400  # > 64, not a multiple of 64 -> exercises the pad tiles
    x = torch.randn(T, hp.n_embd) * 0.5
    pos = torch.arange(T, dtype=torch.int32)
    seq = torch.zeros(T, dtype=torch.int32)
    y_gpu = hip.forward(x.cuda(), pos.cuda(), seq.cuda()).cpu()
    y_cpu = cpu.forward(x.clone(), pos, seq)
    _assert_close(y_gpu, y_cpu, label="native prefill")
    # decode the next token through the decode path on the same KV
    xd = torch.randn(1, hp.n_embd) * 0.5
    pd = torch.tensor([T], dtype=torch.int32)
    sd = torch.zeros(1, dtype=torch.int32)
    y2_gpu = hip.forward(xd.cuda(), pd.cuda(), sd.cuda(),
                         decode=True).cpu()
    y2_cpu = cpu.forward(xd.clone(), pd, sd)
    _assert_close(y2_gpu, y2_cpu, label="decode after native prefill")


def test_native_prefill_mixed_stream_matches_cpu():
    """A mixed multi-span admission stream (two prompts of different
    sequences concatenated in one call) runs through the SAME native
    prefill kernels — per-token pos/seq indexing, no span special-
    casing (the round-1 rocBLAS path was single-span only)."""
    f, hip, cpu = _engines(preset="small", n_ctx=512, max_batch=2)
    hp = f.hparams
    torch.manual_seed(12)
    n0, n1 = 200, 80
    x = torch.randn(n0 + n1, hp.n_embd) * 0.5
    pos = torch.cat([torch.arange(n0), torch.arange(n1)]).to(torch.int32)
    seq = torch.cat([torch.zeros(n0), torch.ones(n1)]).to(torch.int32)
    y_gpu = hip.forward(x.cuda(), pos.cuda(), seq.cuda()).cpu()
    y_cpu = cpu.forward(x.clone(), pos, seq)
    _assert_close(y_gpu, y_cpu, label="mixed-stream native prefill")
    # single-span continuation of sequence 1 against the same KV
    x2 = torch.randn(300, hp.n_embd) * 0.5
    pos2 = torch.arange(n1, n1 + 300, dtype=torch.int32)
    seq2 = torch.ones(300, dtype=torch.int32)
    y2_gpu = hip.forward(x2.cuda(), pos2.cuda(), seq2.cuda()).cpu()
    y2_cpu = cpu.forward(x2.clone(), pos2, seq2)
    _assert_close(y2_gpu, y2_cpu, label="prefill continuation")


@pytest.mark.parametrize("ftype", [ggml.FTYPE_MOSTLY_Q8_0,
                                   ggml.FTYPE_MOSTLY_F16])
def test_native_prefill_other_wtypes(ftype):
    """The byte-stream (W_Q8B) and f16 tile paths run the same _mt
    prefill kernels."""
    f, hip, cpu = _engines(preset="small", ftype=ftype, n_ctx=256,
                           max_batch=1)
    hp = f.hparams
    torch.manual_seed(13)
    T = 130
    x = torch.randn(T, hp.n_embd) * 0.5
    pos = torch.arange(T, dtype=torch.int32)
    seq = torch.zeros(T, dtype=torch.int32)
    y_gpu = hip.forward(x.cuda(), pos.cuda(), seq.cuda()).cpu()
    y_cpu = cpu.forward(x.clone(), pos, seq)
    _assert_close(y_gpu, y_cpu, label=f"native prefill ftype={ftype}")


@pytest.mark.parametrize("preset", ["tiny_gqa", "small_gqa"])
def test_gqa_hip_engine_matches_cpu(preset):
    """GQA end-to-end on the HIP engine: Ekv-wide KV cache, kv-head
    mapping in attention (decode fused + prefill MFMA paths), wk/wv
    with Hkv*D rows — vs the fp32 torch reference."""
    f, hip, cpu = _engines(preset=preset, n_ctx=256, max_batch=4)
    hp = f.hparams
    assert hp.is_gqa
    torch.manual_seed(21)
    # multi-token prefill (<=64: decode=False slab/fused kernels)
    T = 24
    x = torch.randn(T, hp.n_embd) * 0.5
    pos = torch.arange(T, dtype=torch.int32)
    seq = torch.zeros(T, dtype=torch.int32)
    y_gpu = hip.forward(x.cuda(), pos.cuda(), seq.cuda()).cpu()
    y_cpu = cpu.forward(x.clone(), pos, seq)
    _assert_close(y_gpu, y_cpu, label=f"{preset} gqa prefill")
    # batched fused decode (distinct sequences)
    xd = torch.randn(4, hp.n_embd) * 0.5
    pd = torch.tensor([T, 0, 0, 0], dtype=torch.int32)
    sd = torch.arange(4, dtype=torch.int32)
    y2_gpu = hip.forward(xd.cuda(), pd.cuda(), sd.cuda(),
                         decode=True).cpu()
    y2_cpu = cpu.forward(xd.clone(), pd, sd)
    _assert_close(y2_gpu, y2_cpu, label=f"{preset} gqa fused decode")
    # native large-T prefill (_mt kernels + MFMA flash attention)
    T2 = 100
    x3 = torch.randn(T2, hp.n_embd) * 0.5
    p3 = torch.arange(T + 1, T + 1 + T2, dtype=torch.int32)
    s3 = torch.zeros(T2, dtype=torch.int32)
    y3_gpu = hip.forward(x3.cuda(), p3.cuda(), s3.cuda()).cpu()
    y3_cpu = cpu.forward(x3.clone(), p3, s3)
    _assert_close(y3_gpu, y3_cpu, label=f"{preset} gqa native prefill")


def test_gqa_random_engine_runs():
    """HIPSliceEngine.random builds a GQA engine straight on the GPU
    (the bench path for llama2_70b-class models)."""
    from distributedllm_amd.engine import HIPSliceEngine
    from distributedllm_amd.models.llama import PRESETS
    p = PRESETS["small_gqa"]
    hp = p.hparams(ggml.FTYPE_MOSTLY_Q4_0)
    eng = HIPSliceEngine.random(hp, n_layers=p.n_layer, n_ctx=64,
                                max_batch=2, seed=0)
    toks = torch.tensor([3, 7], dtype=torch.int32, device="cuda")
    x = eng.embed(toks)
    pos = torch.zeros(2, dtype=torch.int32, device="cuda")
    seq = torch.arange(2, dtype=torch.int32, device="cuda")
    y = eng.forward(x, pos, seq, decode=True)
    lg = eng.logits(y, all_logits=True)
    assert torch.isfinite(y).all() and torch.isfinite(lg).all()


@pytest.mark.parametrize("ftype", [ggml.FTYPE_MOSTLY_Q4_K_M,
                                   ggml.FTYPE_MOSTLY_Q5_K_M,
                                   ggml.FTYPE_MOSTLY_Q6_K,
                                   ggml.FTYPE_MOSTLY_Q3_K_M,
                                   ggml.FTYPE_MOSTLY_Q2_K])
def test_kquant_hip_engine_matches_cpu(ftype):
    """k-quant weights through the MFMA byte path: q4_K/q5_K expand to
    per-32 (alpha, beta) byte blocks (W_Q8B); q2_K/q3_K/q6_K to the
    per-16 two-plane form (W_Q8B16). Decode, fused decode and the
    native large-T prefill all must match the fp32 reference of the
    same dequantized weights."""
    f, hip, cpu = _engines(preset="small_k", ftype=ftype, n_ctx=256,
                           max_batch=2)
    hp = f.hparams
    torch.manual_seed(31)
    T = 24
    x = torch.randn(T, hp.n_embd) * 0.5
    pos = torch.arange(T, dtype=torch.int32)
    seq = torch.zeros(T, dtype=torch.int32)
    y_gpu = hip.forward(x.cuda(), pos.cuda(), seq.cuda()).cpu()
    y_cpu = cpu.forward(x.clone(), pos, seq)
    _assert_close(y_gpu, y_cpu, label=f"kquant prefill ftype={ftype}")
    # fused batched decode
    xd = torch.randn(2, hp.n_embd) * 0.5
    pd = torch.tensor([T, 0], dtype=torch.int32)
    sd = torch.arange(2, dtype=torch.int32)
    y2 = hip.forward(xd.cuda(), pd.cuda(), sd.cuda(), decode=True).cpu()
    y2c = cpu.forward(xd.clone(), pd, sd)
    _assert_close(y2, y2c, label=f"kquant decode ftype={ftype}")
    # native large-T prefill on sequence 1
    T2 = 100
    x3 = torch.randn(T2, hp.n_embd) * 0.5
    p3 = torch.arange(1, 1 + T2, dtype=torch.int32)
    s3 = torch.ones(T2, dtype=torch.int32)
    y3 = hip.forward(x3.cuda(), p3.cuda(), s3.cuda()).cpu()
    y3c = cpu.forward(x3.clone(), p3, s3)
    _assert_close(y3, y3c, label=f"kquant native prefill ftype={ftype}")


def test_wide_decode_matches_tiled():
    """decode with T > 64 (the wide batched-decode path through the _mt
    kernels) must match the same step run as two <=64-token decode
    calls on identical weights."""
    f = synthetic.build_model("small", ftype=ggml.FTYPE_MOSTLY_Q4_0,
                              seed=5)
    ex = slicer.make_extra_layers(f)
    from distributedllm_amd.engine import HIPSliceEngine
    a = HIPSliceEngine.from_ggml(f, n_ctx=64, max_batch=96)
    b = HIPSliceEngine.from_ggml(f, n_ctx=64, max_batch=96)
    a.attach_extra(ex)
    b.attach_extra(ex)
    hp = f.hparams
    torch.manual_seed(41)
    T = 96
    x = (torch.randn(T, hp.n_embd) * 0.5).cuda()
    pos = torch.zeros(T, dtype=torch.int32, device="cuda")
    seq = torch.arange(T, dtype=torch.int32, device="cuda")
    y_wide = a.forward(x.clone(), pos, seq, decode=True)
    outs = [b.forward(x[i:i + 48].clone(), pos[i:i + 48],
                      seq[i:i + 48].contiguous(), decode=True)
            for i in (0, 48)]
    _assert_close(y_wide.cpu(), torch.cat(outs).cpu(),
                  label="wide vs tiled decode")


def test_argmax_wide_rows():
    """Device argmax must cover EVERY row for T > 64 (rows >= 64 were
    uninitialized before the wide-serving fix — embed of the garbage
    ids faulted the GPU; tools/debug_wide.py is the repro)."""
    from distributedllm_amd.engine import HIPSliceEngine
    p = PRESETS["small"]
    hp = p.hparams(ggml.FTYPE_MOSTLY_Q4_0)
    eng = HIPSliceEngine.random(hp, n_layers=1, n_ctx=16, max_batch=1,
                                seed=0, max_prefill=256)
    torch.manual_seed(3)
    lg = torch.randn(200, hp.n_vocab, device="cuda")
    ids = eng.argmax(lg)
    assert torch.equal(ids.cpu(),
                       lg.argmax(dim=-1).to(torch.int32).cpu())


def test_speculative_pld_exact_on_hip():
    """Prompt-lookup speculative decoding on the HIP engine must emit
    exactly the tokens sequential greedy emits (the drafts ride the
    mixed-admission prefill path; rejections leave stale KV rows that
    must never be read — serving/speculative.py)."""
    from distributedllm_amd.engine import HIPSliceEngine
    from distributedllm_amd.formats import slicer, synthetic
    from distributedllm_amd.serving.speculative import (SpecStats,
                                                        pld_generate)

    def eng():
        f = synthetic.build_model("tiny", seed=0)
        e = HIPSliceEngine.from_ggml(f, n_ctx=128, max_batch=1)
        e.attach_extra(slicer.make_extra_layers(f))
        return e

    def greedy(e, prompt, n):
        ids = list(prompt)
        toks = torch.tensor(ids[:-1], dtype=torch.int32, device="cuda")
        pos = torch.arange(len(ids) - 1, dtype=torch.int32, device="cuda")
        e.forward(e.embed(toks), pos, torch.zeros_like(pos))
        out, cur, p = [], ids[-1], len(ids) - 1
        for _ in range(n):
            y = e.forward(
                e.embed(torch.tensor([cur], dtype=torch.int32,
                                     device="cuda")),
                torch.tensor([p], dtype=torch.int32, device="cuda"),
                torch.zeros(1, dtype=torch.int32, device="cuda"),
                decode=True)
            cur = int(e.argmax(e.logits(y, all_logits=True))[0])
            out.append(cur)
            p += 1
        return out

    for prompt in ([7, 7, 7, 7], [5, 9, 3], [4, 8, 2, 4, 8, 2, 4, 8]):
        want = greedy(eng(), prompt, 24)
        st = SpecStats()
        got = pld_generate(eng(), prompt, 24, ngram=2, k=6, stats=st)
        assert got == want, (prompt, got, want)
        assert st.forwards <= 24


def test_speculative_batcher_exact_on_hip():
    """Batched speculation on the HIP engine: per-request outputs must
    equal the plain batcher's (drafts ride the mixed-admission stream;
    serving/scheduler.py spec_ngram/spec_k)."""
    from distributedllm_amd.engine import HIPSliceEngine
    from distributedllm_amd.formats import slicer, synthetic
    from distributedllm_amd.serving import ContinuousBatcher

    def eng():
        f = synthetic.build_model("tiny", seed=0)
        e = HIPSliceEngine.from_ggml(f, n_ctx=64, max_batch=3)
        e.attach_extra(slicer.make_extra_layers(f))
        return e

    prompts = [[7, 7, 7, 7], [5, 9, 3], [4, 8, 2, 4, 8, 2, 4, 8]]
    steps = [12, 8, 10]
    plain = ContinuousBatcher(eng())
    p_reqs = [plain.submit(p, s) for p, s in zip(prompts, steps)]
    plain.run_all(max_steps=100)
    spec = ContinuousBatcher(eng(), spec_ngram=2, spec_k=4)
    s_reqs = [spec.submit(p, s) for p, s in zip(prompts, steps)]
    n = 0
    while spec.pending:
        spec.step()
        n += 1
        assert n < 100
    for pr, sr in zip(p_reqs, s_reqs):
        assert sr.done and sr.out == pr.out
    assert n < max(steps)  # drafts accepted on the repetitive request
