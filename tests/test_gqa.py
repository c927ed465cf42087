"""GQA (grouped-query attention, llama-v2 70B class) — CPU coverage.

The GGJT v3 header cannot express n_head_kv (the reference era passed it
on the llama.cpp command line); this framework extends the format with
GGJT version 4 (n_head_kv after n_head), written ONLY for GQA models so
every MHA file stays byte-identical to the reference layouts
(slice_model.cpp:253-263 / tensor_processor.cpp:179-188).
"""
import struct
import subprocess
from pathlib import Path

import pytest
import torch

from distributedllm_amd.formats import ggml, slicer, synthetic
from distributedllm_amd.models.llama import PRESETS, LlamaSliceRef

TOOLS = Path(__file__).resolve().parent.parent / "tools" / "bin"


def test_gqa_header_roundtrip(tmp_path):
    f = synthetic.build_model("tiny_gqa", seed=3)
    p = tmp_path / "gqa.bin"
    f.save(str(p))
    head = p.read_bytes()[:8]
    magic, version = struct.unpack("<II", head)
    assert version == ggml.GGJT_VERSION_GQA
    g = ggml.GGMLFile.load(str(p), extended=False)
    assert g.hparams.n_head == 8 and g.hparams.kv_heads == 2
    assert g.hparams.n_embd_kv == 32  # 2 kv heads x head_dim 16
    # wk/wv rows are Ekv
    tm = g.tensor_map()
    assert tm["layers.0.attention.wk.weight"].shape_rows_cols == (32, 128)
    assert tm["layers.0.attention.wq.weight"].shape_rows_cols == (128, 128)


def test_mha_files_stay_version3(tmp_path):
    f = synthetic.build_model("tiny", seed=0)
    p = tmp_path / "mha.bin"
    f.save(str(p))
    _, version = struct.unpack("<II", p.read_bytes()[:8])
    assert version == ggml.GGJT_VERSION


def test_slicer_propagates_n_head_kv(tmp_path):
    f = synthetic.build_model("tiny_gqa", seed=3)
    sl = slicer.make_slice(f, 1, 2)
    assert sl.hparams.kv_heads == 2 and sl.hparams.first_layer == 1
    ex = slicer.make_extra_layers(f)
    assert ex.hparams.kv_heads == 2
    # v4 + extended (9-field) header survives a disk roundtrip
    p = tmp_path / "s.bin"
    sl.save(str(p))
    assert ggml.sniff_extended(str(p))
    back = ggml.GGMLFile.load(str(p), extended=True)
    assert back.hparams.kv_heads == 2
    assert back.hparams.first_layer == 1


def test_cpp_slicer_matches_python_on_gqa(tmp_path):
    """The native slice_model tool must produce byte-identical v4 slices
    (same extended+GQA header, same tensors)."""
    f = synthetic.build_model("tiny_gqa", ftype=ggml.FTYPE_MOSTLY_F16,
                              seed=5)
    src = tmp_path / "m.bin"
    f.save(str(src))
    out_c = tmp_path / "slice_c.bin"
    subprocess.run([str(TOOLS / "slice_model"), "slice", str(src), "1",
                    "2", str(out_c)], check=True, capture_output=True)
    out_py = tmp_path / "slice_py.bin"
    slicer.make_slice(f, 1, 2).save(str(out_py))
    assert out_c.read_bytes() == out_py.read_bytes()
    ex_c = tmp_path / "ex_c.bin"
    subprocess.run([str(TOOLS / "slice_model"), "extra_layers", str(src),
                    str(ex_c)], check=True, capture_output=True)
    ex_py = tmp_path / "ex_py.bin"
    slicer.make_extra_layers(f).save(str(ex_py))
    assert ex_c.read_bytes() == ex_py.read_bytes()


def test_torch_engine_gqa_matches_reference():
    """TorchSliceEngine's kv-head mapping == LlamaSliceRef's (the fp32
    reference applies GQA by index-expanding the kv cache)."""
    from distributedllm_amd.engine import TorchSliceEngine
    from distributedllm_amd.models.llama import weights_from_ggml
    f = synthetic.build_model("tiny_gqa", ftype=ggml.FTYPE_ALL_F32, seed=4)
    w = weights_from_ggml(f)
    hp = f.hparams
    ref = LlamaSliceRef(hp, w, first_layer=0, n_layers=hp.n_layer,
                        n_ctx=32)
    eng = TorchSliceEngine(hp, w, n_layers=hp.n_layer, first_layer=0,
                           n_ctx=32, max_batch=1)
    torch.manual_seed(2)
    n_past = 0
    for n in (3, 1, 2):  # prefill then decode-ish steps
        x = torch.randn(n, hp.n_embd) * 0.5
        y_ref = ref.forward(x.clone())
        pos = torch.arange(n_past, n_past + n, dtype=torch.int32)
        seq = torch.zeros(n, dtype=torch.int32)
        y_eng = eng.forward(x.clone(), pos, seq)
        n_past += n
        assert torch.allclose(y_eng, y_ref, atol=1e-5), (n_past, n)


def test_gqa_kv_cache_memory_shrinks():
    """The point of GQA: the kv cache is Hkv/H the size."""
    from distributedllm_amd.engine import TorchSliceEngine
    from distributedllm_amd.models.llama import weights_from_ggml
    f = synthetic.build_model("tiny_gqa", ftype=ggml.FTYPE_ALL_F32, seed=0)
    eng = TorchSliceEngine(f.hparams, weights_from_ggml(f),
                           n_layers=f.hparams.n_layer, first_layer=0,
                           n_ctx=16, max_batch=1)
    assert eng.k_cache.shape[-2] == 2  # kv heads, not 8 query heads


def test_llama2_70b_preset_dims():
    p = PRESETS["llama2_70b"]
    assert (p.n_embd, p.n_head, p.kv_heads) == (8192, 64, 8)
    assert p.n_ff == 28672  # the era formula with n_mult=28672
    assert p.n_embd_kv == 1024


_GQA_PROMPT, _GQA_STEPS = [5, 9, 3], 4


def _gqa_pipe_rank(rank, world, port, q):
    import os

    import torch.distributed as dist

    from distributedllm_amd.engine import TorchSliceEngine
    from distributedllm_amd.parallel.pipeline import (
        DecodePipeline, PipelineConfig, partition_layers,
        pipeline_generate)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    f = synthetic.build_model("tiny_gqa", ftype=ggml.FTYPE_ALL_F32,
                              seed=0)
    ex = slicer.make_extra_layers(f)
    first, count = partition_layers(f.hparams.n_layer, world)[rank]
    sl = slicer.make_slice(f, first, first + count - 1)
    eng = TorchSliceEngine.from_ggml(sl, n_ctx=32, max_batch=2 * world)
    eng.attach_extra(ex)
    pipe = DecodePipeline(eng, PipelineConfig(mbs=2, n_mb=world,
                                              device="cpu"),
                          rank=rank, world=world)
    out = pipeline_generate(pipe, _GQA_PROMPT, max_steps=_GQA_STEPS)
    if rank == 0:
        q.put(out.tolist())
    dist.destroy_process_group()


def test_gqa_pipeline_matches_single_engine():
    """A GQA model sliced across 2 gloo ranks decodes token-exactly
    like the single engine (v4 slices + Ekv KV through the pipeline)."""
    import multiprocessing as mp

    from distributedllm_amd.engine import TorchSliceEngine
    from distributedllm_amd.parallel.pipeline import (
        DecodePipeline, PipelineConfig, pipeline_generate)

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_gqa_pipe_rank, args=(r, 2, 29741, q))
             for r in range(2)]
    for p in procs:
        p.start()
    got = q.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    f = synthetic.build_model("tiny_gqa", ftype=ggml.FTYPE_ALL_F32,
                              seed=0)
    ex = slicer.make_extra_layers(f)
    eng = TorchSliceEngine.from_ggml(f, n_ctx=32, max_batch=2)
    eng.attach_extra(ex)
    pipe = DecodePipeline(eng, PipelineConfig(mbs=2, n_mb=1,
                                              device="cpu"),
                          rank=0, world=1)
    want = pipeline_generate(pipe, _GQA_PROMPT,
                             max_steps=_GQA_STEPS).tolist()
    assert all(row == want[0] for row in got)


def test_kquant_model_through_node_generate(tmp_path):
    """End to end on CPU: provision-style slice of a q6_K GQA model,
    pushed to a TCP node and generated through the cluster client —
    the full reference workflow on the round-2 formats."""
    import threading

    from distributedllm_amd.cluster.client import Connection
    from distributedllm_amd.cluster.llm_client import DistributedLLM
    from distributedllm_amd.cluster.node import NodeServer
    f = synthetic.build_model("small_k", ftype=ggml.FTYPE_MOSTLY_Q6_K,
                              seed=2)
    mp_ = tmp_path / "m.bin"
    sp = tmp_path / "s.bin"
    ep = tmp_path / "extra.bin"
    f.save(str(mp_))
    slicer.make_slice(f, 0, f.hparams.n_layer - 1).save(str(sp))
    slicer.make_extra_layers(f).save(str(ep))
    srv = NodeServer("127.0.0.1", 0, str(tmp_path / "uploads"),
                     device="cpu", n_ctx=64)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        conn = Connection("127.0.0.1", srv.port)
        conn.push_slice(str(sp), {"name": "mk"})
        conn.load_slice("mk")
        llm = DistributedLLM(
            [(conn, 0, f.hparams.n_layer - 1)], str(ep))
        out = list(llm.generate("hello", max_steps=3))
        assert len(out) == 3
        conn.close()
    finally:
        srv.shutdown()
