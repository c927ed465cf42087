"""Continuous batching across pipeline ranks == single-engine batcher.

2 gloo ranks on CPU (the same PipelineEngine/serve_forever code drives
RCCL on GPUs): per-request prompts of different lengths prefill through
both layer slices, decode steps advance all requests, greedy tokens
must equal the single-engine ContinuousBatcher exactly (f32 math,
slicing cannot change it).
"""
import multiprocessing as mp
import os

import torch

from distributedllm_amd.engine import TorchSliceEngine
from distributedllm_amd.formats import slicer, synthetic
from distributedllm_amd.parallel.pipeline import partition_layers
from distributedllm_amd.serving import ContinuousBatcher
from distributedllm_amd.serving.pipeline_server import (
    PipelineEngine, serve_forever)

PROMPTS = [[5, 9, 3], [7], [11, 2, 8, 4, 1]]
MAX_NEW = [4, 5, 3]


def _single_engine_tokens():
    f = synthetic.build_model("tiny", seed=0)
    ex = slicer.make_extra_layers(f)
    eng = TorchSliceEngine.from_ggml(f, n_ctx=32, max_batch=4)
    eng.attach_extra(ex)
    bat = ContinuousBatcher(eng)
    reqs = [bat.submit(p, m) for p, m in zip(PROMPTS, MAX_NEW)]
    bat.run_all(max_steps=64)
    return [r.out for r in reqs]


def _rank_main(rank, world, port, q):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    f = synthetic.build_model("tiny", seed=0)
    ex = slicer.make_extra_layers(f)
    parts = partition_layers(f.hparams.n_layer, world)
    first, count = parts[rank]
    sl = slicer.make_slice(f, first, first + count - 1)
    eng = TorchSliceEngine.from_ggml(sl, n_ctx=32, max_batch=4)
    eng.attach_extra(ex)
    if rank > 0:
        serve_forever(eng, rank, world)
        dist.destroy_process_group()
        return
    facade = PipelineEngine(eng, rank, world)
    bat = ContinuousBatcher(facade)
    reqs = [bat.submit(p, m) for p, m in zip(PROMPTS, MAX_NEW)]
    bat.run_all(max_steps=64)
    facade.shutdown()
    q.put([r.out for r in reqs])
    dist.destroy_process_group()


def test_pipeline_serving_matches_single_engine():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_rank_main, args=(r, 2, 29721, q))
             for r in range(2)]
    for p in procs:
        p.start()
    got = q.get(timeout=240)
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    want = _single_engine_tokens()
    assert got == want


def test_pipeline_engine_world1_is_plain():
    """world=1 facade behaves like the engine itself (smoke; the CLI
    uses the engine directly in that case)."""
    f = synthetic.build_model("tiny", seed=0)
    ex = slicer.make_extra_layers(f)
    eng = TorchSliceEngine.from_ggml(f, n_ctx=32, max_batch=2)
    eng.attach_extra(ex)
    facade = PipelineEngine(eng, 0, 1)
    bat = ContinuousBatcher(facade)
    r = bat.submit([5, 9], 3)
    bat.run_all(max_steps=16)
    assert len(r.out) == 3
