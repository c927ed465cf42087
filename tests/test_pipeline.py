"""Pipeline correctness: gloo 2-rank layer-sliced decode == single engine.

Runs here on CPU (world_size=2, backend gloo, 127.0.0.1 rendezvous); the
same DecodePipeline code path is what bench.py runs over RCCL on GPUs.
"""
import json
import multiprocessing as mp
import os

import pytest
import torch

from distributedllm_amd.engine import TorchSliceEngine
from distributedllm_amd.formats import slicer, synthetic
from distributedllm_amd.parallel.pipeline import (
    DecodePipeline, PipelineConfig, partition_layers)

STEPS = 3
MBS = 2


def test_partition_layers():
    assert partition_layers(26, 8) == [
        (0, 4), (4, 4), (8, 3), (11, 3), (14, 3), (17, 3), (20, 3), (23, 3)]
    assert partition_layers(3, 1) == [(0, 3)]
    assert partition_layers(4, 2) == [(0, 2), (2, 2)]


def _single_reference_tokens():
    f = synthetic.build_model("tiny", seed=0)
    ex = slicer.make_extra_layers(f)
    eng = TorchSliceEngine.from_ggml(f, n_ctx=32, max_batch=MBS)
    eng.attach_extra(ex)
    cfg = PipelineConfig(mbs=MBS, n_mb=1, device="cpu")
    pipe = DecodePipeline(eng, cfg, rank=0, world=1)
    # deterministic starting tokens
    pipe.tok[0] = torch.tensor([5, 9], dtype=torch.int32)
    pipe.run_steps(STEPS)
    return pipe.current_tokens()[0].tolist()


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
    eng = TorchSliceEngine.from_ggml(sl, n_ctx=32, max_batch=MBS)
    eng.attach_extra(ex)
    cfg = PipelineConfig(mbs=MBS, n_mb=1, device="cpu")
    pipe = DecodePipeline(eng, cfg, rank=rank, world=world)
    pipe.tok[0] = torch.tensor([5, 9], dtype=torch.int32)
    pipe.run_steps(STEPS)
    if rank == 0:
        q.put(pipe.current_tokens()[0].tolist())
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_two_stage_pipeline_matches_single():
    ref = _single_reference_tokens()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29531
    procs = [ctx.Process(target=_rank_main, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    got = q.get(timeout=100)
    for p in procs:
        p.join(timeout=30)
        assert p.exitcode == 0
    assert got == ref
