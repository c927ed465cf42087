#!/usr/bin/env python3
"""Probe: can RCCL form a 2-rank communicator with BOTH ranks on one GPU?

NCCL historically rejects duplicate devices in one communicator; if RCCL
on this image allows it, the whole multi-GPU pipeline (bench.py --gpus 2,
isend/irecv ordering, graph replay + comm interleave) can be shaken out
on a single leased MI355X before the driver's first 8-GPU SCALE run
(VERDICT r1 next-steps #1).

Run: torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 2 \
       tools/rccl_probe.py
Prints one line per rank: PROBE_OK / PROBE_FAIL <reason>.
"""
import os
import sys

import torch
import torch.distributed as dist


def main() -> int:
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    dev = rank % torch.cuda.device_count()
    torch.cuda.set_device(dev)
    try:
        dist.init_process_group("nccl", rank=rank, world_size=world)
        x = torch.full((4, 3200), float(rank + 1), device="cuda")
        if rank == 0:
            dist.send(x, dst=1)
            y = torch.empty_like(x)
            dist.recv(y, src=1)
            ok = bool((y == 2.0).all().item())
        else:
            y = torch.empty_like(x)
            dist.recv(y, src=0)
            ok = bool((y == 1.0).all().item())
            dist.send(x, dst=0)
        # isend/irecv pair as well (the pipeline's actual ops)
        req = dist.isend(x, dst=(rank + 1) % world)
        z = torch.empty_like(x)
        r2 = dist.irecv(z, src=(rank - 1) % world)
        req.wait()
        r2.wait()
        torch.cuda.synchronize()
        ok = ok and bool((z == float((rank - 1) % world + 1)).all().item())
        dist.barrier()
        print(f"PROBE_OK rank={rank} dev={dev} world={world} ok={ok}",
              flush=True)
        dist.destroy_process_group()
        return 0 if ok else 1
    except Exception as e:
        print(f"PROBE_FAIL rank={rank}: {type(e).__name__}: {e}",
              flush=True)
        return 1


if __name__ == "__main__":
    sys.exit(main())
