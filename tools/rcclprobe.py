#!/usr/bin/env python3
"""Probe: can RCCL run world=2 with both ranks on ONE GPU (cuda:0)?

NCCL historically refuses duplicate devices in a communicator; if RCCL
allows it, a 1-GPU lease can exercise real RCCL p2p hops (pp2 pipeline,
compute serialized on the single device but transport/ordering real).
Prints PROBE_OK or the failure mode."""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def worker(rank):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29881"
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = "2"
    torch.cuda.set_device(0)          # both ranks on the same GPU
    dist.init_process_group("nccl", rank=rank, world_size=2)
    t = torch.full((8,), float(rank + 1), device="cuda:0")
    if rank == 0:
        dist.send(t, dst=1)
        r = torch.zeros(8, device="cuda:0")
        dist.recv(r, src=1)
        torch.cuda.synchronize()
        assert r.sum().item() == 16.0, r
        print("PROBE_OK: RCCL p2p world=2 on one GPU works", flush=True)
    else:
        r = torch.zeros(8, device="cuda:0")
        dist.recv(r, src=0)
        dist.send(t, dst=0)
        torch.cuda.synchronize()
        assert r.sum().item() == 8.0, r
    dist.destroy_process_group()


if __name__ == "__main__":
    try:
        mp.spawn(worker, nprocs=2, join=True)
    except Exception as e:
        print(f"PROBE_FAIL: {type(e).__name__}: {e}", flush=True)
        sys.exit(1)
