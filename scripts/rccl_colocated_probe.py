#!/usr/bin/env python3
"""Probe: can RCCL run 2 ranks on ONE GPU?  (NCCL semantics traditionally
forbid it; if RCCL allows it, bench.py multirank rehearsals could use real
RCCL on the 1-GPU box instead of gloo.)  Run under torchrun with
--nproc-per-node 2 and an outer `timeout` — a hang means NO."""

import os
import sys

import torch
import torch.distributed as dist


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    torch.cuda.set_device(0)
    dist.init_process_group("nccl", rank=rank, world_size=world)
    t = torch.ones(4, device="cuda") * (rank + 1)
    dist.all_reduce(t)
    torch.cuda.synchronize()
    expect = sum(range(1, world + 1))
    ok = bool((t == expect).all().item())
    print(f"[rank {rank}] RCCL colocated all_reduce "
          f"{'OK' if ok else 'WRONG RESULT'}: {t.tolist()}", flush=True)
    dist.destroy_process_group()
    sys.exit(0 if ok else 1)


if __name__ == "__main__":
    main()
