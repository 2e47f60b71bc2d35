"""RCCL/gloo smoke payload: init_pg + one all_reduce; prints the computed
world size (parity: torchx/examples/apps/compute_world_size)."""

from __future__ import annotations

import os
import sys

import torch
import torch.distributed as dist

from torchx_amd.distributed import init_pg, rank, world_size


def compute_world_size() -> int:
    device = init_pg()
    ws = world_size()
    t = torch.zeros(ws, device=device)
    t[rank()] = 1.0
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    result = int(t.sum().item())
    print(f"rank {rank()}: computed world size = {result}", flush=True)
    if result != ws:
        raise RuntimeError(f"all_reduce disagreed: {result} != {ws}")
    return result


def main() -> int:
    if "--throws" in sys.argv:
        raise RuntimeError("injected failure (--throws)")
    compute_world_size()
    return 0


if __name__ == "__main__":
    sys.exit(main())
