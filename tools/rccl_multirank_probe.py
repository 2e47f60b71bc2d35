#!/usr/bin/env python3
"""Minimal probe: can RCCL form a 2-rank communicator with both ranks on
one device? Run:
  HIP_VISIBLE_DEVICES=0 python -m torch.distributed.run --standalone \
      --nproc-per-node 2 tools/rccl_multirank_probe.py
"""

import os
import sys

import torch
import torch.distributed as dist


def main() -> int:
    torch.cuda.set_device(0)
    rank = int(os.environ.get("RANK", "0"))
    print(f"[rank {rank}] visible devices: {torch.cuda.device_count()}",
          flush=True)
    try:
        dist.init_process_group("nccl")
        t = torch.ones(8, device="cuda:0")
        dist.all_reduce(t)
        torch.cuda.synchronize()
        print(f"[rank {rank}] all_reduce ok: {t[0].item()}", flush=True)
        dist.destroy_process_group()
        return 0
    except Exception as e:  # noqa: BLE001
        print(f"[rank {rank}] FAILED: {type(e).__name__}: {e}", flush=True)
        return 1


if __name__ == "__main__":
    sys.exit(main())
