#!/usr/bin/env python3
"""Bisect the in-loop fwd-GEMM slowdown (in-step ~1.04 PF/s vs ~1.5
isolated; every microbench — cold/spread/flat/interleave/sustain — fails
to reproduce it). Stages:

  1. probe the gu GEMM on fresh tensors in an empty process
  2. build llama3_8b + FlatParams/FlatDDP/FlatAdamW, run 3 real steps
  3. re-probe the SAME buffers (process/allocator state bisect)
  4. probe with the model's real wgu weight
  5. run a step with CUDA events around every block's wgu fwd GEMM
     (the in-step truth, same measurement as the probe)
"""

import os
import sys
import time

import torch
import torch.nn as nn
import torch.nn.functional as F

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

FLOPS = 2.0 * 16384 * 28672 * 4096


def probe(x, w, iters=10):
    evs = [(torch.cuda.Event(enable_timing=True),
            torch.cuda.Event(enable_timing=True)) for _ in range(iters)]
    x @ w.t()
    torch.cuda.synchronize()
    for s, e in evs:
        s.record()
        x @ w.t()
        e.record()
    torch.cuda.synchronize()
    ts = [s.elapsed_time(e) / 1e3 for s, e in evs]
    return round(FLOPS / (sum(ts) / len(ts)) / 1e12, 1)


class TimedLinear(nn.Module):
    def __init__(self, lin):
        super().__init__()
        self.lin = lin
        self.pairs = []

    @property
    def weight(self):
        return self.lin.weight

    def forward(self, x):
        s = torch.cuda.Event(enable_timing=True)
        e = torch.cuda.Event(enable_timing=True)
        s.record()
        y = self.lin(x)
        e.record()
        self.pairs.append((s, e))
        return y


def main():
    dev = torch.device("cuda:0")
    x = torch.randn(16384, 4096, device=dev, dtype=torch.bfloat16)
    w = torch.randn(28672, 4096, device=dev, dtype=torch.bfloat16)
    print("1. empty-process probe:", probe(x, w), "TF/s", flush=True)

    from torchx_amd.models.llama import LlamaModel, llama3_8b
    from torchx_amd.parallel import FlatAdamW, FlatDDP, FlatParams

    cfg = llama3_8b()
    model = LlamaModel(cfg, device=dev)
    flat = FlatParams(model, dev)
    ddp = FlatDDP(flat)
    opt = FlatAdamW(flat, lr=3e-4)
    tokens = torch.randint(0, cfg.vocab_size, (4, 4096), device=dev)
    targets = torch.roll(tokens, shifts=-1, dims=1)

    def step():
        opt.zero_grad()
        loss = model(tokens, targets)
        loss.backward()
        ddp.finish()
        opt.step()

    for _ in range(3):
        step()
    torch.cuda.synchronize()
    print("2. three steps done", flush=True)
    print("3. post-model probe (same bufs):", probe(x, w), "TF/s", flush=True)
    print("4. probe with model wgu weight:",
          probe(x, model.blocks[10].wgu.weight), "TF/s", flush=True)

    for blk in model.blocks:
        blk.wgu = TimedLinear(blk.wgu)
    step()
    torch.cuda.synchronize()
    ts = [s.elapsed_time(e) / 1e3
          for blk in model.blocks for s, e in blk.wgu.pairs]
    rates = sorted(round(FLOPS / t / 1e12, 1) for t in ts)
    print("5. in-step wgu fwd GEMM TF/s: min", rates[0], "median",
          rates[len(rates) // 2], "max", rates[-1], flush=True)
    # and immediately after that step, the plain probe again:
    print("6. post-instrumented-step probe:", probe(x, w), "TF/s", flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
