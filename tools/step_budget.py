#!/usr/bin/env python3
"""Honest per-category step budget via CUDA events (rocprofv3's
kernel-trace inflates at least the fwd GEMM times ~40% — see
tools/gemm_inbench.py — so attribution here uses events only).

Wraps every nn.Linear forward and every hip-op call with event pairs,
then times one llama3_8b step phase by phase. Event pairs measure stream
time between records == kernel time for single-kernel calls.
"""

import os
import sys
from collections import defaultdict

import torch
import torch.nn as nn

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

PAIRS = defaultdict(list)


def timed(name, fn):
    def wrapper(*a, **kw):
        s = torch.cuda.Event(enable_timing=True)
        e = torch.cuda.Event(enable_timing=True)
        s.record()
        out = fn(*a, **kw)
        e.record()
        PAIRS[name].append((s, e))
        return out

    return wrapper


class TimedLinear(nn.Module):
    def __init__(self, lin, name):
        super().__init__()
        self.lin = lin
        self.name = name

    @property
    def weight(self):
        return self.lin.weight

    def forward(self, x):
        from torchx_amd import ops as _ops

        s = torch.cuda.Event(enable_timing=True)
        e = torch.cuda.Event(enable_timing=True)
        s.record()
        # same path the model takes (fast_linear: NT fwd, NT dgrad)
        y = _ops.fast_linear(x, self.lin.weight)
        e.record()
        PAIRS[self.name].append((s, e))
        return y


def main():
    dev = torch.device("cuda:0")
    from torchx_amd import ops
    from torchx_amd.models.llama import LlamaModel, llama3_8b
    from torchx_amd.parallel import FlatAdamW, FlatDDP, FlatParams

    hip = ops.hip_ops(required=True)
    for name in ["rope_qkv", "attn_fwd_qkv", "attn_bwd_qkv", "rmsnorm_fwd",
                 "rmsnorm_bwd", "swiglu_gu_fwd", "swiglu_gu_bwd", "ce_fwd",
                 "ce_bwd", "adamw_step"]:
        setattr(hip, name, timed(name, getattr(hip, name)))

    ops.fused_linear_cross_entropy = timed("fused_ce_region",
                                           ops.fused_linear_cross_entropy)

    cfg = llama3_8b()
    model = LlamaModel(cfg, device=dev)
    flat = FlatParams(model, dev)
    ddp = FlatDDP(flat)
    opt = FlatAdamW(flat, lr=3e-4)
    for blk in model.blocks:
        blk.wqkv = TimedLinear(blk.wqkv, "fwd_gemm_qkv")
        blk.wo = TimedLinear(blk.wo, "fwd_gemm_wo")
        blk.wgu = TimedLinear(blk.wgu, "fwd_gemm_gu")
        blk.wdown = TimedLinear(blk.wdown, "fwd_gemm_down")

    tokens = torch.randint(0, cfg.vocab_size, (4, 4096), device=dev)
    targets = torch.roll(tokens, shifts=-1, dims=1)

    def step(record=False):
        marks = {}

        def mark(name):
            ev = torch.cuda.Event(enable_timing=True)
            ev.record()
            marks[name] = ev

        mark("t0")
        opt.zero_grad()
        mark("zeroed")
        loss = model(tokens, targets)
        mark("fwd")
        loss.backward()
        mark("bwd")
        ddp.finish()
        opt.step()
        mark("opt")
        torch.cuda.synchronize()
        return marks

    # warmup (not recorded)
    for _ in range(2):
        step()
    PAIRS.clear()
    marks = step(record=True)

    phases = {
        "zero_grad": ("t0", "zeroed"),
        "forward": ("zeroed", "fwd"),
        "backward": ("fwd", "bwd"),
        "optimizer": ("bwd", "opt"),
    }
    total = marks["t0"].elapsed_time(marks["opt"])
    print(f"step total: {total:.1f} ms")
    for name, (a, b) in phases.items():
        print(f"  {name:10s} {marks[a].elapsed_time(marks[b]):8.1f} ms")

    print("\nper-category (events around each call):")
    cat_total = 0.0
    for name in sorted(PAIRS):
        ms = sum(s.elapsed_time(e) for s, e in PAIRS[name])
        cat_total += ms
        n = len(PAIRS[name])
        print(f"  {name:16s} {ms:8.2f} ms  ({n:4d} calls, "
              f"{ms / n * 1000:7.1f} us avg)")
    print(f"  {'[sum tracked]':16s} {cat_total:8.2f} ms  "
          f"(untracked incl. bwd GEMMs + CE-loop GEMMs + elementwise: "
          f"{total - cat_total:.1f} ms)")
    return 0


if __name__ == "__main__":
    sys.exit(main())
