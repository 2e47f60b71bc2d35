#!/usr/bin/env python3
"""Attention kernel microbenchmark: TF/s for fwd and bwd at the bench shape
(B=4, S=4096, Hq=32, Hkv=8, D=128, causal bf16)."""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from torchx_amd import ops


def flops_fwd(B, S, Hq, D, causal):
    f = 2 * 2 * B * Hq * S * S * D  # QK^T + PV
    return f // 2 if causal else f


def time_fn(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--B", type=int, default=4)
    p.add_argument("--S", type=int, default=4096)
    p.add_argument("--Hq", type=int, default=32)
    p.add_argument("--Hkv", type=int, default=8)
    p.add_argument("--iters", type=int, default=20)
    p.add_argument("--mode", type=str, default="both",
                   choices=["fwd", "bwd", "both"])
    args = p.parse_args()
    B, S, Hq, Hkv, D = args.B, args.S, args.Hq, args.Hkv, 128

    dev = torch.device("cuda", 0)
    torch.manual_seed(0)
    q = torch.randn(B, S, Hq, D, device=dev, dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16, requires_grad=True)
    v = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16, requires_grad=True)
    do = torch.randn(B, S, Hq, D, device=dev, dtype=torch.bfloat16)

    hip = ops.hip_ops(required=True)
    scale = D ** -0.5

    def fwd():
        return hip.attn_fwd(q, k, v, scale, True)

    o, lse = fwd()
    if args.mode in ("fwd", "both"):
        t_fwd = time_fn(fwd, args.iters)
        tf_fwd = flops_fwd(B, S, Hq, D, True) / t_fwd / 1e12
        print(f"fwd: {t_fwd*1e3:.3f} ms  {tf_fwd:.0f} TF/s")

    if args.mode in ("bwd", "both"):
        def bwd():
            return hip.attn_bwd(q, k, v, o, do, lse, scale, True)

        t_bwd = time_fn(bwd, args.iters)
        tf_bwd = flops_fwd(B, S, Hq, D, True) * 2.5 / t_bwd / 1e12
        print(f"bwd: {t_bwd*1e3:.3f} ms  {tf_bwd:.0f} TF/s")


if __name__ == "__main__":
    main()
