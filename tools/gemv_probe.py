#!/usr/bin/env python3
"""Per-shape decode-GEMV probe: gemv_bf16 vs torch.matmul (hipBLASLt) at
the Llama-3-8B decode shapes, CUDA-event timed, with achieved weight-
stream GB/s (the op's roof is the ~8 TB/s HBM3E read of W)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from torchx_amd.ops import hip_ops

SHAPES = [  # (M, N, K) = (batch, out-features, in-features)
    ("wqkv", 4, 6144, 4096),
    ("wo", 4, 4096, 4096),
    ("wgu", 4, 28672, 4096),
    ("wdown", 4, 4096, 14336),
    ("lm_head", 4, 128256, 4096),
]


def time_fn(fn, iters=100):
    s, e = torch.cuda.Event(True), torch.cuda.Event(True)
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1e3  # us


def main():
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    ops = hip_ops()
    tot_g = tot_m = 0.0
    for name, M, N, K in SHAPES:
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
        t_g = time_fn(lambda: ops.gemv_bf16(x, w))
        t_m = time_fn(lambda: torch.matmul(x, w.t()))
        gb = N * K * 2 / 1e9
        print(f"{name:8s} M={M} N={N:6d} K={K:5d}  W={gb*1e3:7.1f} MB  "
              f"gemv {t_g:7.1f} us ({gb/(t_g*1e-6):6.0f} GB/s)  "
              f"matmul {t_m:7.1f} us ({gb/(t_m*1e-6):6.0f} GB/s)")
        tot_g += t_g
        tot_m += t_m
    layer_g = tot_g - t_g  # last shape is lm_head
    layer_m = tot_m - t_m
    print(f"one decode step GEMV total: gemv {(layer_g*32 + t_g)/1e3:.3f} ms"
          f"  matmul {(layer_m*32 + t_m)/1e3:.3f} ms")

    # fused wgu+swiglu vs separate gemv + swiglu kernel
    M, I, K = 4, 14336, 4096
    x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    w = torch.randn(2 * I, K, device=dev, dtype=torch.bfloat16)
    t_f = time_fn(lambda: ops.gemv_swiglu_bf16(x, w))
    gb = 2 * I * K * 2 / 1e9
    print(f"wgu+swiglu fused: {t_f:7.1f} us ({gb/(t_f*1e-6):6.0f} GB/s) "
          f"vs separate gemv {time_fn(lambda: ops.gemv_bf16(x, w)):7.1f} us"
          f" + swiglu kernel")
    return 0


if __name__ == "__main__":
    sys.exit(main())
