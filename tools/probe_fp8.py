#!/usr/bin/env python3
"""Probe hipBLASLt fp8 (OCP e4m3) GEMM throughput via torch._scaled_mm on
gfx950, vs bf16 torch.mm at the model's fattest shape."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def bench(fn, iters=20):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    dev = "cuda"
    M, K, N = 16384, 4096, 28672  # wgu fwd shape
    a = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    w = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
    fl = 2 * M * K * N

    t = bench(lambda: a @ w.t())
    print(f"bf16  mm: {t*1e3:8.3f} ms  {fl/t/1e12:7.1f} TF/s")

    sa = a.abs().amax() / 448.0
    sw = w.abs().amax() / 448.0
    a8 = (a / sa).to(torch.float8_e4m3fn)
    w8 = (w / sw).to(torch.float8_e4m3fn)
    # _scaled_mm wants b column-major: w8.t() is a column-major [K, N] view
    b8 = w8.t()
    out = torch._scaled_mm(a8, b8, scale_a=sa.float(), scale_b=sw.float(),
                           out_dtype=torch.bfloat16)
    t = bench(lambda: torch._scaled_mm(a8, b8, scale_a=sa.float(),
                                       scale_b=sw.float(),
                                       out_dtype=torch.bfloat16))
    print(f"fp8  smm: {t*1e3:8.3f} ms  {fl/t/1e12:7.1f} TF/s")
    # numerics sanity
    ref = (a @ w.t()).float()
    err = (out.float() - ref).abs().max() / ref.abs().max()
    print(f"fp8 rel max err vs bf16: {err.item():.4f}")

    # quantization cost (the overhead an Fp8Linear pays per call)
    t = bench(lambda: (a / sa).to(torch.float8_e4m3fn))
    print(f"quantize a: {t*1e3:8.3f} ms")


if __name__ == "__main__":
    main()
