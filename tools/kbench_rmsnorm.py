#!/usr/bin/env python3
"""RMSNorm fwd/bwd kernel microbench at the llama3_8b bench shape
(rows=16384, H=4096). Reports us/call and achieved TB/s vs the HBM
roofline (fwd: 3 tensor passes; bwd: dx 3 passes + dw re-read 2 passes)."""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def bench(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    from torchx_amd import ops

    hip = ops.hip_ops(required=True)
    dev = torch.device("cuda:0")
    R, H = 16384, 4096
    x = torch.randn(R, H, device=dev, dtype=torch.bfloat16)
    w = torch.randn(H, device=dev, dtype=torch.bfloat16)
    dy = torch.randn(R, H, device=dev, dtype=torch.bfloat16)
    y, invrms = hip.rmsnorm_fwd(x, w, 1e-5)

    t_fwd = bench(lambda: hip.rmsnorm_fwd(x, w, 1e-5))
    t_bwd = bench(lambda: hip.rmsnorm_bwd(dy, x, w, invrms))

    bytes_fwd = R * H * 2 * 3  # x read, y write (+w tiny)
    bytes_bwd = R * H * 2 * 5  # dx pass: dy+x read, dx write; dw: dy+x read
    print(f"fwd: {t_fwd*1e6:7.1f} us  {bytes_fwd/t_fwd/1e12:5.2f} TB/s")
    print(f"bwd: {t_bwd*1e6:7.1f} us  {bytes_bwd/t_bwd/1e12:5.2f} TB/s "
          f"(5-pass roofline ~{bytes_bwd/8e12*1e6:.0f} us at 8 TB/s)")

    # numerics vs fp32 reference
    xr = x.float().requires_grad_(True)
    wr = w.float().requires_grad_(True)
    yr = xr * torch.rsqrt(xr.pow(2).mean(-1, keepdim=True) + 1e-5) * wr
    yr.backward(dy.float())
    dx, dw = hip.rmsnorm_bwd(dy, x, w, invrms)
    ex = (dx.float() - xr.grad).abs().max() / xr.grad.abs().max()
    ew = (dw.float() - wr.grad).abs().max() / wr.grad.abs().max()
    print(f"rel err dx {ex.item():.3e} dw {ew.item():.3e}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
