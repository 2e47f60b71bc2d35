#!/usr/bin/env python3
"""Decode ds_read_b64_tr_b16 semantics (guide T10): runs the probe kernel
(B [16,32] staged row-major in LDS, fragment assembled with two tr reads)
and reports (a) whether the MFMA result matches torch, (b) for each lane,
WHICH B elements actually landed in its fragment — so a wrong hypothesis
prints the true permutation instead of just failing."""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    from torchx_amd import ops

    hip = ops.hip_ops(required=True)
    dev = torch.device("cuda:0")
    # unique identifiable values: B[k][n] = k*100 + n (exact in bf16 for
    # k<16, n<32 -> values < 1600, all integers representable)
    k_idx = torch.arange(16, device=dev).unsqueeze(1)
    n_idx = torch.arange(32, device=dev).unsqueeze(0)
    b = (k_idx * 100 + n_idx).to(torch.bfloat16)
    a = torch.randn(32, 16, device=dev).to(torch.bfloat16)

    c, raw = hip.mfma_probe_tr(a, b)
    ref = a.float() @ b.float()
    err = (c - ref).abs().max().item()
    print(f"mfma-with-tr-read max err: {err:.4f} "
          f"({'OK' if err < 2 else 'MISMATCH'})")

    # expected fragment: lane l, j -> B[(l>>5)*8 + j][l&31]
    raw = raw.float()  # [64, 8]
    bad = 0
    for lane in range(64):
        for j in range(8):
            exp = ((lane >> 5) * 8 + j) * 100 + (lane & 31)
            got = int(raw[lane, j].item())
            if got != exp:
                if bad < 16:
                    gk, gn = got // 100, got % 100
                    print(f"lane {lane:2d} j {j}: expected B[{(lane>>5)*8+j}]"
                          f"[{lane & 31}], got B[{gk}][{gn}]")
                bad += 1
    print(f"fragment mismatches: {bad}/512")
    return 0 if err < 2 and bad == 0 else 1


if __name__ == "__main__":
    sys.exit(main())
