#!/usr/bin/env python3
"""Per-shape GEMM efficiency probe for the Llama-3-8B step shapes.

The r2a profile shows the three GEMM passes running at very different
rates (fwd NT 1.04 PF/s, dgrad NN 1.34, wgrad split-K TN 1.52) — this
isolates each (shape, layout) pair to see whether NT is intrinsically slow
on these shapes (=> store weights transposed) or only slow in-loop
(=> cache/clock effect).

Layouts measured per shape (M tokens, N out, K in), weight W [N,K]:
  fwd-NT    x[M,K] @ W^T          (what nn.Linear does today)
  fwd-NN    x[M,K] @ Wt           (pre-transposed weight Wt=[K,N])
  dgrad-NN  dy[M,N] @ W
  dgrad-NT  dy[M,N] @ Wt^T
  wgrad-TN  dy^T[N,M] @ x         (dW [N,K])
  wgrad-TN' x^T[K,M] @ dy         (dWt [K,N])
"""

import argparse
import json
import sys
import time

import torch


def bench_mm(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


_FLUSH = None


def bench_mm_cold(fn, iters=8, warmup=2):
    """Times only the GEMM (CUDA events) with an LLC-sized flush write
    between iterations — the in-loop situation, where the weights are
    never cache-resident."""
    global _FLUSH
    if _FLUSH is None:
        _FLUSH = torch.empty(512 * 1024 * 1024 // 4, device="cuda",
                             dtype=torch.float32)  # 512 MB > 256 MB LLC
    start = torch.cuda.Event(enable_timing=True)
    end = torch.cuda.Event(enable_timing=True)
    total = 0.0
    for i in range(warmup + iters):
        _FLUSH.fill_(float(i))
        start.record()
        fn()
        end.record()
        torch.cuda.synchronize()
        if i >= warmup:
            total += start.elapsed_time(end) / 1e3
    return total / iters


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--m", type=int, default=16384)
    p.add_argument("--iters", type=int, default=20)
    p.add_argument("--cold", action="store_true",
                   help="flush the LLC between iterations (in-loop truth)")
    p.add_argument("--sustain", type=float, default=0.0,
                   help="loop the gu fwd GEMM for this many seconds and "
                        "report the rate per 0.5s window (detects clock "
                        "throttling under sustained load)")
    p.add_argument("--interleave", action="store_true",
                   help="alternate the gu GEMM with non-GEMM kernels and "
                        "time only the GEMM (kernel-transition effects)")
    p.add_argument("--flat", action="store_true",
                   help="weights as views into one flat buffer "
                        "(FlatParams layout)")
    p.add_argument("--spread", action="store_true",
                   help="cycle 32 distinct weight + 8 activation buffers "
                        "(~9 GB working set) like a real 32-layer step — "
                        "detects TLB/working-set effects the single-buffer "
                        "probe misses")
    args = p.parse_args()
    M = args.m

    if args.spread:
        dev = torch.device("cuda:0")
        N, K = 28672, 4096
        ws = [torch.randn(N, K, device=dev, dtype=torch.bfloat16)
              for _ in range(32)]
        xs = [torch.randn(M, K, device=dev, dtype=torch.bfloat16)
              for _ in range(8)]
        flops = 2.0 * M * N * K
        idx = [0]

        def fn():
            i = idx[0]
            idx[0] += 1
            return xs[i % 8] @ ws[i % 32].t()

        dt = bench_mm(fn, iters=64, warmup=8)
        print(json.dumps({"mode": "spread32w8x",
                          "gu_fwd_TFs": round(flops / dt / 1e12, 1)}))
        return 0

    if args.interleave:
        # alternate the gu fwd GEMM with non-GEMM kernels (rmsnorm /
        # attention / a memory sweep) and time ONLY the GEMM via events —
        # reproduces the in-step situation where fwd GEMMs run at ~1.05
        # PF/s vs 1.5 isolated. If rates drop here, the gap is a
        # power/clock state transition between kernel types.
        import os as _os
        import sys as _sys
        _sys.path.insert(0, _os.path.dirname(_os.path.dirname(
            _os.path.abspath(__file__))))
        from torchx_amd import ops

        dev = torch.device("cuda:0")
        N, K = 28672, 4096
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
        flops = 2.0 * M * N * K
        rx = torch.randn(16384, 4096, device=dev, dtype=torch.bfloat16)
        rw = torch.randn(4096, device=dev, dtype=torch.bfloat16)
        qkv = torch.randn(4, 512, 64 * 128, device=dev, dtype=torch.bfloat16)
        cos, sin = ops.rope_tables(512, 128, device=dev)
        sweep = torch.empty(2 << 30, device=dev, dtype=torch.uint8)

        fillers = {
            "none": lambda: None,
            "rmsnorm": lambda: ops.rmsnorm(rx, rw),
            "attn": lambda: ops.fused_attention_qkv(qkv, cos, sin, 32, 16),
            "memsweep": lambda: sweep.fill_(1),
        }
        start = torch.cuda.Event(enable_timing=True)
        end = torch.cuda.Event(enable_timing=True)
        for fname, filler in fillers.items():
            total = 0.0
            iters, warmup = 16, 4
            for i in range(warmup + iters):
                filler()
                start.record()
                x @ w.t()
                end.record()
                torch.cuda.synchronize()
                if i >= warmup:
                    total += start.elapsed_time(end) / 1e3
            rate = flops / (total / iters) / 1e12
            print(json.dumps({"filler": fname,
                              "gu_fwd_TFs": round(rate, 1)}), flush=True)
        return 0

    if args.sustain > 0:
        dev = torch.device("cuda:0")
        N, K = 28672, 4096
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
        flops = 2.0 * M * N * K
        t_end = time.perf_counter() + args.sustain
        window_t0 = time.perf_counter()
        n = 0
        while time.perf_counter() < t_end:
            x @ w.t()
            n += 1
            if n % 50 == 0:
                torch.cuda.synchronize()
                now = time.perf_counter()
                rate = flops * 50 / (now - window_t0) / 1e12
                print(json.dumps({"t": round(now - (t_end - args.sustain), 2),
                                  "gu_fwd_TFs": round(rate, 1)}), flush=True)
                window_t0 = now
        return 0

    shapes = [
        ("qkv", 6144, 4096),
        ("wo", 4096, 4096),
        ("gu", 28672, 4096),
        ("down", 4096, 14336),
        ("lm_head", 128256, 4096),
    ]

    dev = torch.device("cuda:0")
    flat = None
    if args.flat:
        # weights as views into one big flat allocation (the FlatParams
        # situation in the model) — catches pointer-offset effects
        total = sum(N * K for _, N, K in shapes)
        flat = torch.randn(total, device=dev, dtype=torch.bfloat16)
    results = []
    off = 0
    for name, N, K in shapes:
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        if flat is not None:
            w = flat[off:off + N * K].view(N, K)
            off += N * K
        else:
            w = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
        wt = w.t().contiguous()  # [K, N]
        dy = torch.randn(M, N, device=dev, dtype=torch.bfloat16)
        flops = 2.0 * M * N * K

        try:
            sys.path.insert(0, __import__("os").path.dirname(
                __import__("os").path.dirname(
                    __import__("os").path.abspath(__file__))))
            from torchx_amd import ops as _txops
            _hip = _txops.hip_ops(required=True)
        except Exception:
            _hip = None

        def wgrad_via_t():
            # transpose dy with the LDS-tiled kernel, then run wgrad as a
            # plain NN GEMM (includes the transpose cost)
            dyt = _hip.transpose_bf16(dy)
            return dyt @ x

        cases = {
            "fwd-NT": lambda: x @ w.t(),
            "fwd-NN": lambda: x @ wt,
            "dgrad-NN": lambda: dy @ w,
            "dgrad-NT": lambda: dy @ wt.t(),
            "wgrad-TN": lambda: dy.t() @ x,
            "wgrad-TNp": lambda: x.t() @ dy,
        }
        if _hip is not None:
            cases["wgrad-viaT"] = wgrad_via_t
        row = {"shape": name, "M": M, "N": N, "K": K,
               "mode": ("cold" if args.cold else "hot")
                       + ("+flat" if args.flat else "")}
        for cname, fn in cases.items():
            if args.cold:
                dt = bench_mm_cold(fn)
            else:
                dt = bench_mm(fn, iters=args.iters)
            row[cname] = round(flops / dt / 1e12, 1)  # TF/s
        results.append(row)
        print(json.dumps(row), flush=True)
        del x, w, wt, dy
        if flat is None:
            torch.cuda.empty_cache()

    # totals at current vs best-layout assignment
    tot_cur = tot_best = 0.0
    for r in results:
        fl = 2.0 * r["M"] * r["N"] * r["K"] / 1e12
        tot_cur += fl / r["fwd-NT"] + fl / r["dgrad-NN"] + fl / r["wgrad-TN"]
        best_fwd = max(r["fwd-NT"], r["fwd-NN"])
        best_dgrad = max(r["dgrad-NN"], r["dgrad-NT"])
        best_wgrad = max(r["wgrad-TN"], r["wgrad-TNp"])
        tot_best += fl / best_fwd + fl / best_dgrad + fl / best_wgrad
    print(json.dumps({"per_layer_ms_current": tot_cur * 1e3,
                      "per_layer_ms_best": tot_best * 1e3}))
    return 0


if __name__ == "__main__":
    sys.exit(main())
