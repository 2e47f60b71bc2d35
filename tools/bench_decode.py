#!/usr/bin/env python3
"""Serving benchmark: prefill + KV-cache decode throughput for
Llama-3-8B on one MI355X (random weights, synthetic prompt)."""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama3_8b",
                   choices=["llama3_8b", "gpu_tiny", "mixtral_8x7b",
                            "mixtral_gpu_tiny"])
    p.add_argument("--batch", type=int, default=4)
    p.add_argument("--prompt", type=int, default=512)
    p.add_argument("--new", type=int, default=128)
    p.add_argument("--graph", action="store_true",
                   help="hipGraph decode loop (whole step as one replay)")
    args = p.parse_args()

    from torchx_amd.models.generate import KVCache, prefill
    from torchx_amd.models.generate import decode_step as decode_step_dense
    from torchx_amd.models.llama import LlamaModel, llama3_8b, llama_gpu_tiny

    dev = torch.device("cuda:0")
    moe = args.model.startswith("mixtral")
    if moe:
        from torchx_amd.models.generate_moe import (
            decode_step_moe, prefill_moe,
        )
        from torchx_amd.models.mixtral import (
            MixtralModel, mixtral_8x7b, mixtral_gpu_tiny,
        )

        if args.graph:
            print("--graph is dense-only (MoE routing is data-dependent)",
                  file=sys.stderr)
            return 1
        cfg = (mixtral_8x7b() if args.model == "mixtral_8x7b"
               else mixtral_gpu_tiny())
        torch.manual_seed(0)
        model = MixtralModel(cfg, device=dev)
        prefill, decode_step = prefill_moe, decode_step_moe
    else:
        cfg = llama3_8b() if args.model == "llama3_8b" else llama_gpu_tiny()
        torch.manual_seed(0)
        model = LlamaModel(cfg, device=dev)
        decode_step = decode_step_dense
    B, S0, N = args.batch, args.prompt, args.new
    tokens = torch.randint(0, cfg.vocab_size, (B, S0), device=dev)
    caches = [KVCache.empty(cfg, B, S0 + N + 8, dev)
              for _ in range(cfg.num_layers)]

    # warm prefill once (hipBLASLt algo selection etc.), then re-time it
    # on fresh caches so the printed number is the steady-state rate
    warm = [KVCache.empty(cfg, B, S0 + N + 8, dev)
            for _ in range(cfg.num_layers)]
    prefill(model, tokens, warm)
    del warm
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    logits = prefill(model, tokens, caches)
    torch.cuda.synchronize()
    t_prefill = time.perf_counter() - t0

    nxt = logits.argmax(-1, keepdim=True)
    steps = N
    if args.graph:
        from torchx_amd.models.generate import GraphedDecoder

        dec = GraphedDecoder(model, caches, B, nxt, start_pos=S0)
        for _ in range(2):
            dec.step()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(steps):
            dec.step()
        torch.cuda.synchronize()
        t_dec = time.perf_counter() - t0
    else:
        # warm a few decode steps, then time
        for _ in range(4):
            nxt = decode_step(model, nxt, caches).argmax(-1, keepdim=True)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(steps):
            nxt = decode_step(model, nxt, caches).argmax(-1, keepdim=True)
        torch.cuda.synchronize()
        t_dec = time.perf_counter() - t0

    print(json.dumps({
        "metric": "decode_tokens_per_second",
        "value": B * steps / t_dec,
        "ms_per_decode_step": t_dec / steps * 1e3,
        "prefill_tokens_per_second": B * S0 / t_prefill,
        "batch": B, "prompt": S0, "new_tokens": steps,
        "model": args.model, "dtype": "bf16", "data": "synthetic", "graphed": bool(args.graph),
    }))
    return 0


if __name__ == "__main__":
    sys.exit(main())
