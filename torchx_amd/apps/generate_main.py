"""Reference serving app: KV-cache generation on the CDNA4 decode path.

The deployment/serving counterpart of ``apps/trainer.py`` — the reference
(a launcher) ships trainer-style example apps only
(torchx/examples/apps/lightning/train.py); this app exposes the
MI355X-native serving stack (skinny-M GEMV linears, fused
rope+cache-append, split-K flash-decode attention, optional hipGraph
decode loop) behind the same "launchable app" convention, so
``torchx run utils.python -m torchx_amd.apps.generate_main`` serves a
model the way ``dist.ddp -m torchx_amd.apps.trainer`` trains one.

Synthetic prompts + random-init weights (no network in the target
environment); prints one JSON line with decode/prefill throughput.
"""

from __future__ import annotations

import argparse
import json
import sys
import time

import torch


def parse_args(argv):
    p = argparse.ArgumentParser(description="torchx_amd serving app")
    p.add_argument("--model", default="llama3_8b",
                   choices=["llama3_8b", "gpu_tiny", "tiny",
                            "mixtral_8x7b", "mixtral_tiny"])
    p.add_argument("--batch", type=int, default=4)
    p.add_argument("--prompt-len", type=int, default=512)
    p.add_argument("--new-tokens", type=int, default=64)
    p.add_argument("--temperature", type=float, default=0.0)
    p.add_argument("--top-k", type=int, default=None)
    p.add_argument("--graph", action="store_true",
                   help="hipGraph decode loop (GPU, greedy only)")
    p.add_argument("--seed", type=int, default=0)
    return p.parse_args(argv)


def main(argv=None) -> int:
    args = parse_args(argv)
    from torchx_amd.models.generate import generate, generate_graphed
    from torchx_amd.models.generate_moe import generate_moe
    from torchx_amd.models.llama import (
        LlamaModel, llama3_8b, llama_gpu_tiny, llama_tiny,
    )
    from torchx_amd.models.mixtral import (
        MixtralModel, mixtral_8x7b, mixtral_tiny,
    )

    moe = args.model.startswith("mixtral")
    cfg = {"llama3_8b": llama3_8b, "gpu_tiny": llama_gpu_tiny,
           "tiny": llama_tiny, "mixtral_8x7b": mixtral_8x7b,
           "mixtral_tiny": mixtral_tiny}[args.model]()
    use_gpu = torch.cuda.is_available()
    dev = torch.device("cuda:0" if use_gpu else "cpu")
    torch.manual_seed(args.seed)
    cls = MixtralModel if moe else LlamaModel
    model = cls(cfg, device=dev if use_gpu else None)
    tokens = torch.randint(0, cfg.vocab_size,
                           (args.batch, args.prompt_len), device=dev)
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    if args.graph:
        if not use_gpu or moe:
            print("--graph needs a GPU and a dense model", file=sys.stderr)
            return 1
        out = generate_graphed(model, tokens, args.new_tokens)
    elif moe:
        out = generate_moe(model, tokens, args.new_tokens,
                           temperature=args.temperature, top_k=args.top_k)
    else:
        out = generate(model, tokens, args.new_tokens,
                       temperature=args.temperature, top_k=args.top_k)
    if use_gpu:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    assert out.shape == (args.batch, args.prompt_len + args.new_tokens)
    print(json.dumps({
        "app": "generate", "model": args.model, "device": str(dev),
        "batch": args.batch, "prompt_len": args.prompt_len,
        "new_tokens": args.new_tokens, "graphed": bool(args.graph),
        "total_s": round(dt, 3),
        "tokens_per_second": round(args.batch * args.new_tokens / dt, 1),
    }))
    return 0


if __name__ == "__main__":
    sys.exit(main())
