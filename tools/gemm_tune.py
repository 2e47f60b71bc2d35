#!/usr/bin/env python3
"""Tune hipBLASLt algorithm selection (PyTorch TunableOp) for the bench
model's GEMM shapes and write the result CSV.

Run on a GPU box:
  python tools/gemm_tune.py --model llama3_8b --out tunableop_gfx950.csv
Commit the CSV; bench.py and the models load it read-only when present
(torchx_amd/ops/tunableop_gfx950.csv), giving tuned GEMM algorithms
without per-run tuning cost.
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama3_8b")
    p.add_argument("--out", default="gpurun_out/tunableop_gfx950.csv")
    p.add_argument("--steps", type=int, default=2)
    p.add_argument("--micro-batch", type=int, default=4)
    p.add_argument("--seq-len", type=int, default=4096)
    args = p.parse_args()

    assert torch.cuda.is_available()
    torch.cuda.tunable.enable(True)
    torch.cuda.tunable.tuning_enable(True)
    # cap per-candidate cost: the defaults (30 ms / 100 iters per
    # solution x ~600 hipblaslt solutions) take ~2.5 min per GEMM shape
    torch.cuda.tunable.set_max_tuning_duration(5)
    torch.cuda.tunable.set_max_tuning_iterations(10)
    # filename must be set BEFORE ops run; write_file at the end
    torch.cuda.tunable.set_filename(args.out)

    from torchx_amd.models.llama import LlamaModel, llama3_8b
    from torchx_amd.models.mixtral import MixtralModel, mixtral_8x7b
    from torchx_amd.parallel import FlatAdamW, FlatDDP, FlatParams

    dev = torch.device("cuda", 0)
    torch.manual_seed(0)
    if args.model == "llama3_8b":
        cfg = llama3_8b()
        model = LlamaModel(cfg, device=dev)
    else:
        raise SystemExit(f"unknown model {args.model}")

    flat = FlatParams(model, dev)
    ddp = FlatDDP(flat)
    opt = FlatAdamW(flat, lr=1e-4)
    B, S = args.micro_batch, args.seq_len
    tokens = torch.randint(0, cfg.vocab_size, (B, S), device=dev)
    targets = torch.roll(tokens, shifts=-1, dims=1)
    for i in range(args.steps):
        opt.zero_grad()
        loss = model(tokens, targets)
        loss.backward()
        ddp.finish()
        opt.step()
        torch.cuda.synchronize()
        print(f"step {i} done", flush=True)
    torch.cuda.tunable.write_file(args.out)
    print("wrote", args.out)


if __name__ == "__main__":
    main()
