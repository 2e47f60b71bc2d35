"""Reference training app: Llama step loop with checkpoint/resume and
experiment tracking.

Parity: the reference ships a Lightning trainer example
(torchx/examples/apps/lightning/train.py) whose conventions this app keeps —
checkpoints are app-owned files under fsspec paths, resumable from latest,
surfaced as tracker artifacts (AppRun.add_artifact, torchx/tracker/api.py:246).
MI355X-native: the model runs through the CDNA4 HIP kernel path
(FlatParams/FlatDDP/FlatAdamW) with RCCL for multi-process.

Run standalone or via the launcher:
  torchx run dist.ddp -j 1x2 -m torchx_amd.apps.trainer -- --steps 10
"""

from __future__ import annotations

import argparse
import os
import sys
from typing import Optional

import torch

from torchx_amd.distributed import init_pg, rank, world_size


def parse_args(argv) -> argparse.Namespace:
    p = argparse.ArgumentParser(description="torchx_amd reference trainer")
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--micro-batch", type=int, default=2)
    p.add_argument("--seq-len", type=int, default=128)
    p.add_argument("--lr", type=float, default=1e-3)
    p.add_argument("--model", type=str, default="tiny",
                   choices=["tiny", "gpu_tiny", "llama3_8b"])
    p.add_argument("--checkpoint-dir", type=str, default=None,
                   help="fsspec path for periodic checkpoints")
    p.add_argument("--checkpoint-every", type=int, default=5)
    p.add_argument("--resume", action="store_true",
                   help="resume from the latest checkpoint in checkpoint-dir")
    return p.parse_args(argv)


def _latest_checkpoint(ckpt_dir: str) -> Optional[str]:
    import fsspec

    fs, path = fsspec.core.url_to_fs(ckpt_dir)
    if not fs.exists(path):
        return None
    steps = []
    for f in fs.ls(path):
        base = os.path.basename(f.rstrip("/"))
        if base.startswith("step_") and base.endswith(".pt"):
            steps.append((int(base[5:-3]), f))
    if not steps:
        return None
    return max(steps)[1]


def save_checkpoint(ckpt_dir: str, step: int, model, opt) -> str:
    import fsspec

    fs, path = fsspec.core.url_to_fs(ckpt_dir)
    fs.makedirs(path, exist_ok=True)
    target = f"{path}/step_{step}.pt"
    with fs.open(target, "wb") as f:
        torch.save(
            {"step": step,
             "model": model.state_dict(),
             "opt": opt.state_dict()},
            f,
        )
    return target


def load_checkpoint(path: str, model, opt) -> int:
    import fsspec

    fs, p = fsspec.core.url_to_fs(path)
    with fs.open(p, "rb") as f:
        # weights_only: the payload is plain tensors/ints; never unpickle
        # arbitrary objects from an fsspec URI (shared/writable dirs)
        sd = torch.load(f, map_location="cpu", weights_only=True)
    model.load_state_dict(sd["model"])
    opt.load_state_dict(sd["opt"])
    return int(sd["step"])


def main(argv=None) -> int:
    args = parse_args(argv if argv is not None else sys.argv[1:])
    device = init_pg()

    from torchx_amd.models.llama import (
        LlamaModel, llama3_8b, llama_gpu_tiny, llama_tiny,
    )
    from torchx_amd.parallel import FlatAdamW, FlatDDP, FlatParams

    cfg = {"tiny": llama_tiny, "gpu_tiny": llama_gpu_tiny,
           "llama3_8b": llama3_8b}[args.model]()
    args.seq_len = min(args.seq_len, cfg.max_seq_len)

    torch.manual_seed(1234 + rank())
    model = LlamaModel(cfg, device=device)
    flat = FlatParams(model, device)
    ddp = FlatDDP(flat)
    opt = FlatAdamW(flat, lr=args.lr)

    start_step = 0
    if args.resume and args.checkpoint_dir:
        latest = _latest_checkpoint(args.checkpoint_dir)
        if latest:
            # load_state_dict copies into the flat-buffer views in place
            start_step = load_checkpoint(latest, model, opt)
            print(f"rank {rank()}: resumed from {latest} (step {start_step})",
                  flush=True)

    # experiment tracking via the env-var contract (no-op when unconfigured)
    from torchx_amd.tracker.api import app_run_from_env

    app_run = app_run_from_env()
    app_run.add_metadata(model=args.model, lr=args.lr, steps=args.steps)

    B, S = args.micro_batch, args.seq_len
    tokens = torch.randint(0, cfg.vocab_size, (B, S), device=device)
    targets = torch.roll(tokens, shifts=-1, dims=1)

    loss_val = 0.0
    for step in range(start_step, args.steps):
        opt.zero_grad()
        loss = model(tokens, targets)
        loss.backward()
        ddp.finish()
        opt.step()
        loss_val = float(loss.detach())
        if rank() == 0:
            print(f"step {step + 1} loss {loss_val:.4f}", flush=True)
        if (args.checkpoint_dir and rank() == 0
                and (step + 1) % args.checkpoint_every == 0):
            path = save_checkpoint(args.checkpoint_dir, step + 1, model, opt)
            app_run.add_artifact("checkpoint", path)

    if args.checkpoint_dir and rank() == 0:
        path = save_checkpoint(args.checkpoint_dir, args.steps, model, opt)
        app_run.add_artifact("checkpoint", path)
    if world_size() > 1:
        import torch.distributed as dist

        dist.barrier()
    return 0


if __name__ == "__main__":
    sys.exit(main())
