"""Model interpretability app for the classifier example.

Counterpart of the reference's captum example
(torchx/examples/apps/lightning/interpret.py — Integrated Gradients over
a trained checkpoint, attributions written as images). captum is not in
the target image, so the attribution method here is pure torch:
gradient * input saliency plus a small occlusion sweep — the same
"load checkpoint -> attribute a batch -> write artifacts" app shape, run
as a launchable job whose outputs land on an fsspec path.

Run standalone or via the launcher::

    python -m torchx_amd.examples.apps.classifier.interpret \
        --load-path /ckpts/epoch_2.pt --data-path /data --output-path /out
"""

from __future__ import annotations

import argparse
import json
import logging
import os
import sys
from typing import List, Optional

import fsspec
import torch

from .data import create_random_data, load_data
from .model import TinyImageModel

log = logging.getLogger(__name__)


def saliency(model: TinyImageModel, x: torch.Tensor) -> torch.Tensor:
    """Gradient-times-input attribution for the predicted class,
    [B, 3, H, W] -> per-pixel importance [B, H, W] (channel-summed abs)."""
    x = x.clone().requires_grad_(True)
    logits = model(x)
    score = logits.gather(1, logits.argmax(-1, keepdim=True)).sum()
    score.backward()
    assert x.grad is not None
    return (x.grad * x).abs().sum(1)


@torch.no_grad()
def occlusion(model: TinyImageModel, x: torch.Tensor,
              patch: int = 8) -> torch.Tensor:
    """Occlusion map: drop in the predicted-class probability when each
    ``patch``x``patch`` square is zeroed. [B, 3, H, W] -> [B, Hp, Wp]."""
    probs = torch.softmax(model(x), dim=-1)
    cls = probs.argmax(-1)
    base = probs.gather(1, cls[:, None]).squeeze(1)
    H, W = x.shape[-2:]
    out = torch.zeros(x.shape[0], H // patch, W // patch)
    for i in range(0, H - patch + 1, patch):
        for j in range(0, W - patch + 1, patch):
            occluded = x.clone()
            occluded[:, :, i:i + patch, j:j + patch] = 0
            p = torch.softmax(model(occluded), dim=-1)
            out[:, i // patch, j // patch] = (
                base - p.gather(1, cls[:, None]).squeeze(1))
    return out


def parse_args(argv: List[str]) -> argparse.Namespace:
    p = argparse.ArgumentParser(description="classifier interpretability")
    p.add_argument("--load-path", default="",
                   help="checkpoint from train.py (empty: random init)")
    p.add_argument("--data-path", default="",
                   help="dataset dir (empty: synthesize in-memory)")
    p.add_argument("--output-path", required=True,
                   help="fsspec dir for attribution artifacts")
    p.add_argument("--batch", type=int, default=8)
    p.add_argument("--patch", type=int, default=8)
    return p.parse_args(argv)


def main(argv: Optional[List[str]] = None) -> int:
    logging.basicConfig(level=logging.INFO)
    args = parse_args(argv or sys.argv[1:])
    if args.data_path:
        ds, num_classes = load_data(args.data_path)
    else:
        synth = create_random_data(
            os.path.join(args.output_path, "_synth"), n=args.batch * 2)
        ds, num_classes = load_data(synth)
    model = TinyImageModel(num_classes=num_classes)
    if args.load_path:
        with fsspec.open(args.load_path, "rb") as f:
            model.load_state_dict(
                torch.load(f, map_location="cpu",
                           weights_only=True)["model"])
    model.eval()

    x = torch.stack([ds[i][0] for i in range(min(args.batch, len(ds)))])
    sal = saliency(model, x)
    occ = occlusion(model, x, patch=args.patch)

    fs, root = fsspec.core.url_to_fs(args.output_path)
    fs.makedirs(root, exist_ok=True)
    with fsspec.open(os.path.join(args.output_path, "attributions.pt"),
                     "wb") as f:
        torch.save({"saliency": sal, "occlusion": occ}, f)
    summary = {
        "n": int(x.shape[0]),
        "saliency_mean": float(sal.mean()),
        "occlusion_max_drop": float(occ.max()),
    }
    with fsspec.open(os.path.join(args.output_path, "summary.json"),
                     "w") as f:
        json.dump(summary, f)
    log.info("attributions written: %s", summary)
    return 0


if __name__ == "__main__":
    sys.exit(main())
