"""Small CNN classifier (counterpart of the reference's
TinyImageNetModel, torchx/examples/apps/lightning/model.py — a plain
nn.Module instead of a LightningModule)."""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


class TinyImageModel(nn.Module):
    def __init__(self, num_classes: int = 8, width: int = 16) -> None:
        super().__init__()
        self.features = nn.Sequential(
            nn.Conv2d(3, width, 3, stride=2, padding=1),
            nn.BatchNorm2d(width),
            nn.ReLU(inplace=True),
            nn.Conv2d(width, width * 2, 3, stride=2, padding=1),
            nn.BatchNorm2d(width * 2),
            nn.ReLU(inplace=True),
            nn.AdaptiveAvgPool2d(1),
        )
        self.head = nn.Linear(width * 2, num_classes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.head(self.features(x).flatten(1))

    def loss(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        return F.cross_entropy(self(x), y)

    @torch.no_grad()
    def accuracy(self, x: torch.Tensor, y: torch.Tensor) -> float:
        return float((self(x).argmax(-1) == y).float().mean())


def export_inference(model: TinyImageModel, out_path: str) -> str:
    """TorchScript-export the model for serving (fsspec URI)."""
    import fsspec

    model.eval()
    scripted = torch.jit.script(model)
    fs, path = fsspec.core.url_to_fs(out_path)
    if "/" in path:
        fs.makedirs(path.rsplit("/", 1)[0], exist_ok=True)
    with fs.open(path, "wb") as f:
        torch.jit.save(scripted, f)
    return out_path
