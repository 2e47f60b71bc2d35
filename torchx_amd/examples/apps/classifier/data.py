"""Synthetic image data module for the classifier example.

Counterpart of the reference's TinyImageNetDataModule
(torchx/examples/apps/lightning/data.py) without the Lightning
dependency: generates a random labeled image set (there is no network on
the training boxes — same reason the reference ships
``create_random_data``), saves/loads it through fsspec so the app stays
storage-agnostic, and hands out rank-sharded loaders.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch.utils.data import DataLoader, Dataset, DistributedSampler


class SyntheticImages(Dataset):
    """Random [3, 64, 64] images with integer labels."""

    def __init__(self, n: int, num_classes: int, seed: int = 0) -> None:
        g = torch.Generator().manual_seed(seed)
        self.images = torch.randn(n, 3, 64, 64, generator=g)
        self.labels = torch.randint(0, num_classes, (n,), generator=g)
        # make the task learnable: shift each image's mean by its label
        self.images += self.labels.view(-1, 1, 1, 1).float() * 0.1

    def __len__(self) -> int:
        return len(self.labels)

    def __getitem__(self, i: int) -> Tuple[torch.Tensor, torch.Tensor]:
        return self.images[i], self.labels[i]


def create_random_data(output_path: str, n: int = 256,
                       num_classes: int = 8, seed: int = 0) -> str:
    """Materialize a synthetic dataset to ``output_path`` (any fsspec URI)."""
    import fsspec

    ds = SyntheticImages(n, num_classes, seed)
    fs, path = fsspec.core.url_to_fs(output_path)
    fs.makedirs(path, exist_ok=True)
    with fs.open(f"{path}/data.pt", "wb") as f:
        torch.save({"images": ds.images, "labels": ds.labels,
                    "num_classes": num_classes}, f)
    return f"{path}/data.pt"


def load_data(path: str) -> Tuple[SyntheticImages, int]:
    import fsspec

    fs, p = fsspec.core.url_to_fs(path)
    with fs.open(p, "rb") as f:
        blob = torch.load(f, map_location="cpu", weights_only=True)
    ds = SyntheticImages.__new__(SyntheticImages)
    ds.images = blob["images"]
    ds.labels = blob["labels"]
    return ds, int(blob["num_classes"])


class ImageDataModule:
    """Train/val split with rank-sharded loaders (DistributedSampler when
    a process group is up)."""

    def __init__(self, dataset: SyntheticImages, batch_size: int = 32,
                 val_fraction: float = 0.25) -> None:
        n_val = max(1, int(len(dataset) * val_fraction))
        n_train = len(dataset) - n_val
        self.train_set, self.val_set = torch.utils.data.random_split(
            dataset, [n_train, n_val],
            generator=torch.Generator().manual_seed(1),
        )
        self.batch_size = batch_size

    def _loader(self, ds: Dataset, shuffle: bool) -> DataLoader:
        import torch.distributed as dist

        sampler: Optional[DistributedSampler] = None
        if dist.is_available() and dist.is_initialized() \
                and dist.get_world_size() > 1:
            sampler = DistributedSampler(ds, shuffle=shuffle)
            shuffle = False
        return DataLoader(ds, batch_size=self.batch_size, shuffle=shuffle,
                          sampler=sampler)

    def train_loader(self) -> DataLoader:
        return self._loader(self.train_set, shuffle=True)

    def val_loader(self) -> DataLoader:
        return self._loader(self.val_set, shuffle=False)
