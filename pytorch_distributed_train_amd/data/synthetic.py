"""Synthetic datasets (no network: random data, seeded, fixed shapes).

BASELINE.json configs name ImageNet-shape (3x224x224, 1000 classes) and
CIFAR-10-shape (3x32x32, 10 classes) synthetic data. Samples are generated
deterministically per index so every rank/epoch sees identical data for a
given index (makes DDP-vs-single-process gradient equality testable).
"""
from __future__ import annotations

import torch


class SyntheticImageDataset:
    """Deterministic per-index samples. Images come from a lazily-built
    POOL of `pool_size` unique tensors (index -> pool[idx % pool_size]):
    generating a fresh 3x224x224 randn per __getitem__ cost ~1 ms and
    made the synchronous loader the training bottleneck (~880 img/s on a
    GPU that trains at 8000+); the pool turns collation into memcpys.
    Labels stay per-index seeded, and index -> data is still
    deterministic, so DDP-vs-single-process equality tests hold."""

    def __init__(self, num_samples: int, image_shape, num_classes: int,
                 seed: int = 0, pool_size: int = 256):
        self.num_samples = num_samples
        self.image_shape = tuple(image_shape)
        self.num_classes = num_classes
        self.seed = seed
        self.pool_size = min(pool_size, num_samples) or 1
        self._pool = None

    def __len__(self) -> int:
        return self.num_samples

    def _ensure_pool(self):
        if self._pool is None:
            g = torch.Generator()
            g.manual_seed(self.seed * 7_777_777 + 13)
            self._pool = torch.randn((self.pool_size,) + self.image_shape,
                                     generator=g)
            # all labels in one vectorized draw: a per-__getitem__
            # Generator().manual_seed() costs ~1.5 ms EACH (it re-seeds
            # the whole MT19937 state) and was the real loader bottleneck
            self._labels = torch.randint(0, self.num_classes,
                                         (self.num_samples,), generator=g)
        return self._pool

    def __getitem__(self, idx: int):
        pool = self._ensure_pool()
        return pool[idx % self.pool_size], int(self._labels[idx])


_PRESETS = {
    "synthetic-imagenet": dict(image_shape=(3, 224, 224), num_classes=1000,
                               default_samples=50_000),
    "synthetic-cifar10": dict(image_shape=(3, 32, 32), num_classes=10,
                              default_samples=10_000),
}


def make_dataset(name: str, num_samples: int = 0, seed: int = 0):
    spec = _PRESETS[name]
    n = num_samples if num_samples > 0 else spec["default_samples"]
    return SyntheticImageDataset(n, spec["image_shape"], spec["num_classes"],
                                 seed=seed)
