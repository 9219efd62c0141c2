"""Distributed sampler, re-implemented from the behavioral contract.

Semantics match torch's DistributedSampler (verified
against $TORCH/utils/data/distributed.py:94-146 behavior, re-implemented):
  - pad the index list so len is divisible by world size (wrap-around pad),
  - epoch-seeded shuffle so every epoch reshuffles identically on all ranks,
  - rank-strided subsample indices[rank::num_replicas],
  - set_epoch(epoch) must be called each epoch for reshuffling.
"""
from __future__ import annotations

import torch


class DistributedSampler:
    def __init__(self, dataset, num_replicas: int, rank: int,
                 shuffle: bool = True, seed: int = 0, drop_last: bool = False):
        if rank < 0 or rank >= num_replicas:
            raise ValueError(f"rank {rank} out of range for {num_replicas} replicas")
        self.dataset = dataset
        self.num_replicas = num_replicas
        self.rank = rank
        self.shuffle = shuffle
        self.seed = seed
        self.drop_last = drop_last
        self.epoch = 0
        n = len(dataset)
        if drop_last and n % num_replicas != 0:
            self.num_samples = n // num_replicas
        else:
            self.num_samples = (n + num_replicas - 1) // num_replicas
        self.total_size = self.num_samples * num_replicas

    def set_epoch(self, epoch: int) -> None:
        self.epoch = epoch

    def __len__(self) -> int:
        return self.num_samples

    def __iter__(self):
        n = len(self.dataset)
        if self.shuffle:
            g = torch.Generator()
            g.manual_seed(self.seed + self.epoch)
            indices = torch.randperm(n, generator=g).tolist()
        else:
            indices = list(range(n))
        if not self.drop_last:
            pad = self.total_size - len(indices)
            if pad > 0:
                # wrap-around padding, as the torch sampler does
                reps = (pad + len(indices) - 1) // len(indices)
                indices += (indices * reps)[:pad]
        else:
            indices = indices[: self.total_size]
        assert len(indices) == self.total_size
        sub = indices[self.rank :: self.num_replicas]
        assert len(sub) == self.num_samples
        return iter(sub)
