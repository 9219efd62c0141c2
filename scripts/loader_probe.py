#!/usr/bin/env python3
"""Time the loader variants and the bare train loop stages."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from pytorch_distributed_train_amd.data import (DistributedSampler, Loader,
                                                make_dataset)


def time_loader(workers, mode, pin, n=20):
    ds = make_dataset("synthetic-imagenet", 256 * (n + 2), seed=1)
    sa = DistributedSampler(ds, 1, 0, shuffle=True, seed=1)
    ld = Loader(ds, sa, 256, pin_memory=pin, workers=workers,
                worker_mode=mode)
    it = iter(ld)
    next(it)
    t0 = time.perf_counter()
    k = 0
    for x, y in it:
        k += 1
        if k >= n:
            break
    dt = (time.perf_counter() - t0) / n
    print(f"workers={workers} mode={mode} pin={pin}: {dt*1000:8.1f} ms/batch"
          f"  ({256/dt:.0f} img/s)")


def main():
    for workers, mode, pin in [(0, "thread", False), (0, "thread", True),
                               (4, "thread", True), (8, "thread", True),
                               (4, "process", True)]:
        time_loader(workers, mode, pin)


if __name__ == "__main__":
    main()
