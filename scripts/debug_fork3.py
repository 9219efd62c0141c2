#!/usr/bin/env python3
"""Fork-fusion bug hunt, stage 3: Trainer path, per-param, determinism."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from pytorch_distributed_train_amd.flags import build_parser
from pytorch_distributed_train_amd.engine import Trainer


def rel(a, b):
    return float((a - b).norm() / b.norm().clamp(min=1e-9))


def run(fuse, steps=1):
    if fuse:
        os.environ.pop("PDTA_NO_FORK_FUSE", None)
    else:
        os.environ["PDTA_NO_FORK_FUSE"] = "1"
    args = build_parser().parse_args([
        "--model", "resnet50", "--data", "synthetic-imagenet",
        "--dtype", "bf16", "--batch-size", "8", "--seed", "41"])
    t = Trainer(args)
    t.model.train()
    g = torch.Generator().manual_seed(3)
    x = torch.randn(8, 3, 224, 224, generator=g).cuda()
    y = torch.randint(0, 1000, (8,), generator=g).cuda()
    for _ in range(steps):
        loss = t.train_step(x, y)
    torch.cuda.synchronize()
    per = {}
    for i, name in enumerate(t.flat.names):
        off, n = t.flat.offsets[i], t.flat.sizes[i]
        per[name] = t.flat.grad[off:off + n].cpu().clone()
    return float(loss), per


def main():
    l1, p1 = run(True)
    l1b, p1b = run(True)
    l0, p0 = run(False)
    det = [n for n in p1 if rel(p1[n], p1b[n]) > 1e-6]
    print(f"determinism: loss {l1} vs {l1b}; nondet params: {len(det)}"
          f" {det[:3]}")
    bads = sorted(((rel(p1[n], p0[n]), n) for n in p1), reverse=True)
    nbad = sum(1 for r, _ in bads if r > 0.02)
    print(f"fused vs unfused: loss {l1} vs {l0}; bad {nbad}/{len(bads)}")
    for r, n in bads[:25]:
        print(f"  {n:45s} rel={r:.4f}")
    print("DONE")


if __name__ == "__main__":
    main()
