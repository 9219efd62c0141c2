#!/usr/bin/env python3
"""Isolate the residual-fork gradient fusion, block by block."""
import os
import sys

import torch
import torch.nn as nn

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from pytorch_distributed_train_amd.models.resnet import Bottleneck, conv1x1
from pytorch_distributed_train_amd.models.layers import BatchNorm2dNHWC


def build(kind, seed=7):
    torch.manual_seed(seed)
    if kind == "ds":
        d = nn.Sequential(conv1x1(256, 512, 2), BatchNorm2dNHWC(512))
        d[0].emit_bn_stats = True
        return Bottleneck(256, 128, stride=2, downsample=d)
    if kind == "chain":
        return nn.Sequential(Bottleneck(256, 64), Bottleneck(256, 64))
    return Bottleneck(256, 64)


def grads(kind, fuse):
    if fuse:
        os.environ.pop("PDTA_NO_FORK_FUSE", None)
    else:
        os.environ["PDTA_NO_FORK_FUSE"] = "1"
    blk = build(kind).cuda()
    blk.train()
    g = torch.Generator().manual_seed(1)
    x = torch.randn(4, 14, 14, 256, generator=g).cuda().to(torch.bfloat16)
    x.requires_grad_(True)
    y = blk(x)
    y.float().pow(2).mean().backward()
    torch.cuda.synchronize()
    pg = {n: p.grad.float().cpu() for n, p in blk.named_parameters()}
    return x.grad.float().cpu(), pg


def rel(a, b):
    return float((a - b).norm() / b.norm().clamp(min=1e-9))


def main():
    for kind in ("plain", "ds", "chain"):
        dx_f, pg_f = grads(kind, True)
        dx_p, pg_p = grads(kind, False)
        print(f"[{kind}] dx rel={rel(dx_f, dx_p):.5f}")
        for n in pg_f:
            r = rel(pg_f[n], pg_p[n])
            flag = "  <-- BAD" if r > 0.02 else ""
            print(f"  {n:40s} rel={r:.5f}{flag}")
    print("DONE")


if __name__ == "__main__":
    main()
