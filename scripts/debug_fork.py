#!/usr/bin/env python3
"""Isolate the residual-fork gradient fusion: one Bottleneck block (with
and without downsample), grads fused vs unfused vs CPU fp32."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from pytorch_distributed_train_amd.models.resnet import Bottleneck, conv1x1
from pytorch_distributed_train_amd.models.layers import BatchNorm2dNHWC
import torch.nn as nn


def run(block, x, train=True):
    block.train(train)
    xc = x.clone().requires_grad_(True)
    y = block(xc)
    y.float().pow(2).mean().backward()
    return (y.detach().float().cpu(), xc.grad.float().cpu(),
            {n: p.grad.float().cpu() for n, p in block.named_parameters()})


def rel(a, b):
    return float((a - b).norm() / b.norm().clamp(min=1e-9))


def build(ds, seed=7):
    torch.manual_seed(seed)
    if ds:
        d = nn.Sequential(conv1x1(256, 512, 2), BatchNorm2dNHWC(512))
        d[0].emit_bn_stats = True
        return Bottleneck(256, 128, stride=2, downsample=d)
    return Bottleneck(256, 64)


def main():
    g = torch.Generator().manual_seed(1)
    x = torch.randn(4, 14, 14, 256, generator=g)
    for ds in (False, True):
        cpu = build(ds)
        y0, dx0, pg0 = run(cpu, x)

        for mode, env in (("fused", None), ("nofuse", "1")):
            if env:
                os.environ["PDTA_NO_FORK_FUSE"] = env
            else:
                os.environ.pop("PDTA_NO_FORK_FUSE", None)
            blk = build(ds).to("cuda").to(torch.float32)
            # bf16 compute dtype path
            blk_b = build(ds)
            blk_b = blk_b.to("cuda")
            for p in blk_b.parameters():
                pass
            xb = x.to("cuda", torch.bfloat16)
            y, dx, pg = run(blk_b, xb)
            print(f"ds={ds} {mode}: y rel={rel(y, y0):.4f} "
                  f"dx rel={rel(dx, dx0):.4f} "
                  f"conv1.dw rel={rel(pg['conv1.weight'], pg0['conv1.weight']):.4f} "
                  f"conv3.dw rel={rel(pg['conv3.weight'], pg0['conv3.weight']):.4f}")
    print("OK")


if __name__ == "__main__":
    main()
