#!/usr/bin/env python3
"""Fork-fusion bug hunt, stage 2: emit_bn_stats interplay + full model."""
import os
import sys

import torch
import torch.nn as nn

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from pytorch_distributed_train_amd.models.resnet import Bottleneck, conv1x1
from pytorch_distributed_train_amd.models.layers import (BatchNorm2dNHWC,
                                                         Conv2dNHWC)
from pytorch_distributed_train_amd.models import build_model


def wire_emit(m):
    for mod in m.modules():
        for cn, bn in (("conv1", "bn1"), ("conv2", "bn2"), ("conv3", "bn3")):
            c = getattr(mod, cn, None)
            if isinstance(c, Conv2dNHWC) and \
                    isinstance(getattr(mod, bn, None), BatchNorm2dNHWC):
                c.emit_bn_stats = True
        if isinstance(mod, nn.Sequential) and len(mod) == 2 and \
                isinstance(mod[0], Conv2dNHWC):
            mod[0].emit_bn_stats = True


def rel(a, b):
    return float((a - b).norm() / b.norm().clamp(min=1e-9))


def grads_chain(fuse, emit):
    if fuse:
        os.environ.pop("PDTA_NO_FORK_FUSE", None)
    else:
        os.environ["PDTA_NO_FORK_FUSE"] = "1"
    torch.manual_seed(7)
    blk = nn.Sequential(Bottleneck(256, 64), Bottleneck(256, 64))
    if emit:
        wire_emit(blk)
    blk.cuda().train()
    g = torch.Generator().manual_seed(1)
    x = torch.randn(4, 14, 14, 256, generator=g).cuda().to(torch.bfloat16)
    x.requires_grad_(True)
    y = blk(x)
    y.float().pow(2).mean().backward()
    torch.cuda.synchronize()
    return x.grad.float().cpu(), \
        {n: p.grad.float().cpu() for n, p in blk.named_parameters()}


def grads_model(fuse):
    if fuse:
        os.environ.pop("PDTA_NO_FORK_FUSE", None)
    else:
        os.environ["PDTA_NO_FORK_FUSE"] = "1"
    torch.manual_seed(11)
    m = build_model("resnet50", 1000, torch.bfloat16).cuda()
    m.train()
    g = torch.Generator().manual_seed(2)
    x = torch.randn(4, 3, 224, 224, generator=g).cuda()
    y = m(x)
    y.float().pow(2).mean().backward()
    torch.cuda.synchronize()
    return {n: p.grad.float().cpu() for n, p in m.named_parameters()}


def main():
    for emit in (False, True):
        dxf, pf = grads_chain(True, emit)
        dxp, pp = grads_chain(False, emit)
        bad = [n for n in pf if rel(pf[n], pp[n]) > 0.02]
        print(f"chain emit={emit}: dx rel={rel(dxf, dxp):.5f} "
              f"bad={len(bad)}/{len(pf)} {bad[:4]}")
    pf = grads_model(True)
    pp = grads_model(False)
    bads = [(n, rel(pf[n], pp[n])) for n in pf]
    nbad = sum(1 for _, r in bads if r > 0.02)
    print(f"resnet50: bad={nbad}/{len(bads)}")
    for n, r in bads:
        if r > 0.02:
            print(f"  {n:45s} rel={r:.4f}")
    print("DONE")


if __name__ == "__main__":
    main()
