#!/usr/bin/env python3
"""Shape fuzz: random conv/BN/LN shapes through the HIP kernels vs the
CPU fp32 references (the fixed-shape numerics tests' random-shape
companion). Seeded; prints per-case rel errors and FAILs loudly."""
import os
import random
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from pytorch_distributed_train_amd.ops import functional as OF


def rel(a, b):
    return float((a.float() - b.float()).norm() /
                 b.float().norm().clamp(min=1e-9))


def fuzz_conv(rng, i):
    C = rng.choice([64, 128, 192, 256, 320, 512])
    K = rng.choice([64, 128, 256, 512])
    ks, st = rng.choice([(1, 1), (3, 1), (3, 2), (1, 2)])
    pad = 1 if ks == 3 else 0
    H = rng.choice([7, 8, 14, 15, 28])
    N = rng.choice([2, 3, 4])
    x = torch.randn(N, H, H, C) * 0.5
    w = torch.randn(K, C, ks, ks) * 0.1
    wt = w.permute(0, 2, 3, 1).contiguous()
    w2 = w.permute(2, 3, 1, 0).contiguous()
    xg = x.cuda().bfloat16().requires_grad_(True)
    wg = w.cuda().requires_grad_(True)
    y = OF.conv2d_nhwc(xg, wg, wt.cuda().bfloat16(), w2.cuda().bfloat16(),
                       st, pad)
    dy = torch.randn_like(y)
    y.backward(dy)
    xc = x.requires_grad_(True)
    wc = w.clone().requires_grad_(True)
    yc = OF.conv2d_nhwc(xc, wc, wt, w2, st, pad)
    yc.backward(dy.cpu().float())
    ry = rel(y.detach().cpu(), yc.detach())
    rx = rel(xg.grad.cpu(), xc.grad)
    rw = rel(wg.grad.cpu(), wc.grad)
    ok = ry < 0.02 and rx < 0.03 and rw < 0.03
    print(f"conv[{i}] N{N} H{H} C{C} K{K} k{ks}s{st}: y={ry:.4f} "
          f"dx={rx:.4f} dw={rw:.4f}{'' if ok else '  FAIL'}")
    return ok


def fuzz_bn(rng, i):
    C = rng.choice([64, 128, 256, 512])
    N, H = rng.choice([(2, 14), (3, 8), (4, 7)])
    x = torch.randn(N, H, H, C)
    res = torch.randn(N, H, H, C) if rng.random() < 0.5 else None
    from pytorch_distributed_train_amd.models.layers import BatchNorm2dNHWC
    torch.manual_seed(i)
    bn_g = BatchNorm2dNHWC(C).cuda()
    bn_g.fuse_relu = res is not None
    torch.manual_seed(i)
    bn_c = BatchNorm2dNHWC(C)
    bn_c.fuse_relu = res is not None
    bn_g.train()
    bn_c.train()
    xg = x.cuda().bfloat16().requires_grad_(True)
    rg = res.cuda().bfloat16() if res is not None else None
    y = bn_g(xg, rg)
    dy = torch.randn_like(y)
    y.backward(dy)
    xc = x.requires_grad_(True)
    yc = bn_c(xc, res)
    yc.backward(dy.cpu().float())
    ry = rel(y.detach().cpu(), yc.detach())
    rx = rel(xg.grad.cpu(), xc.grad)
    ok = ry < 0.02 and rx < 0.05
    print(f"bn[{i}] N{N} H{H} C{C} res={res is not None}: y={ry:.4f} "
          f"dx={rx:.4f}{'' if ok else '  FAIL'}")
    return ok


def fuzz_ln(rng, i):
    d = rng.choice([256, 512, 768, 1024])
    rows = rng.choice([64, 197, 256, 1000])
    a = torch.randn(rows, d)
    b = torch.randn(rows, d)
    w = torch.randn(d).abs() + 0.5
    bias = torch.randn(d)
    ag = a.cuda().bfloat16().requires_grad_(True)
    bg = b.cuda().bfloat16().requires_grad_(True)
    y, s = OF.add_layer_norm(ag, bg, w.cuda(), bias.cuda(), 1e-6)
    dy = torch.randn_like(y)
    (y * dy).sum().backward()
    ac = a.requires_grad_(True)
    bc = b.requires_grad_(True)
    yc, sc = OF.add_layer_norm(ac, bc, w, bias, 1e-6)
    (yc * dy.cpu().float()).sum().backward()
    ry = rel(y.detach().cpu(), yc.detach())
    ra = rel(ag.grad.cpu(), ac.grad)
    ok = ry < 0.02 and ra < 0.05
    print(f"addln[{i}] rows{rows} d{d}: y={ry:.4f} da={ra:.4f}"
          f"{'' if ok else '  FAIL'}")
    return ok


def main():
    rng = random.Random(int(os.environ.get("PDTA_FUZZ_SEED", "7")))
    n = int(os.environ.get("PDTA_FUZZ_N", "8"))
    torch.manual_seed(1)
    fails = 0
    for i in range(n):
        fails += not fuzz_conv(rng, i)
    for i in range(n):
        fails += not fuzz_bn(rng, i)
    for i in range(n):
        fails += not fuzz_ln(rng, i)
    print("FUZZ", "FAILED" if fails else "OK", f"({fails} failures)")
    sys.exit(1 if fails else 0)


if __name__ == "__main__":
    main()
