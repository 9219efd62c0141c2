#!/usr/bin/env python3
"""Microbench: gemm_nt on ViT shapes, current vs experimental schedules.

    PDTA_GEMM_8P=1 python scripts/gemm_probe.py
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from pytorch_distributed_train_amd.ops.backend import ext

SHAPES = [  # (M, N, K) — ViT-B/16 B=128 padded-seq GEMMs + a square
    (32768, 2304, 768),   # qkv
    (32768, 768, 768),    # proj
    (32768, 3072, 768),   # mlp up
    (32768, 768, 3072),   # mlp down
    (4096, 4096, 4096),   # reference square
    (8192, 8192, 8192),
]


def bench(M, N, K, iters=30):
    e = ext()
    g = torch.Generator(device="cpu").manual_seed(0)
    A = (torch.rand(M, K, generator=g) * 2 - 1).to("cuda", torch.bfloat16)
    B = (torch.rand(N, K, generator=g) * 2 - 1).to("cuda", torch.bfloat16)
    bias = (torch.rand(N, generator=g).to("cuda", torch.float32)
            if os.environ.get("PDTA_PROBE_BIAS") else None)
    hold = bias
    if os.environ.get("PDTA_PROBE_BIAS") == "alloc-only":
        bias = None  # `hold` keeps the tensor allocated: same address
        # layout, no bias work — separates epilogue cost from addresses
    C = e.gemm_nt(A, B, bias)
    # correctness vs torch matmul in fp32 (sampled rows to bound cost)
    idx = torch.randint(0, M, (256,))
    ref = A[idx].float() @ B.float().t()
    if bias is not None:
        ref = ref + bias
    got = C[idx].float()
    rel = float((got - ref).norm() / ref.norm().clamp(min=1e-9))
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        e.gemm_nt(A, B, bias)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    tf = 2.0 * M * N * K / dt / 1e12
    print(f"M={M:6d} N={N:5d} K={K:5d}  {dt*1e6:9.1f} us  {tf:8.1f} TF"
          f"  rel={rel:.2e}{'  <-- BAD' if rel > 0.02 else ''}")
    return tf


def main():
    mode = ("BD" if os.environ.get("PDTA_GEMM_BD") else
            "8P2" if os.environ.get("PDTA_GEMM_8P2") else
            "R4" if os.environ.get("PDTA_GEMM_R4") else
            "8P" if os.environ.get("PDTA_GEMM_8P") else
            "V2" if os.environ.get("PDTA_GEMM_V2") else "base")
    print(f"schedule: {mode}")
    for shape in SHAPES:
        bench(*shape)




def pad_sweep():
    """Same shape, varying allocator offsets: if timing swings with the
    PAD (which only shifts base addresses), the variance is HBM channel
    camping, not kernel schedule."""
    e = ext()
    M, N, K = 32768, 768, 3072
    for pad in [0, 4096, 8192, 12288, 65536, 131072, 262144, 524288]:
        hold = torch.empty(pad, dtype=torch.bfloat16, device="cuda") \
            if pad else None
        g = torch.Generator(device="cpu").manual_seed(0)
        A = (torch.rand(M, K, generator=g) * 2 - 1).to("cuda",
                                                       torch.bfloat16)
        B = (torch.rand(N, K, generator=g) * 2 - 1).to("cuda",
                                                       torch.bfloat16)
        C = e.gemm_nt(A, B, None)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(30):
            e.gemm_nt(A, B, None)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 30
        print(f"pad={pad:7d}  A=0x{A.data_ptr():x} B=0x{B.data_ptr():x} "
              f"C=0x{C.data_ptr():x}  {dt*1e6:8.1f} us")
        del hold, A, B, C
        torch.cuda.empty_cache()


if __name__ == "__main__":
    if os.environ.get("PDTA_PROBE_PADSWEEP"):
        pad_sweep()
    else:
        main()
