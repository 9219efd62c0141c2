#!/usr/bin/env python3
"""bench.py — driver contract benchmark (BASELINE.json).

Measures the flagship training step: ResNet-50, ImageNet-shape synthetic
data, bf16 compute, images/sec for the WHOLE job. Run directly for 1 GPU or
under torch.distributed.run for N ranks (one per GPU, RCCL over xGMI):

    python bench.py --gpus 1 --steps 30 --warmup 10
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 bench.py --gpus 8 --steps 30 --warmup 10

Prints ONE JSON line from rank 0 (whole-job images/sec, max step time over
ranks). Weak scaling: per-GPU batch is fixed as N grows.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

from pytorch_distributed_train_amd.flags import build_parser, DistEnv
from pytorch_distributed_train_amd.engine import Trainer


def parse(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--model", default="resnet50")
    p.add_argument("--batch-size", type=int, default=256,
                   help="per-GPU batch size")
    p.add_argument("--dtype", default="bf16")
    p.add_argument("--bucket-cap-mb", type=int, default=50)
    p.add_argument("--grad-comm-dtype", default="same")
    p.add_argument("--sync-bn", action="store_true")
    p.add_argument("--local_rank", "--local-rank", type=int, default=None)
    p.add_argument("--hip-graph", dest="hip_graph", action="store_true",
                   default=True,
                   help="capture the step into a HIP graph and replay it "
                        "(single-GPU; default on, falls back to eager)")
    p.add_argument("--no-hip-graph", dest="hip_graph", action="store_false")
    return p.parse_args(argv)


def _free_port() -> int:
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _self_launch(argv) -> int:
    """`python bench.py --gpus N` with no rendezvous env: spawn N local
    ranks of this script via the gang launcher (driver contract — the
    bare invocation must produce an n_gpus=N record)."""
    from pytorch_distributed_train_amd.launch import main as launch_main
    return launch_main([
        "--nproc_per_node", str(parse(argv).gpus),
        "--use-env",
        "--master_addr", "127.0.0.1",
        "--master_port", str(_free_port()),
        "--max-restarts", "0",
        os.path.abspath(__file__),
        *(argv if argv is not None else sys.argv[1:]),
    ])


def main(argv=None) -> int:
    b = parse(argv)
    if b.gpus > 1 and "WORLD_SIZE" not in os.environ:
        return _self_launch(argv)
    env = DistEnv.from_env(b.local_rank)
    if env.world_size > 1 and env.world_size != b.gpus and env.rank == 0:
        print(f"warning: WORLD_SIZE={env.world_size} != --gpus {b.gpus}; "
              "using WORLD_SIZE", file=sys.stderr)
    n_gpus = env.world_size if env.world_size > 1 else 1

    targs = build_parser().parse_args([
        "--model", b.model, "--dtype", b.dtype,
        "--batch-size", str(b.batch_size),
        "--bucket-cap-mb", str(b.bucket_cap_mb),
        "--grad-comm-dtype", b.grad_comm_dtype,
        *(["--sync-bn"] if b.sync_bn else []),
    ])
    trainer = Trainer(targs, env)
    dev = trainer.device
    on_gpu = dev.type == "cuda"

    # prefab synthetic batch, device-resident (BASELINE: synthetic data,
    # random-init weights; the loader is off the hot path by design)
    g = torch.Generator(device="cpu").manual_seed(1234 + env.rank)
    x = torch.randn(b.batch_size, 3, 224, 224, generator=g).to(dev)
    y = torch.randint(0, 1000, (b.batch_size,), generator=g).to(dev)
    trainer.model.train()

    def barrier_sync():
        if env.world_size > 1:
            trainer.comm.barrier()
        if on_gpu:
            torch.cuda.synchronize()

    use_graph = False
    if b.hip_graph and on_gpu:
        try:
            use_graph = trainer.enable_hip_graph(x, y)
        except Exception as e:  # never let capture break the bench contract
            print(f"warning: hip-graph capture failed ({e}); eager steps",
                  file=sys.stderr)
            use_graph = False
        if env.rank == 0 and not use_graph:
            print("warning: hip-graph capture unavailable; eager steps",
                  file=sys.stderr)
    for _ in range(b.warmup):
        trainer.graph_step() if use_graph else trainer.train_step(x, y)
    barrier_sync()

    t0 = time.perf_counter()
    for _ in range(b.steps):
        trainer.graph_step() if use_graph else trainer.train_step(x, y)
    if on_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    barrier_sync()

    # max over ranks
    t = torch.tensor([elapsed], dtype=torch.float64)
    if env.world_size > 1:
        import torch.distributed as dist
        dist.all_reduce(t, op=dist.ReduceOp.MAX)  # gloo control plane
    t_max = float(t[0])

    if env.rank == 0:
        ms = t_max / b.steps * 1000.0
        total_images = b.batch_size * n_gpus * b.steps
        line = json.dumps({
            "metric": ("images/sec (whole node) ResNet-50 ImageNet-shape"
                       if b.model == "resnet50" else
                       f"images/sec (whole node) {b.model} ImageNet-shape"),
            "value": round(total_images / t_max, 2),
            "unit": "images/sec",
            "n_gpus": n_gpus,
            "steps": b.steps,
            "warmup": b.warmup,
            "ms_per_step": round(ms, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": b.dtype,
            "data": "synthetic",
            "config": {"model": b.model,
                       "global_batch": b.batch_size * n_gpus,
                       "input": "3x224x224",
                       "parallelism": f"dp{n_gpus}"},
        })
        print(line)
        try:
            os.makedirs("gpurun_out", exist_ok=True)
            with open(os.path.join("gpurun_out", "bench.jsonl"), "a") as f:
                f.write(line + "\n")
        except OSError:
            pass
    return 0


if __name__ == "__main__":
    sys.exit(main())
