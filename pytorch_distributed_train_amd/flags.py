"""Argparse + env-var contract.

Mirrors the reference's pure-argparse flag surface (reference train.py is a
torch.distributed.launch-style script) plus the env contract
RANK / WORLD_SIZE / LOCAL_RANK / MASTER_ADDR / MASTER_PORT used by the
launcher (SURVEY.md §1 L1->L7 interface).
"""
from __future__ import annotations

import argparse
import os
from dataclasses import dataclass


MODEL_CHOICES = ("resnet18", "resnet34", "resnet50", "resnet101",
                 "resnet152", "vit_b_16")


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(
        description="MI355X-native distributed data-parallel training harness"
    )
    # legacy torch.distributed.launch passes --local-rank/--local_rank argv
    # unless --use-env is given.
    p.add_argument("--local_rank", "--local-rank", type=int, default=None,
                   dest="local_rank",
                   help="set by the legacy launcher; prefer LOCAL_RANK env")
    p.add_argument("--model", default="resnet50", choices=MODEL_CHOICES)
    p.add_argument("--batch-size", type=int, default=256,
                   help="per-process (per-GPU) batch size")
    p.add_argument("--epochs", type=int, default=1)
    p.add_argument("--steps-per-epoch", type=int, default=0,
                   help="cap steps per epoch (0 = full sampler length)")
    p.add_argument("--lr", type=float, default=0.1)
    p.add_argument("--lr-schedule", default="none",
                   choices=("none", "step", "cosine"),
                   help="step: x0.1 at 30/60/80%% of epochs; cosine anneal")
    p.add_argument("--warmup-epochs", type=int, default=0)
    p.add_argument("--momentum", type=float, default=0.9)
    p.add_argument("--weight-decay", type=float, default=1e-4)
    p.add_argument("--dtype", default="bf16", choices=("fp32", "bf16", "fp16"),
                   help="compute dtype for activations/GEMMs; master weights fp32")
    p.add_argument("--sync-bn", action="store_true",
                   help="use SyncBatchNorm (cross-rank stats)")
    p.add_argument("--data", default="synthetic-imagenet",
                   choices=("synthetic-imagenet", "synthetic-cifar10"),
                   help="synthetic dataset shape (no network: random data)")
    p.add_argument("--num-samples", type=int, default=0,
                   help="synthetic dataset size (0 = default per dataset)")
    p.add_argument("--workers", type=int, default=0,
                   help="data loader workers (0 = synchronous)")
    p.add_argument("--worker-mode", default="thread",
                   choices=("thread", "process"),
                   help="prefetch via GIL-releasing threads (synthetic "
                        "default) or torch-DataLoader-style worker "
                        "processes")
    p.add_argument("--bucket-cap-mb", type=int, default=50,
                   help="DDP gradient bucket size; tuned for 7-link xGMI "
                        "(fewer, larger buckets than NVSwitch defaults)")
    p.add_argument("--grad-comm-dtype", default="same",
                   choices=("same", "bf16", "fp16"),
                   help="compress gradient all-reduce payload")
    p.add_argument("--no-overlap", action="store_true",
                   help="disable backward/all-reduce overlap (debugging)")
    p.add_argument("--broadcast-buffers", dest="broadcast_buffers",
                   action="store_true", default=True,
                   help="broadcast module buffers (BN running stats) from "
                        "rank 0 each training step (torch DDP "
                        "broadcast_buffers=True parity)")
    p.add_argument("--no-broadcast-buffers", dest="broadcast_buffers",
                   action="store_false")
    p.add_argument("--backend", default="auto", choices=("auto", "rccl", "gloo"),
                   help="collective backend; auto = rccl on GPU, gloo on CPU")
    p.add_argument("--device", default="auto", choices=("auto", "cuda", "cpu"))
    p.add_argument("--seed", type=int, default=1234)
    p.add_argument("--print-freq", type=int, default=20)
    p.add_argument("--checkpoint-dir", default="")
    p.add_argument("--checkpoint-freq", type=int, default=0,
                   help="save every N epochs on rank 0 (0 = only at end)")
    p.add_argument("--resume", default="", help="checkpoint path to resume from")
    p.add_argument("--label-smoothing", type=float, default=0.0)
    p.add_argument("--amp", action="store_true",
                   help="fp16 loss-scaling path (GradScaler semantics)")
    p.add_argument("--eval-freq", type=int, default=0,
                   help="evaluate top-1/top-5 every N epochs (0 = never)")
    p.add_argument("--eval-batches", type=int, default=10,
                   help="cap eval batches (synthetic data)")
    p.add_argument("--watchdog-sec", type=float,
                   default=float(os.environ.get("PDTA_WATCHDOG_SEC", "-1")),
                   help="abort + dump flight recorder + ncclCommAbort if no "
                        "training progress for this long; -1 = auto (120 s "
                        "when distributed, off single-process — the "
                        "always-on ProcessGroupNCCL watchdog analog); 0 = "
                        "off")
    p.add_argument("--torch-profile", type=int, default=0,
                   help="capture a torch.profiler (Kineto/roctracer) "
                        "trace of the first N steps of the first epoch; "
                        "rank 0 writes a chrome trace to "
                        "--torch-profile-out (SURVEY.md §5.1 step-level "
                        "overlap visualization; kernel-level counters "
                        "come from rocprofv3 instead)")
    p.add_argument("--torch-profile-out",
                   default="gpurun_out/torch_trace.json")
    p.add_argument("--fault-inject", default="",
                   help="step:N:rank:R -> raise at step N on rank R; "
                        "hang:step:N:rank:R -> block forever instead "
                        "(watchdog/desync testing)")
    p.add_argument("--nan-check", action="store_true",
                   help="after each backward+all-reduce, verify the flat "
                        "gradient is finite and abort with rank info if "
                        "not (ProcessGroupNCCL NanCheck analog; adds one "
                        "host sync per step)")
    return p


@dataclass
class DistEnv:
    """The rendezvous env contract (rendezvous.py:243 analog)."""

    rank: int = 0
    world_size: int = 1
    local_rank: int = 0
    master_addr: str = "127.0.0.1"
    master_port: int = 29500

    @staticmethod
    def from_env(local_rank_arg: int | None = None) -> "DistEnv":
        env = os.environ
        local_rank = local_rank_arg
        if local_rank is None:
            local_rank = int(env.get("LOCAL_RANK", "0"))
        return DistEnv(
            rank=int(env.get("RANK", "0")),
            world_size=int(env.get("WORLD_SIZE", "1")),
            local_rank=local_rank,
            master_addr=env.get("MASTER_ADDR", "127.0.0.1"),
            master_port=int(env.get("MASTER_PORT", "29500")),
        )

    @property
    def distributed(self) -> bool:
        return self.world_size > 1
