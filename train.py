#!/usr/bin/env python3
"""train.py — reference-style entrypoint.

Same launch contract as the reference (torch.distributed.launch-style):

  # our launcher:
  python -m pytorch_distributed_train_amd.launch --nproc_per_node=8 \
      train.py --model resnet50 --dtype bf16 --batch-size 256
  # or torchrun:
  python -m torch.distributed.run --nproc-per-node 8 \
      --master-addr 127.0.0.1 train.py --model resnet50

Reads RANK / LOCAL_RANK / WORLD_SIZE / MASTER_ADDR / MASTER_PORT (or legacy
--local_rank argv), trains on synthetic data, prints rank-0 loss/images/sec,
checkpoints on rank 0, resumes with --resume.
"""
from __future__ import annotations

import os
import sys

import torch

from pytorch_distributed_train_amd.flags import build_parser, DistEnv
from pytorch_distributed_train_amd.engine import Trainer
from pytorch_distributed_train_amd.data import (DistributedSampler, Loader,
                                                make_dataset)
from pytorch_distributed_train_amd.utils.checkpoint import (load_checkpoint,
                                                            save_checkpoint)
from pytorch_distributed_train_amd.utils.metrics import RankLogger, StepTimer


def evaluate(trainer, dataset, env, batch_size, max_batches=0):
    """Top-1/top-5 on a synthetic validation split. Counts are summed
    across ranks (one tiny all-reduce), so every rank returns the same
    global accuracy."""
    from pytorch_distributed_train_amd.data import DistributedSampler, Loader
    sampler = DistributedSampler(dataset, env.world_size, env.rank,
                                 shuffle=False)
    loader = Loader(dataset, sampler, batch_size)
    trainer.model.eval()
    top1 = top5 = total = 0
    with torch.no_grad():
        for i, (images, labels) in enumerate(loader):
            if max_batches and i >= max_batches:
                break
            images = images.to(trainer.device)
            labels = labels.to(trainer.device)
            logits = trainer.model(images).float()
            _, pred = logits.topk(5, dim=1)
            correct = pred.eq(labels.view(-1, 1))
            top1 += int(correct[:, 0].sum())
            top5 += int(correct.any(dim=1).sum())
            total += labels.numel()
    trainer.model.train()
    if env.world_size > 1:
        counts = torch.tensor([top1, top5, total], dtype=torch.float32,
                              device=trainer.device)
        trainer.comm.allreduce_sum(counts).wait()
        if counts.is_cuda:
            torch.cuda.synchronize()
        top1, top5, total = (int(v) for v in counts.tolist())
    return (top1 / max(total, 1), top5 / max(total, 1))


def lr_at(args, epoch: int) -> float:
    """Step (x0.1 at 30/60/80% of epochs) or cosine schedule with linear
    warmup — the reference-style training recipes."""
    import math
    if args.warmup_epochs and epoch < args.warmup_epochs:
        return args.lr * (epoch + 1) / args.warmup_epochs
    if args.lr_schedule == "step":
        frac = epoch / max(args.epochs, 1)
        scale = 1.0
        for boundary in (0.3, 0.6, 0.8):
            if frac >= boundary:
                scale *= 0.1
        return args.lr * scale
    if args.lr_schedule == "cosine":
        t = (epoch - args.warmup_epochs) / max(
            args.epochs - args.warmup_epochs, 1)
        return args.lr * 0.5 * (1 + math.cos(math.pi * t))
    return args.lr


def main(argv=None):
    args = build_parser().parse_args(argv)
    env = DistEnv.from_env(args.local_rank)
    trainer = Trainer(args, env)

    dataset = make_dataset(args.data, args.num_samples, seed=args.seed)
    sampler = DistributedSampler(dataset, env.world_size, env.rank,
                                 shuffle=True, seed=args.seed)
    loader = Loader(dataset, sampler, args.batch_size,
                    pin_memory=(trainer.device.type == "cuda"),
                    workers=args.workers, worker_mode=args.worker_mode)

    start_epoch = 0
    if args.resume:
        state = load_checkpoint(args.resume, trainer.model, trainer.optimizer,
                                trainer.scaler, map_location=trainer.device)
        start_epoch = state.get("epoch", 0)
        trainer.global_step = state.get("step", 0)
        if env.rank == 0:
            print(f"resumed from {args.resume} at epoch {start_epoch}")
    # DDP init broadcast keeps ranks identical after resume as well
    trainer.reducer.broadcast_params(src=0)

    logger = RankLogger(env.rank, os.path.join("gpurun_out", "metrics.jsonl")
                        if os.path.isdir("gpurun_out") else None)
    timer = StepTimer()

    ckpt_path = None
    if args.checkpoint_dir:
        ckpt_path = os.path.join(args.checkpoint_dir,
                                 f"{args.model}_checkpoint.pt")

    profiler = None
    if args.torch_profile > 0 and start_epoch < args.epochs:
        # step-level overlap trace (compute vs reducer-stream collectives)
        # via torch.profiler -> Kineto -> roctracer; spans come from the
        # record_function markers in engine.Trainer.train_step
        from torch.profiler import (profile, schedule, ProfilerActivity,
                                    tensorboard_trace_handler)  # noqa: F401
        acts = [ProfilerActivity.CPU]
        if torch.cuda.is_available():
            acts.append(ProfilerActivity.CUDA)

        def _save(prof):
            if env.rank == 0:
                out = args.torch_profile_out
                os.makedirs(os.path.dirname(out) or ".", exist_ok=True)
                prof.export_chrome_trace(out)
                print(f"torch-profile trace -> {out}")

        profiler = profile(
            activities=acts,
            schedule=schedule(wait=0, warmup=min(2, args.torch_profile - 1),
                              active=max(1, args.torch_profile -
                                         min(2, args.torch_profile - 1)),
                              repeat=1),
            on_trace_ready=_save)
        profiler.__enter__()

    for epoch in range(start_epoch, args.epochs):
        trainer.epoch = epoch
        trainer.optimizer.lr = lr_at(args, epoch)
        sampler.set_epoch(epoch)
        trainer.model.train()
        timer.start()
        for step, (images, labels) in enumerate(loader):
            if args.steps_per_epoch and step >= args.steps_per_epoch:
                break
            images = images.to(trainer.device, non_blocking=True)
            labels = labels.to(trainer.device, non_blocking=True)
            loss = trainer.train_step(images, labels)
            if profiler is not None:
                profiler.step()
                if epoch == start_epoch and step + 1 >= args.torch_profile:
                    profiler.__exit__(None, None, None)
                    profiler = None
            ips = timer.tick(args.batch_size * env.world_size)
            if step % args.print_freq == 0:
                logger.log(epoch=epoch, step=step, loss=float(loss),
                           images_per_sec=round(ips, 1),
                           lr=trainer.optimizer.lr)
        if args.eval_freq and (epoch + 1) % args.eval_freq == 0:
            t1, t5 = evaluate(trainer, dataset, env, args.batch_size,
                              max_batches=args.eval_batches)
            logger.log(epoch=epoch, top1=round(t1, 4), top5=round(t5, 4))
        if ckpt_path and env.rank == 0 and (
                epoch == args.epochs - 1 or
                (args.checkpoint_freq and
                 (epoch + 1) % args.checkpoint_freq == 0)):
            save_checkpoint(ckpt_path, epoch + 1, trainer.global_step,
                            trainer.model, trainer.optimizer, trainer.scaler)
            logger.log(epoch=epoch, checkpoint=ckpt_path)
    if profiler is not None:  # fewer steps ran than --torch-profile
        profiler.__exit__(None, None, None)
    logger.close()
    return 0


def _recorded_main():
    """@record analog: serialize worker failures so
    the launcher/operator can see which rank died and why."""
    try:
        return main()
    except Exception:
        import traceback
        rank = os.environ.get("RANK", "0")
        os.makedirs("gpurun_out", exist_ok=True)
        path = os.path.join("gpurun_out", f"worker_error_rank{rank}.txt")
        with open(path, "w") as f:
            traceback.print_exc(file=f)
        traceback.print_exc()
        return 1


if __name__ == "__main__":
    sys.exit(_recorded_main())
