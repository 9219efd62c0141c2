"""Process launcher (SURVEY.md §2b rows 1-3, §3a — re-designed, not ported).

torch.distributed.launch-style CLI:

    python -m pytorch_distributed_train_amd.launch --nproc_per_node=8 \
        train.py --model resnet50 ...

Spawns nproc_per_node worker processes with the env contract
RANK / LOCAL_RANK / WORLD_SIZE / MASTER_ADDR / MASTER_PORT (and legacy
--local_rank argv unless --use-env), monitors them every 0.1 s, and on any
worker failure kills the whole gang and restarts it (DDP is gang-scheduled,
not per-rank elastic — api.py:716 behavior) up to --max-restarts times.
Workers are expected to resume from their last checkpoint via --resume.
"""
from __future__ import annotations

import argparse
import os
import signal
import subprocess
import sys
import time


def parse_args(argv=None):
    p = argparse.ArgumentParser(description="MI355X-native gang launcher")
    p.add_argument("--nproc_per_node", "--nproc-per-node", type=int, default=1,
                   dest="nproc_per_node")
    p.add_argument("--nnodes", type=int, default=1)
    p.add_argument("--node_rank", "--node-rank", type=int, default=0,
                   dest="node_rank")
    p.add_argument("--master_addr", "--master-addr", default="127.0.0.1",
                   dest="master_addr")
    p.add_argument("--master_port", "--master-port", type=int, default=29500,
                   dest="master_port")
    p.add_argument("--use-env", "--use_env", action="store_true",
                   dest="use_env",
                   help="pass LOCAL_RANK via env only (no --local_rank argv)")
    p.add_argument("--max-restarts", type=int, default=3)
    p.add_argument("--monitor-interval", type=float, default=0.1)
    p.add_argument("script")
    p.add_argument("script_args", nargs=argparse.REMAINDER)
    return p.parse_args(argv)


def _spawn_gang(args) -> list[subprocess.Popen]:
    procs = []
    world_size = args.nproc_per_node * args.nnodes
    for local_rank in range(args.nproc_per_node):
        rank = args.node_rank * args.nproc_per_node + local_rank
        env = dict(os.environ)
        env.update(
            RANK=str(rank),
            LOCAL_RANK=str(local_rank),
            WORLD_SIZE=str(world_size),
            LOCAL_WORLD_SIZE=str(args.nproc_per_node),
            MASTER_ADDR=args.master_addr,
            MASTER_PORT=str(args.master_port),
        )
        cmd = [sys.executable, args.script]
        if not args.use_env:
            cmd.append(f"--local_rank={local_rank}")
        cmd += args.script_args
        procs.append(subprocess.Popen(cmd, env=env))
    return procs


def _kill_gang(procs) -> None:
    for p in procs:
        if p.poll() is None:
            p.send_signal(signal.SIGTERM)
    deadline = time.time() + 10
    for p in procs:
        while p.poll() is None and time.time() < deadline:
            time.sleep(0.1)
        if p.poll() is None:
            p.kill()
            p.wait()


def main(argv=None) -> int:
    args = parse_args(argv)
    restarts = 0
    while True:
        procs = _spawn_gang(args)
        failed_rc = None
        while True:
            time.sleep(args.monitor_interval)
            codes = [p.poll() for p in procs]
            if any(c is not None and c != 0 for c in codes):
                failed_rc = next(c for c in codes if c is not None and c != 0)
                break
            if all(c == 0 for c in codes):
                return 0
        _kill_gang(procs)
        if restarts >= args.max_restarts:
            print(f"[launch] worker failed (rc={failed_rc}); "
                  f"max restarts ({args.max_restarts}) exhausted",
                  file=sys.stderr)
            return failed_rc or 1
        restarts += 1
        print(f"[launch] worker failed (rc={failed_rc}); restarting gang "
              f"({restarts}/{args.max_restarts})", file=sys.stderr)


if __name__ == "__main__":
    sys.exit(main())
