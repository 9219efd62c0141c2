"""End-to-end train.py main() on CPU: epoch loop, lr schedule, eval,
checkpoint write, resume (single process — the gloo ws=2 configs are in
test_ddp_gloo.py)."""
import json
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import train as train_mod  # repo-root entrypoint

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _argv(tmp_path, extra=()):
    return ["--model", "resnet18", "--data", "synthetic-cifar10",
            "--dtype", "fp32", "--device", "cpu", "--batch-size", "4",
            "--num-samples", "16", "--epochs", "2", "--lr", "0.01",
            "--eval-freq", "1", "--eval-batches", "2",
            "--checkpoint-dir", str(tmp_path), "--checkpoint-freq", "1",
            "--seed", "7", *extra]


def test_main_trains_evals_checkpoints(tmp_path, monkeypatch):
    monkeypatch.delenv("RANK", raising=False)
    monkeypatch.delenv("WORLD_SIZE", raising=False)
    assert train_mod.main(_argv(tmp_path)) == 0
    ckpt = tmp_path / "resnet18_checkpoint.pt"
    assert ckpt.exists()
    state = torch.load(str(ckpt), weights_only=False)
    assert state["epoch"] == 2

    # resume continues from the saved epoch without error
    assert train_mod.main(_argv(tmp_path,
                                ("--resume", str(ckpt),
                                 "--epochs", "3"))) == 0
    state = torch.load(str(ckpt), weights_only=False)
    assert state["epoch"] == 3


def test_lr_schedules():
    from pytorch_distributed_train_amd.flags import build_parser
    a = build_parser().parse_args([
        "--lr", "1.0", "--epochs", "10", "--warmup-epochs", "2",
        "--lr-schedule", "step"])
    assert train_mod.lr_at(a, 0) == 0.5        # linear warmup 1/2
    assert train_mod.lr_at(a, 1) == 1.0        # warmup done
    assert train_mod.lr_at(a, 2) == 1.0        # before 30%
    assert abs(train_mod.lr_at(a, 3) - 0.1) < 1e-9   # past 30%
    assert abs(train_mod.lr_at(a, 6) - 0.01) < 1e-9  # past 60%
    assert abs(train_mod.lr_at(a, 8) - 0.001) < 1e-9  # past 80%
    a = build_parser().parse_args([
        "--lr", "1.0", "--epochs", "10", "--lr-schedule", "cosine"])
    assert abs(train_mod.lr_at(a, 0) - 1.0) < 1e-9
    assert train_mod.lr_at(a, 9) < 0.05


def test_bench_under_torchrun_cpu(tmp_path):
    """The driver launches bench.py via `python -m torch.distributed.run
    --nnodes=1 --nproc-per-node N --master-addr 127.0.0.1 bench.py
    --gpus N ...` — verify that exact invocation produces an n_gpus=N
    record (CPU/gloo here; RCCL on the GPU node)."""
    import json
    import subprocess
    import sys
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29741", "bench.py", "--gpus", "2",
         "--steps", "1", "--warmup", "0", "--model", "resnet18",
         "--batch-size", "2"],
        cwd=REPO, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    rec = json.loads(line)
    assert rec["n_gpus"] == 2 and rec["config"]["parallelism"] == "dp2"


def test_train_deterministic_across_runs(tmp_path):
    """Same seed, same flags => bit-identical loss trajectory (CPU):
    the loaders, sampler, init and step are all deterministic."""
    import subprocess
    outs = []
    for _ in range(2):
        r = subprocess.run(
            [sys.executable, "train.py", "--model", "resnet18",
             "--data", "synthetic-cifar10", "--batch-size", "8",
             "--num-samples", "64", "--epochs", "2", "--device", "cpu",
             "--print-freq", "1", "--seed", "123"],
            cwd=REPO, capture_output=True, text=True, timeout=600)
        assert r.returncode == 0, r.stderr[-1500:]
        import re
        outs.append([re.sub(r"images_per_sec=[0-9.]+", "", l)
                     for l in r.stdout.splitlines() if "loss=" in l])
    assert outs[0] == outs[1] and len(outs[0]) >= 8


def test_torch_profile_flag_writes_trace(tmp_path, monkeypatch):
    """--torch-profile N captures the first N steps and rank 0 exports a
    chrome trace (SURVEY.md §5.1 step-level overlap visualization)."""
    import train as train_mod
    out = tmp_path / "trace.json"
    monkeypatch.chdir(tmp_path)
    rc = train_mod.main([
        "--model", "resnet18", "--data", "synthetic-cifar10",
        "--num-samples", "64", "--batch-size", "16", "--epochs", "1",
        "--steps-per-epoch", "4", "--device", "cpu",
        "--torch-profile", "3", "--torch-profile-out", str(out)])
    assert rc == 0
    assert out.exists() and out.stat().st_size > 100


def test_bench_self_launch_2ranks_cpu(tmp_path):
    """`python bench.py --gpus 2` with no rendezvous env must spawn 2
    local ranks and print an n_gpus=2 record — CPU/gloo variant of the
    driver contract so every round covers the bare invocation."""
    import json
    import subprocess
    env = dict(os.environ)
    for k in ("RANK", "LOCAL_RANK", "WORLD_SIZE", "MASTER_ADDR",
              "MASTER_PORT"):
        env.pop(k, None)
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "bench.py", "--gpus", "2", "--steps", "2",
         "--warmup", "1", "--model", "resnet18", "--batch-size", "4"],
        cwd=repo, env=env, capture_output=True, text=True, timeout=420)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    rec = json.loads(line)
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "dp2"
    assert rec["value"] > 0
