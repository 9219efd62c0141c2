"""Unit coverage for utils: GradScaler cycle, watchdog mechanics,
metrics logger, synthetic-dataset determinism."""
import json
import os
import time

import torch

from pytorch_distributed_train_amd.utils.amp import GradScaler
from pytorch_distributed_train_amd.utils.watchdog import (Watchdog,
                                                          record_collective,
                                                          dump_flight_recorder)
from pytorch_distributed_train_amd.utils.metrics import RankLogger
from pytorch_distributed_train_amd.data import make_dataset


def test_gradscaler_backoff_and_growth_cpu():
    """torch GradScaler semantics: overflow halves the scale and skips;
    growth_interval clean steps double it."""
    s = GradScaler(enabled=True, init_scale=2.0 ** 10, growth_interval=3)
    g = torch.randn(64)
    s0 = s.get_scale()
    # clean steps: scale grows after growth_interval
    skips = [s.unscale_and_check(g.clone() * s.get_scale()) for _ in range(3)]
    assert not any(skips)
    assert s.get_scale() == s0 * 2
    # overflow: skip + halve
    bad = g.clone()
    bad[0] = float("inf")
    assert s.unscale_and_check(bad)
    assert s.get_scale() == s0  # 2x then /2


def test_gradscaler_disabled_passthrough():
    s = GradScaler(enabled=False)
    g = torch.randn(8)
    assert s.scale(g.sum()).item() == g.sum().item()
    assert not s.unscale_and_check(g)


def test_watchdog_fires_only_after_heartbeat_stall():
    """Armed only after the first heartbeat; the flight recorder dump
    lists recorded collectives (explicit sink: the default-arg stderr is
    bound at import time and bypasses pytest capture)."""
    import io
    w = Watchdog(timeout_sec=0.2, rank=3)
    assert not w._armed
    w.heartbeat()
    assert w._armed
    record_collective("unit.allreduce", 1234)
    sink = io.StringIO()
    dump_flight_recorder(file=sink)
    out = sink.getvalue()
    assert "unit.allreduce" in out and "1234" in out


def test_watchdog_stall_hooks_called_before_abort(monkeypatch):
    called = []
    w = Watchdog(timeout_sec=0.05, rank=0)
    w.add_stall_hook(lambda: called.append("abort_comm"))
    monkeypatch.setattr(os, "abort", lambda: called.append("os_abort"))
    w.heartbeat()
    w._beat -= 10  # simulate a stall
    w.start()
    t0 = time.time()
    while "os_abort" not in called and time.time() - t0 < 5:
        time.sleep(0.02)
    w.stop()
    assert called == ["abort_comm", "os_abort"]


def test_rank_logger_jsonl(tmp_path):
    path = tmp_path / "m.jsonl"
    lg = RankLogger(rank=0, jsonl_path=str(path))
    lg.log(epoch=1, step=2, loss=3.5)
    lg.close()
    rec = json.loads(path.read_text().splitlines()[0])
    assert rec["epoch"] == 1 and rec["loss"] == 3.5
    # non-zero ranks stay silent and write nothing
    path2 = tmp_path / "m2.jsonl"
    lg2 = RankLogger(rank=1, jsonl_path=str(path2))
    lg2.log(epoch=1, loss=1.0)
    lg2.close()
    assert not path2.exists() or path2.read_text() == ""


def test_synthetic_dataset_deterministic_across_instances():
    """DDP equality depends on every rank seeing identical data for a
    given index — two instances with the same seed must agree."""
    a = make_dataset("synthetic-imagenet", 64, seed=9)
    b = make_dataset("synthetic-imagenet", 64, seed=9)
    for i in (0, 7, 63):
        xa, ya = a[i]
        xb, yb = b[i]
        assert torch.equal(xa, xb) and ya == yb
    c = make_dataset("synthetic-imagenet", 64, seed=10)
    xc, _ = c[0]
    x0, _ = a[0]
    assert not torch.equal(x0, xc)
