"""Rank-0 metrics: stdout lines + JSONL file.

Lines carry epoch, step, loss, images/sec, lr. The JSONL file defaults to
gpurun_out/metrics.jsonl so results survive the gpurun merge-back.
"""
from __future__ import annotations

import json
import os
import time


class StepTimer:
    """Images/sec over a sliding window; call tick(batch) each step."""

    def __init__(self, window: int = 20):
        self.window = window
        self._times = []
        self._t0 = None

    def start(self):
        self._t0 = time.perf_counter()

    def tick(self, batch_size: int) -> float:
        now = time.perf_counter()
        if self._t0 is None:
            self._t0 = now
            return 0.0
        self._times.append((now - self._t0, batch_size))
        self._t0 = now
        if len(self._times) > self.window:
            self._times.pop(0)
        dt = sum(t for t, _ in self._times)
        n = sum(b for _, b in self._times)
        return n / dt if dt > 0 else 0.0


class RankLogger:
    def __init__(self, rank: int, jsonl_path: str | None = None):
        self.rank = rank
        self.jsonl = None
        if rank == 0 and jsonl_path:
            os.makedirs(os.path.dirname(os.path.abspath(jsonl_path)),
                        exist_ok=True)
            self.jsonl = open(jsonl_path, "a")

    def log(self, **kv) -> None:
        if self.rank != 0:
            return
        msg = " ".join(f"{k}={v:.4g}" if isinstance(v, float) else f"{k}={v}"
                       for k, v in kv.items())
        print(msg, flush=True)
        if self.jsonl:
            self.jsonl.write(json.dumps(kv) + "\n")
            self.jsonl.flush()

    def close(self) -> None:
        if self.jsonl:
            self.jsonl.close()
            self.jsonl = None
