"""Collective watchdog + flight recorder (SURVEY.md §5.2 — the slim 90%
of ProcessGroupNCCL's watchdog/HeartbeatMonitor/FlightRecorder).

Every communicator call is recorded in a per-process ring buffer
(op, tensor bytes, seq, timestamp). The watchdog thread checks that the
trainer's heartbeat (bumped at the end of every reducer.finalize and every
train step) keeps moving; if it stalls longer than the timeout, the last N
collectives and all python stacks are dumped to stderr and the process
aborts (so the gang launcher can restart from the last checkpoint rather
than hang in a desynchronized collective forever).

Enable via Trainer (PDTA_WATCHDOG_SEC env or --watchdog-sec flag; 0 = off).
"""
from __future__ import annotations

import collections
import faulthandler
import os
import sys
import threading
import time

_RING = collections.deque(maxlen=64)
_LOCK = threading.Lock()
_SEQ = [0]


def record_collective(op: str, nbytes: int) -> int:
    with _LOCK:
        _SEQ[0] += 1
        seq = _SEQ[0]
        _RING.append((seq, op, nbytes, time.time()))
    return seq


def dump_flight_recorder(file=sys.stderr) -> None:
    with _LOCK:
        entries = list(_RING)
    print(f"[watchdog] flight recorder — last {len(entries)} collectives:",
          file=file)
    now = time.time()
    for seq, op, nbytes, ts in entries:
        print(f"  #{seq:6d} {op:16s} {nbytes:>12d} B  {now - ts:8.2f}s ago",
              file=file)


class Watchdog:
    def __init__(self, timeout_sec: float, rank: int = 0):
        self.timeout = timeout_sec
        self.rank = rank
        self._beat = time.time()
        self._armed = False
        self._stop = threading.Event()
        self._thread = None
        self._on_stall = []

    def add_stall_hook(self, fn) -> None:
        """Called (best-effort) on stall before the process aborts —
        e.g. RcclComm.abort so in-flight collectives are torn down
        (ncclCommAbort analog of ProcessGroupNCCL's watchdog)."""
        self._on_stall.append(fn)

    def start(self) -> None:
        if self.timeout <= 0 or self._thread is not None:
            return
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="pdta-watchdog")
        self._thread.start()

    def heartbeat(self) -> None:
        self._beat = time.time()
        self._armed = True

    def stop(self) -> None:
        self._stop.set()

    def _run(self) -> None:
        while not self._stop.wait(min(self.timeout / 4, 10.0)):
            if not self._armed:
                continue
            stalled = time.time() - self._beat
            if stalled > self.timeout:
                print(f"[watchdog] rank {self.rank}: no progress for "
                      f"{stalled:.1f}s (timeout {self.timeout}s) — "
                      "dumping state and aborting", file=sys.stderr)
                dump_flight_recorder()
                faulthandler.dump_traceback(file=sys.stderr)
                for fn in self._on_stall:
                    try:
                        fn()
                    except Exception:
                        pass
                sys.stderr.flush()
                os.abort()  # SIGABRT -> launcher restarts the gang
