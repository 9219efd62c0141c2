"""Batch loader over a sampler.

The reference uses torch DataLoader with worker processes + pinned memory
(SURVEY.md §2b row 14). For synthetic data the cost is batch assembly
(host randn), not disk IO, so the MI355X-native loader defaults to a
background prefetch thread pool (torch ops release the GIL) filling a
bounded queue of pinned batches; the trainer overlaps H2D copy with
compute via non_blocking copies. worker_mode="process" gives torch-
DataLoader-style worker PROCESSES (fork, shared-memory tensor transport)
for datasets whose assembly holds the GIL. workers=0 falls back to
synchronous assembly.
"""
from __future__ import annotations

import queue
import threading

import torch


def _proc_worker(dataset, batch_size, task_q, out_q):
    """Worker-process loop: pull (index, idx-list) tasks, collate, ship
    the batch back through a shared-memory queue. A None task stops."""
    while True:
        task = task_q.get()
        if task is None:
            return
        bi, idxs = task
        imgs = []
        labels = torch.empty(len(idxs), dtype=torch.long)
        for j, i in enumerate(idxs):
            img, lab = dataset[i]
            imgs.append(img)
            labels[j] = lab
        out_q.put((bi, torch.stack(imgs), labels))


class Loader:
    def __init__(self, dataset, sampler, batch_size: int, pin_memory: bool = False,
                 drop_last: bool = True, workers: int = 0, prefetch: int = 4,
                 worker_mode: str = "thread"):
        self.dataset = dataset
        self.sampler = sampler
        self.batch_size = batch_size
        self.pin_memory = pin_memory
        self.drop_last = drop_last
        self.workers = workers
        self.prefetch = max(prefetch, 2)
        self.worker_mode = worker_mode

    def __len__(self) -> int:
        n = len(self.sampler)
        if self.drop_last:
            return n // self.batch_size
        return (n + self.batch_size - 1) // self.batch_size

    def _batches(self):
        idxs = []
        for i in self.sampler:
            idxs.append(i)
            if len(idxs) == self.batch_size:
                yield idxs
                idxs = []
        if idxs and not self.drop_last:
            yield idxs

    def _collate(self, idxs):
        imgs = []
        labels = torch.empty(len(idxs), dtype=torch.long)
        for j, i in enumerate(idxs):
            img, lab = self.dataset[i]
            imgs.append(img)
            labels[j] = lab
        x = torch.stack(imgs)
        if self.pin_memory:
            x = x.pin_memory()
            labels = labels.pin_memory()
        return x, labels

    def __iter__(self):
        if self.workers <= 0:
            for idxs in self._batches():
                yield self._collate(idxs)
            return
        if self.worker_mode == "process":
            yield from self._iter_procs()
        else:
            yield from self._iter_prefetch()

    def _iter_procs(self):
        """torch-DataLoader-style worker processes (reference row 2b#14):
        SPAWN context — fork would inherit whatever locks other threads
        (watchdog, prefetch, torch internals) hold at fork time and
        deadlock (observed); tensors return through shared memory;
        batches are re-ordered so every rank sees the identical
        sequence; pinning happens in the parent (pinned pages don't
        survive pickling)."""
        import torch.multiprocessing as mp
        ctx = mp.get_context("spawn")
        task_q = ctx.Queue()
        out_q = ctx.Queue()
        batches = list(self._batches())
        for bi, idxs in enumerate(batches):
            task_q.put((bi, idxs))
        # no stop sentinels: a worker that exits before the consumer maps
        # its shared-memory FDs would EOF the queue — workers idle on
        # task_q.get() until the finally terminates them
        procs = [ctx.Process(target=_proc_worker,
                             args=(self.dataset, self.batch_size, task_q,
                                   out_q), daemon=True)
                 for _ in range(self.workers)]
        for p in procs:
            p.start()
        results = {}
        try:
            for next_bi in range(len(batches)):
                while next_bi not in results:
                    try:
                        bi, x, labels = out_q.get(timeout=5.0)
                    except queue.Empty:
                        if not any(p.is_alive() for p in procs):
                            raise RuntimeError(
                                "all loader worker processes died "
                                "(spawn re-imports __main__: run from a "
                                "script, not stdin)")
                        continue
                    results[bi] = (x, labels)
                x, labels = results.pop(next_bi)
                if self.pin_memory:
                    x = x.pin_memory()
                    labels = labels.pin_memory()
                yield x, labels
        finally:
            for p in procs:
                if p.is_alive():
                    p.terminate()
            for p in procs:
                p.join(timeout=5.0)

    def _iter_prefetch(self):
        batch_q: "queue.Queue" = queue.Queue()
        out_q: "queue.Queue" = queue.Queue(maxsize=self.prefetch)
        batches = list(self._batches())
        for bi, idxs in enumerate(batches):
            batch_q.put((bi, idxs))
        stop = threading.Event()
        results: dict = {}
        lock = threading.Lock()

        def worker():
            while not stop.is_set():
                try:
                    bi, idxs = batch_q.get_nowait()
                except queue.Empty:
                    return
                data = self._collate(idxs)
                with lock:
                    results[bi] = data
                # interruptible put: a consumer that abandons the iterator
                # early (steps-per-epoch break, exception) sets stop; an
                # unconditional blocking put would leak this thread and its
                # pinned batch forever
                while not stop.is_set():
                    try:
                        out_q.put(bi, timeout=0.1)
                        break
                    except queue.Full:
                        continue

        threads = [threading.Thread(target=worker, daemon=True)
                   for _ in range(self.workers)]
        for t in threads:
            t.start()
        try:
            # deliver in order (ranks must see identical batch sequence)
            next_bi = 0
            ready = set()
            while next_bi < len(batches):
                while next_bi not in ready:
                    ready.add(out_q.get())
                with lock:
                    data = results.pop(next_bi)
                yield data
                next_bi += 1
        finally:
            stop.set()
            # unblock any worker already parked in a pre-stop put()
            try:
                while True:
                    out_q.get_nowait()
            except queue.Empty:
                pass
            for t in threads:
                t.join(timeout=5.0)
            with lock:
                results.clear()
