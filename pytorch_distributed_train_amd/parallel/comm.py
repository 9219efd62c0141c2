"""Collective communicator abstraction.

Data plane on GPU: our own RCCL binding (ops/_hip_ops owns the ncclComm_t;
collectives are enqueued on caller-chosen HIP streams) — RCCL's topology
detection drives the 7-link xGMI all-to-all pattern. Control plane /
bootstrap and the CPU plumbing config: torch.distributed with gloo
(init_method env://), kept cleanly behind this interface (no dual GPU
paths: RcclComm is the only GPU communicator).

Collective surface mirrors what the reference exercises (SURVEY.md §2b
row 12): all_reduce, broadcast, all_gather_into_tensor, barrier.
"""
from __future__ import annotations

import os

import torch
import torch.distributed as torch_dist

from ..flags import DistEnv
from ..utils.watchdog import record_collective


class Work:
    def wait(self) -> None:  # host-side wait (CPU backends)
        pass

    def sync_stream(self, stream=None) -> None:
        """Make `stream` (default: current) wait for completion, device-side."""
        self.wait()


class _DistWork(Work):
    def __init__(self, work, post=None):
        self._work = work
        self._post = post

    def wait(self) -> None:
        if self._work is not None:
            self._work.wait()
        if self._post is not None:
            self._post()
            self._post = None


class _EventWork(Work):
    def __init__(self, event: "torch.cuda.Event"):
        self._event = event

    def wait(self) -> None:
        self._event.synchronize()

    def sync_stream(self, stream=None) -> None:
        if stream is None:
            stream = torch.cuda.current_stream()
        stream.wait_event(self._event)


class NullComm:
    """world_size == 1: every collective is a no-op."""

    rank = 0
    world_size = 1

    def allreduce_avg(self, tensor, stream=None) -> Work:
        return Work()

    def allreduce_sum(self, tensor, stream=None) -> Work:
        return Work()

    def broadcast(self, tensor, src: int = 0, stream=None) -> Work:
        return Work()

    def allgather_into(self, out, inp, stream=None) -> Work:
        out.copy_(inp)
        return Work()

    def barrier(self) -> None:
        pass

    def abort(self) -> None:
        pass

    def dup(self) -> "NullComm":
        return self


class TorchDistComm:
    """gloo via torch.distributed — CPU plumbing config (BASELINE config #1)
    and rendezvous substrate for RcclComm bootstrap."""

    def __init__(self, env: DistEnv, backend: str = "gloo"):
        if not torch_dist.is_initialized():
            os.environ.setdefault("MASTER_ADDR", env.master_addr)
            os.environ.setdefault("MASTER_PORT", str(env.master_port))
            if env.master_addr in ("127.0.0.1", "localhost"):
                # single-node: bind gloo to loopback (container hostnames
                # may not resolve — see the torchrun 127.0.0.1 contract)
                os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
            torch_dist.init_process_group(
                backend=backend, init_method="env://",
                rank=env.rank, world_size=env.world_size)
        self.rank = torch_dist.get_rank()
        self.world_size = torch_dist.get_world_size()

    def allreduce_sum(self, tensor, stream=None) -> Work:
        record_collective("gloo.allreduce_sum",
                          tensor.numel() * tensor.element_size())
        w = torch_dist.all_reduce(tensor, op=torch_dist.ReduceOp.SUM,
                                  async_op=True)
        return _DistWork(w)

    def allreduce_avg(self, tensor, stream=None) -> Work:
        record_collective("gloo.allreduce_avg",
                          tensor.numel() * tensor.element_size())
        w = torch_dist.all_reduce(tensor, op=torch_dist.ReduceOp.SUM,
                                  async_op=True)
        ws = self.world_size
        return _DistWork(w, post=lambda: tensor.div_(ws))

    def broadcast(self, tensor, src: int = 0, stream=None) -> Work:
        return _DistWork(torch_dist.broadcast(tensor, src=src, async_op=True))

    def allgather_into(self, out, inp, stream=None) -> Work:
        return _DistWork(
            torch_dist.all_gather_into_tensor(out, inp, async_op=True))

    def barrier(self) -> None:
        torch_dist.barrier()

    def abort(self) -> None:
        pass  # gloo collectives are host-blocking; os.abort reaps them

    def dup(self) -> "TorchDistComm":
        return self  # gloo enqueue is host-ordered; one PG suffices on CPU


class StagedGlooComm(TorchDistComm):
    """GPU tensors staged through host memory over gloo.

    RCCL rejects two ranks of one communicator on the same device
    ("Duplicate GPU detected", hard error in transport init), so
    multi-rank-per-GPU validation runs — on the single-GPU box only —
    with collectives staged D2H → gloo → H2D. Every other piece of the
    distributed path (reducer buckets, SyncBN packed stats, flat
    broadcast) is exercised unchanged on the HIP kernels. Never selected
    when ranks map 1:1 onto devices (the 8-GPU node runs RcclComm)."""

    def _staged(self, t, run):
        if not t.is_cuda:
            return run(t)
        if os.environ.get("PDTA_STAGED_SYNC"):
            torch.cuda.synchronize()
        host = t.cpu()  # sync copy: grads on the caller's stream are done
        w = run(host)
        return _StagedWork(w, lambda: t.copy_(host))

    def allreduce_sum(self, tensor, stream=None) -> Work:
        return self._staged(tensor,
                            lambda h: super(StagedGlooComm, self)
                            .allreduce_sum(h))

    def allreduce_avg(self, tensor, stream=None) -> Work:
        return self._staged(tensor,
                            lambda h: super(StagedGlooComm, self)
                            .allreduce_avg(h))

    def broadcast(self, tensor, src: int = 0, stream=None) -> Work:
        return self._staged(tensor,
                            lambda h: super(StagedGlooComm, self)
                            .broadcast(h, src=src))

    def allgather_into(self, out, inp, stream=None) -> Work:
        if not out.is_cuda:
            return super().allgather_into(out, inp)
        hout = out.cpu()
        hinp = inp.cpu()
        w = super().allgather_into(hout, hinp)
        return _StagedWork(w, lambda: out.copy_(hout))

    def dup(self) -> "StagedGlooComm":
        return self


class _StagedWork(Work):
    def __init__(self, inner: Work, post):
        self._inner = inner
        self._post = post

    def wait(self) -> None:
        if self._inner is not None:
            self._inner.wait()
            self._inner = None
            self._post()

    def sync_stream(self, stream=None) -> None:
        self.wait()


class RcclComm:
    """RCCL over xGMI; the extension owns the ncclComm_t.

    Bootstrap: ncclUniqueId from rank 0, exchanged over the gloo control
    plane; ncclCommInitRank on the local device. Collectives enqueue on the
    caller's HIP stream and complete device-side (sync_stream), so the
    reducer can overlap them with backward on a side stream.
    """

    def __init__(self, env: DistEnv, control: TorchDistComm, device):
        from ..ops.backend import ext
        self._ext = ext()
        self.rank = env.rank
        self.world_size = env.world_size
        self.device = device
        uid = torch.zeros(128, dtype=torch.uint8)  # NCCL_UNIQUE_ID_BYTES
        if self.rank == 0:
            uid.copy_(self._ext.rccl_unique_id())
        control.broadcast(uid, src=0).wait()
        torch.cuda.set_device(device)
        self._comm = self._ext.rccl_comm_init(uid, self.rank, self.world_size)
        self._control = control

    def _stream_ptr(self, stream):
        s = stream if stream is not None else torch.cuda.current_stream()
        return s.cuda_stream

    def _run(self, fn, tensor, stream, *args) -> Work:
        record_collective(fn.__name__,
                          tensor.numel() * tensor.element_size())
        fn(tensor, *args, self._comm, self._stream_ptr(stream))
        ev = torch.cuda.Event()
        ev.record(stream if stream is not None
                  else torch.cuda.current_stream())
        return _EventWork(ev)

    def allreduce_avg(self, tensor, stream=None) -> Work:
        return self._run(self._ext.rccl_allreduce_avg, tensor, stream)

    def allreduce_sum(self, tensor, stream=None) -> Work:
        return self._run(self._ext.rccl_allreduce_sum, tensor, stream)

    def broadcast(self, tensor, src: int = 0, stream=None) -> Work:
        return self._run(self._ext.rccl_broadcast, tensor, stream, src)

    def allgather_into(self, out, inp, stream=None) -> Work:
        record_collective("rccl_allgather",
                          inp.numel() * inp.element_size())
        self._ext.rccl_allgather(inp, out, self._comm,
                                 self._stream_ptr(stream))
        ev = torch.cuda.Event()
        ev.record(stream if stream is not None
                  else torch.cuda.current_stream())
        return _EventWork(ev)

    def barrier(self) -> None:
        self._control.barrier()
        torch.cuda.synchronize()

    def abort(self) -> None:
        """Tear down in-flight RCCL work (watchdog stall path) so a hung
        collective cannot wedge the device past process death."""
        try:
            self._ext.rccl_comm_abort(self._comm)
        except Exception:
            pass

    def dup(self) -> "RcclComm":
        """A second communicator (own ncclComm_t) for collectives that must
        not share enqueue ordering with the reducer's (e.g. SyncBN on the
        compute stream while buckets fly on the side stream)."""
        clone = object.__new__(RcclComm)
        clone._ext = self._ext
        clone.rank, clone.world_size = self.rank, self.world_size
        clone.device = self.device
        uid = torch.zeros(128, dtype=torch.uint8)
        if self.rank == 0:
            uid.copy_(self._ext.rccl_unique_id())
        self._control.broadcast(uid, src=0).wait()
        clone._comm = self._ext.rccl_comm_init(uid, self.rank,
                                               self.world_size)
        clone._control = self._control
        return clone


class DebugComm:
    """`TORCH_DISTRIBUTED_DEBUG=DETAIL` analog (SURVEY.md §5.2 — torch's
    ProcessGroupWrapper, c10d/ProcessGroupWrapper.hpp): with
    PDTA_DEBUG_COLLECTIVES=1, every collective first cross-checks its
    signature (op, dtype, numel, src/out-numel) across ranks over the
    gloo control plane and raises a descriptive error on EVERY rank
    instead of letting a mismatched collective deadlock RCCL or silently
    corrupt. Debug tool: adds one host-synchronous gloo allgather per
    collective — never enable for performance runs."""

    _OPS = {"allreduce_avg": 1, "allreduce_sum": 2, "broadcast": 3,
            "allgather_into": 4, "barrier": 5}

    def __init__(self, inner, control):
        self._inner = inner
        self._control = control
        self.rank = inner.rank
        self.world_size = inner.world_size

    @staticmethod
    def _dt_code(dtype) -> int:
        # deterministic across processes (str hash() is randomized)
        name = str(dtype)
        code = 0
        for ch in name:
            code = (code * 31 + ord(ch)) % (1 << 31)
        return code

    def _verify(self, op: str, a: int, b: int, c: int) -> None:
        sig = torch.tensor([self._OPS[op], a, b, c], dtype=torch.long)
        table = torch.zeros(self.world_size * 4, dtype=torch.long)
        # flat out: gloo's all_gather chunks it into inp-shaped pieces
        self._control.allgather_into(table, sig).wait()
        rows = table.view(self.world_size, 4)
        if not (rows == sig).all():
            names = {v: k for k, v in self._OPS.items()}
            desc = "; ".join(
                f"rank {r}: op={names.get(int(rows[r][0]), '?')} "
                f"dt={int(rows[r][1])} n={int(rows[r][2])} "
                f"x={int(rows[r][3])}" for r in range(self.world_size))
            raise RuntimeError(
                f"[PDTA_DEBUG_COLLECTIVES] collective mismatch at rank "
                f"{self.rank} issuing {op}: {desc}")

    def allreduce_avg(self, tensor, stream=None) -> Work:
        self._verify("allreduce_avg", self._dt_code(tensor.dtype),
                     tensor.numel(), 0)
        return self._inner.allreduce_avg(tensor, stream)

    def allreduce_sum(self, tensor, stream=None) -> Work:
        self._verify("allreduce_sum", self._dt_code(tensor.dtype),
                     tensor.numel(), 0)
        return self._inner.allreduce_sum(tensor, stream)

    def broadcast(self, tensor, src: int = 0, stream=None) -> Work:
        self._verify("broadcast", self._dt_code(tensor.dtype),
                     tensor.numel(), src)
        return self._inner.broadcast(tensor, src, stream)

    def allgather_into(self, out, inp, stream=None) -> Work:
        self._verify("allgather_into", self._dt_code(inp.dtype),
                     inp.numel(), out.numel())
        return self._inner.allgather_into(out, inp, stream)

    def barrier(self) -> None:
        self._verify("barrier", 0, 0, 0)
        self._inner.barrier()

    def abort(self) -> None:
        self._inner.abort()

    def dup(self):
        return DebugComm(self._inner.dup(), self._control)


def _maybe_debug(comm, control):
    if os.environ.get("PDTA_DEBUG_COLLECTIVES") == "1":
        return DebugComm(comm, control)
    return comm


def make_communicator(env: DistEnv, device, backend: str = "auto"):
    """backend: auto|rccl|gloo. auto = rccl on cuda devices, gloo on cpu."""
    if env.world_size <= 1:
        if os.environ.get("PDTA_RCCL_WS1") == "1" \
                and torch.device(device).type == "cuda":
            # validation mode: a real world-1 RCCL communicator so the
            # reducer's bucket collectives (and their hipGraph capture)
            # run the genuine enqueue path on a single-GPU box
            control = TorchDistComm(env, backend="gloo")
            return RcclComm(env, control, torch.device(device))
        return NullComm()
    dev = torch.device(device)
    if backend == "auto":
        backend = "rccl" if dev.type == "cuda" else "gloo"
    if backend == "rccl" and dev.type == "cuda" \
            and env.world_size > torch.cuda.device_count():
        # more ranks than GPUs: RCCL hard-errors on duplicate devices in
        # one communicator — stage through gloo (single-GPU validation
        # only; never taken when ranks map 1:1 to devices)
        import sys
        print(f"[comm] rank {env.rank}: {env.world_size} ranks > "
              f"{torch.cuda.device_count()} GPUs — RCCL cannot share a "
              "device; using host-staged gloo collectives",
              file=sys.stderr)
        staged = StagedGlooComm(env, backend="gloo")
        return _maybe_debug(staged, staged.dup())
    control = TorchDistComm(env, backend="gloo")
    if backend == "gloo":
        return _maybe_debug(control, control)
    if backend == "rccl":
        return _maybe_debug(RcclComm(env, control, dev), control)
    raise ValueError(f"unknown backend {backend}")
