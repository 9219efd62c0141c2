"""Training engine: the per-step hot loop.

step:
    forward (NHWC bf16/fp16 kernels)
    fused softmax-CE loss
    backward — per-param post-accumulate hooks fill flat-gradient buckets;
               full buckets all-reduce (AVG) on the side HIP stream
    reducer.finalize() — compute stream waits bucket events
    fused SGD over the flat buffers (one kernel)
"""
from __future__ import annotations

import os
import time

import torch
from torch.profiler import record_function

from .flags import DistEnv
from .models import build_model
from .ops import functional as OF
from .parallel.comm import make_communicator, NullComm
from .parallel.ddp import GradReducer
from .parallel.flat import FlatParams
from .parallel.optim import FusedSGD
from .parallel.sync_bn import convert_sync_batchnorm
from .utils.amp import GradScaler
from .utils.watchdog import Watchdog

_DTYPES = {"fp32": torch.float32, "bf16": torch.bfloat16,
           "fp16": torch.float16}


class Trainer:
    def __init__(self, args, env: DistEnv | None = None):
        self.args = args
        self.env = env or DistEnv.from_env(getattr(args, "local_rank", None))
        dev = args.device
        if dev == "auto":
            dev = "cuda" if torch.cuda.is_available() else "cpu"
        if dev == "cuda":
            # modulo: more ranks than GPUs is legal (2-ranks-on-1-GPU RCCL
            # validation on the single-GPU box; 8-GPU node maps 1:1)
            idx = self.env.local_rank % max(torch.cuda.device_count(), 1)
            torch.cuda.set_device(idx)
            self.device = torch.device("cuda", idx)
        else:
            self.device = torch.device("cpu")
        self.compute_dtype = _DTYPES[args.dtype]

        torch.manual_seed(args.seed)  # same init on all ranks (+ broadcast)

        num_classes = 10 if args.data == "synthetic-cifar10" else 1000
        self.model = build_model(args.model, num_classes, self.compute_dtype)
        self.model.to(self.device)

        self.comm = make_communicator(self.env, self.device, args.backend)
        self._bn_comm = None
        from .parallel.comm import RcclComm
        if args.sync_bn and (self.env.world_size > 1
                             or isinstance(self.comm, RcclComm)):
            # ws-1 RCCL (PDTA_RCCL_WS1) also converts: the packed-stats
            # allgather and [sum_dy‖sum_dy_xmu‖count] allreduce then run
            # the REAL enqueue path on a single-GPU box (config #4 audit)
            self._bn_comm = self.comm.dup()
            self.model = convert_sync_batchnorm(self.model, self._bn_comm)

        self.flat = FlatParams(self.model.named_parameters(),
                               device=self.device)
        # buffer sync (torch DDP broadcast_buffers analog,
        # nn/parallel/distributed.py:1557 behavior): float buffers are
        # re-viewed into ONE flat tensor at init, so the per-step sync is a
        # single small broadcast with zero pack/unpack copies
        self._buffer_flat = None
        if getattr(args, "broadcast_buffers", False) \
                and self.env.world_size > 1:
            self._buffer_flat = self._flatten_buffers()
        # num_batches_tracked counters advance in lockstep: one flat long
        # buffer, ONE add kernel per step instead of 53 scalar adds
        self._nbt_flat = self._flatten_nbt() \
            if self.device.type == "cuda" else None
        comm_dtype = None
        if args.grad_comm_dtype == "bf16":
            comm_dtype = torch.bfloat16
        elif args.grad_comm_dtype == "fp16":
            comm_dtype = torch.float16
        self.reducer = GradReducer(self.flat, self.comm,
                                   bucket_cap_mb=args.bucket_cap_mb,
                                   overlap=not args.no_overlap,
                                   comm_dtype=comm_dtype)
        self.reducer.broadcast_params(src=0)
        if self.device.type == "cuda" \
                and not os.environ.get("PDTA_NO_DIRECT_GRAD"):
            self._wire_direct_grads()
        self.optimizer = FusedSGD(self.flat, lr=args.lr,
                                  momentum=args.momentum,
                                  weight_decay=args.weight_decay)
        self.scaler = GradScaler(enabled=(args.amp and args.dtype == "fp16"))
        self.packs = None
        if self.device.type == "cuda" and self.compute_dtype != torch.float32:
            from .ops.packs import PackManager
            self.packs = PackManager(self.model, self.flat,
                                     self.compute_dtype, self.device)
        self.epoch = 0
        self.global_step = 0
        self._fault = self._parse_fault(args.fault_inject)
        wd_sec = float(getattr(args, "watchdog_sec", -1.0))
        if wd_sec < 0:  # auto: on for distributed runs (PGNCCL is always-on)
            wd_sec = 120.0 if self.env.world_size > 1 else 0.0
        self.watchdog = Watchdog(wd_sec, rank=self.env.rank)
        for c in (self.comm, getattr(self, "_bn_comm", None)):
            if c is not None and hasattr(c, "abort"):
                self.watchdog.add_stall_hook(c.abort)
        self.watchdog.start()

    def _wire_direct_grads(self) -> None:
        """Attach (flat-gradient view, reducer notify) slots to every
        module whose backward can write its parameter gradient straight
        into flat storage — no AccumulateGrad add kernels (161 small
        launches/step on ResNet-50), no zero+add HBM round trip. The
        notify keeps the DDP bucket accounting exact (param_ready).

        ASSUMES each wired parameter contributes exactly ONE gradient
        per step (true for every model here: no weight sharing). A
        double contribution would overwrite instead of accumulate — the
        reducer raises loudly on a duplicate arrival, so a future
        weight-shared model fails fast rather than training wrong."""
        from .models.layers import (BatchNorm2dNHWC, Conv2dNHWC, Linear)
        from .models.vit import LayerNorm as VitLayerNorm
        from .models.vit import MultiheadSelfAttention

        def slot(p):
            i = self.flat.index_by_param.get(id(p))
            if i is None:
                return None
            view = self.flat.grad_view(i)
            notify = None
            if self.reducer.enabled:
                notify = (lambda idx=i: self.reducer.param_ready(idx))
                self.reducer.take_over_arrival(i)
            return (view, notify)

        for m in self.model.modules():
            if isinstance(m, Conv2dNHWC):
                m._gslot_w = slot(m.weight)
            elif isinstance(m, (BatchNorm2dNHWC, VitLayerNorm)):
                m._gslot_w = slot(m.weight)
                m._gslot_b = slot(m.bias)
            elif isinstance(m, Linear):
                m._gslot_w = slot(m.weight)
                if m.bias is not None:
                    m._gslot_b = slot(m.bias)
            elif isinstance(m, MultiheadSelfAttention):
                m._gslot_inw = slot(m.in_proj_weight)
                m._gslot_inb = slot(m.in_proj_bias)

    def _flatten_nbt(self):
        from .models.layers import BatchNorm2dNHWC
        mods = [m for m in self.model.modules()
                if isinstance(m, BatchNorm2dNHWC)]
        if not mods:
            return None
        flat = torch.zeros(len(mods), dtype=torch.long, device=self.device)
        for i, m in enumerate(mods):
            flat[i] = m.num_batches_tracked
            m._buffers["num_batches_tracked"] = flat[i]
            m._nbt_batched = True
        return flat

    def _flatten_buffers(self):
        mods = [(m, n, b) for m in self.model.modules()
                for n, b in m.named_buffers(recurse=False)
                if b.dtype == torch.float32]
        if not mods:
            return None
        total = sum(b.numel() for _, _, b in mods)
        flat = torch.empty(total, dtype=torch.float32, device=self.device)
        off = 0
        for m, name, b in mods:
            n = b.numel()
            flat[off:off + n].copy_(b.reshape(-1))
            m._buffers[name] = flat[off:off + n].view_as(b)
            off += n
        return flat

    def _sync_buffers(self) -> None:
        """Rank-0 buffer broadcast before forward; enqueued on the compute
        stream so no host sync is added on GPU."""
        if self._buffer_flat is not None and self.model.training:
            self.comm.broadcast(self._buffer_flat, src=0).sync_stream()

    @staticmethod
    def _parse_fault(spec: str):
        # "step:N:rank:R" -> (N, R, "raise"); "hang:step:N:rank:R" hangs
        if not spec:
            return None
        mode = "raise"
        if spec.startswith("hang:"):
            mode = "hang"
            spec = spec[len("hang:"):]
        parts = spec.split(":")
        return int(parts[1]), int(parts[3]), mode

    def _check_grads_finite(self):
        """NanCheck analog: one reduction over the flat gradient; raises
        with rank info so the launcher's gang restart (or the operator)
        can attribute the failure."""
        if not bool(torch.isfinite(self.flat.grad).all()):
            n_bad = int((~torch.isfinite(self.flat.grad)).sum())
            raise RuntimeError(
                f"non-finite gradient after all-reduce: rank "
                f"{self.env.rank} step {self.global_step} "
                f"({n_bad} bad elements)")

    def train_step(self, images, labels) -> torch.Tensor:
        """One optimization step; returns the (device) scalar loss."""
        if self._fault and self.global_step == self._fault[0] \
                and self.env.rank == self._fault[1]:
            if self._fault[2] == "hang":
                # desync scenario: this rank never reaches the collective;
                # peers block in all-reduce until THEIR watchdog fires
                while True:
                    time.sleep(3600)
            raise RuntimeError(
                f"fault injection: step {self.global_step} rank {self.env.rank}")
        self.optimizer.zero_grad()
        self.reducer.prepare()
        self._sync_buffers()
        if self._nbt_flat is not None and self.model.training:
            with torch.no_grad():
                self._nbt_flat += 1
        if self.packs is not None:
            self.packs.refresh()  # one kernel repacks every layer's weights
        with record_function("pdta.forward"):
            logits = self.model(images)
            loss = OF.softmax_cross_entropy(logits, labels,
                                            self.args.label_smoothing)
        with record_function("pdta.backward"):
            self.scaler.scale(loss).backward()
        with record_function("pdta.reduce_finalize"):
            self.reducer.finalize()
        skip = self.scaler.unscale_and_check(self.flat.grad, self.comm)
        # nan-check AFTER unscale and only on non-skipped steps: under fp16
        # AMP a loss-scale overflow is a recoverable event the scaler
        # handles by skipping + backoff, not a reason to kill the gang
        if not skip and getattr(self.args, 'nan_check', False):
            self._check_grads_finite()
        if not skip:
            with record_function("pdta.optimizer"):
                self.optimizer.step()
        self.global_step += 1
        self.watchdog.heartbeat()
        return loss.detach()

    # ---- hipGraph-captured step ----------------------------------------
    def enable_hip_graph(self, example_x, example_y) -> bool:
        """Capture one whole training step (weight repack, forward, loss,
        backward — including the reducer's RCCL bucket all-reduces on the
        side stream — fused SGD) into a HIP graph; graph_step() replays
        it. Launch-bound overhead (200+ kernel launches/step) collapses
        into one graph launch.

        RCCL collectives are graph-capturable (their enqueue lands on the
        side stream, which joins the capture through the bucket events);
        validated on hardware at world-1 RCCL (PDTA_RCCL_WS1, see
        tests/test_multirank_gpu.py). Multi-rank capture is opt-in via
        PDTA_DIST_GRAPH=1 until it has run on an 8-GPU node; default at
        ws>1 is eager (guaranteed-progress choice for the driver bench).
        Host-staged gloo cannot be captured."""
        from .parallel.comm import RcclComm
        if not isinstance(self.comm, (NullComm, RcclComm)) \
                or self.scaler.enabled or self.device.type != "cuda":
            return False
        if self.env.world_size > 1 \
                and os.environ.get("PDTA_DIST_GRAPH") != "1":
            return False
        self._gx = example_x.clone()
        self._gy = example_y.clone()

        def body():
            self.optimizer.zero_grad()
            self.reducer.prepare()
            self._sync_buffers()
            if self.packs is not None:
                self.packs.refresh()
            logits = self.model(self._gx)
            loss = OF.softmax_cross_entropy(logits, self._gy,
                                            self.args.label_smoothing)
            loss.backward()
            self.reducer.finalize()
            self.optimizer.step()
            return loss.detach()

        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):
                body()
        torch.cuda.current_stream().wait_stream(s)
        self._graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self._graph):
            self._gloss = body()
        return True

    def graph_step(self, images=None, labels=None) -> torch.Tensor:
        if images is not None:
            self._gx.copy_(images, non_blocking=True)
            self._gy.copy_(labels, non_blocking=True)
        self._graph.replay()
        self.global_step += 1
        self.watchdog.heartbeat()
        return self._gloss

    def state(self):
        return dict(model=self.model, optimizer=self.optimizer,
                    scaler=self.scaler, epoch=self.epoch,
                    step=self.global_step)
