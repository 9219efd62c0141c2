"""SyncBatchNorm: cross-rank batch statistics.

Forward: each rank computes local (mean, var, count), all-gathers the
packed [mean ‖ var ‖ count] vector (2C+1 fp32 — the tiny latency-bound
collective of SURVEY.md §2d), and combines count-weighted global stats.
Backward: the BN Function all-reduces [sum_dy ‖ sum_dy_xmu] and normalizes
with the GLOBAL count (wired through ops.functional.BatchNorm2dNHWCFn via
comm/count_override).

The collective runs on the COMPUTE stream through a communicator that is
`dup()`-ed from the reducer's, so SyncBN traffic never shares enqueue
ordering with gradient-bucket traffic (deadlock-free by construction).
"""
from __future__ import annotations

import torch

from ..models.layers import BatchNorm2dNHWC
from ..ops import functional as OF
from ..ops.backend import ext


class SyncBatchNorm2dNHWC(BatchNorm2dNHWC):
    def forward(self, x, residual=None):
        """GPU fast path: conv-emitted stats slab (no activation re-read)
        + fused pack/combine kernels around the packed allgather — the
        per-layer python op chain (~12 small launches) collapses to 2
        kernels + 1 collective. CPU falls back to the generic oracle
        path."""
        comm = self.comm
        if not (self.training and x.is_cuda and comm is not None):
            return super().forward(x, residual)
        c = self.num_features
        rows = x.shape[0] * x.shape[1] * x.shape[2]
        with torch.no_grad():
            part = getattr(x, "_pdta_bn_part", None)
            if part is not None:
                mean, var = ext().bn_stats_from_part(part, rows, c)
            else:
                mean, var = OF.bn_stats(x)
            packed = ext().syncbn_pack(mean, var, float(rows))
            gathered = torch.empty(comm.world_size * (2 * c + 1),
                                   dtype=torch.float32, device=x.device)
            comm.allgather_into(gathered, packed).sync_stream()
            mean, invstd, total = ext().syncbn_combine(
                gathered, comm.world_size, c, self.eps, self.momentum,
                self.running_mean, self.running_var)
            if not getattr(self, "_nbt_batched", False):
                self.num_batches_tracked += 1
        fuse = OF.fork_fuse_enabled()
        return OF.bn_train(
            x, self.weight, self.bias, mean, invstd, self.fuse_relu, comm,
            None, residual,
            dres_sink=getattr(self, "_dres_sink", None) if fuse else None,
            dy2_source=(getattr(self, "_dy2_source", None)
                        if fuse else None),
            gslot_w=getattr(self, "_gslot_w", None),
            gslot_b=getattr(self, "_gslot_b", None))

    def _compute_stats(self, x):
        comm = self.comm
        local_count = self._batch_count(x)
        with torch.no_grad():
            mean, var = OF.bn_stats(x)
            if comm is None or comm.world_size == 1:
                return mean, var, local_count
            c = self.num_features
            packed = torch.empty(2 * c + 1, dtype=torch.float32,
                                 device=x.device)
            packed[:c] = mean * local_count                 # sum(x)
            packed[c:2 * c] = (var + mean * mean) * local_count  # sum(x^2)
            packed[2 * c] = local_count
            gathered = torch.empty(comm.world_size * (2 * c + 1),
                                   dtype=torch.float32, device=x.device)
            comm.allgather_into(gathered, packed).sync_stream()
            g = gathered.view(comm.world_size, 2 * c + 1)
            total = g[:, 2 * c].sum()
            gmean = g[:, :c].sum(dim=0) / total
            gvar = g[:, c:2 * c].sum(dim=0) / total - gmean * gmean
            gvar.clamp_(min=0.0)
            # the REDUCED count is returned as a device scalar (no host
            # sync, and exact even for ragged last batches — the
            # equal-shapes shortcut it replaces silently broke there)
            return gmean, gvar, total


def convert_sync_batchnorm(module: torch.nn.Module, comm):
    """Replace every BatchNorm2dNHWC with a SyncBatchNorm2dNHWC sharing
    parameters/buffers (torch convert_sync_batchnorm analog,
    nn/modules/batchnorm.py:842 behavior)."""
    out = module
    if isinstance(module, BatchNorm2dNHWC) and \
            not isinstance(module, SyncBatchNorm2dNHWC):
        out = SyncBatchNorm2dNHWC(module.num_features, module.eps,
                                  module.momentum)
        out.fuse_relu = module.fuse_relu
        out.comm = comm
        if hasattr(module, "_dres_sink"):
            out._dres_sink = module._dres_sink
        if hasattr(module, "_dy2_source"):
            out._dy2_source = module._dy2_source
        out.weight = module.weight
        out.bias = module.bias
        out.running_mean = module.running_mean
        out.running_var = module.running_var
        out.num_batches_tracked = module.num_batches_tracked
        out.training = module.training
    else:
        for name, child in module.named_children():
            new = convert_sync_batchnorm(child, comm)
            if new is not child:
                setattr(module, name, new)
    return out
