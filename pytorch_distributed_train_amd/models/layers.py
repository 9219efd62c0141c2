"""NHWC modules over ops.functional.

Master parameters are fp32 in torch checkpoint layouts (conv [K,C,R,S],
linear [N,K]) so state_dicts interchange with torchvision-style models;
compute copies are packed per forward into the compute dtype (weights
change every step, so a cross-step cache would never hit during training;
packing is weight-sized traffic, negligible next to the conv/GEMM work).
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn

from ..ops import functional as OF


class Conv2dNHWC(nn.Module):
    def __init__(self, in_channels: int, out_channels: int, kernel_size: int,
                 stride: int = 1, padding: int = 0, bias: bool = False):
        super().__init__()
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.kernel_size = kernel_size
        self.stride = stride
        self.padding = padding
        self.weight = nn.Parameter(torch.empty(
            out_channels, in_channels, kernel_size, kernel_size))
        # kaiming normal, fan_out, relu — the torchvision resnet init
        fan_out = kernel_size * kernel_size * out_channels
        nn.init.normal_(self.weight, 0, math.sqrt(2.0 / fan_out))
        if bias:
            self.bias = nn.Parameter(torch.zeros(out_channels))
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        # two compute-dtype packs keep fwd AND bwd-data K-contiguous
        # (conv.hip header): wt [K,R,S,C] for fwd, w2 [R,S,C,K] for
        # bwd-data. With a PackManager (ops/packs.py) the packs are pool
        # views refreshed by one batched kernel per step.
        w = self.weight
        wt = getattr(self, "_pack_wt", None)
        if wt is None or wt.dtype != x.dtype:
            wt = w.permute(0, 2, 3, 1).to(x.dtype).contiguous()
            w2 = w.permute(2, 3, 1, 0).to(x.dtype).contiguous()
        else:
            w2 = self._pack_w2
        # emit_bn_stats (set by the model builder on conv->BN pairs): the
        # fwd kernel's epilogue writes per-tile channel (sum, sumsq)
        # partials so the BN that follows never re-reads y for statistics
        fuse_grads = x.is_cuda and self.training \
            and torch.is_grad_enabled() and OF.fork_fuse_enabled()
        sink = getattr(self, "_dx_sink", None) if fuse_grads else None
        gslot = (getattr(self, "_gslot_w", None)
                 if x.is_cuda and self.training else None)
        if (getattr(self, "emit_bn_stats", False) and self.training
                and x.is_cuda and self.bias is None
                and torch.is_grad_enabled()):
            y, part = OF.conv2d_nhwc(x, w, wt, w2, self.stride,
                                     self.padding, want_bn_stats=True,
                                     dx_sink=sink, gslot=gslot)
            y._pdta_bn_part = part
            return y
        y = OF.conv2d_nhwc(x, w, wt, w2, self.stride, self.padding,
                           dx_sink=sink, gslot=gslot)
        if self.bias is not None:
            y = y + self.bias.to(y.dtype)  # broadcast over trailing C
        return y

    def extra_repr(self):
        return (f"{self.in_channels}, {self.out_channels}, "
                f"kernel_size={self.kernel_size}, stride={self.stride}, "
                f"padding={self.padding}")


class BatchNorm2dNHWC(nn.Module):
    """BatchNorm over [N,H,W,C] with optional fused ReLU epilogue.

    Stats/apply decomposition mirrors the kernels; SyncBatchNorm subclasses
    override _compute_stats with the cross-rank combine.
    """

    def __init__(self, num_features: int, eps: float = 1e-5,
                 momentum: float = 0.1):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.fuse_relu = False  # set by the model builder when ReLU follows
        self.comm = None  # set on SyncBatchNorm conversion (parallel.sync_bn)
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.register_buffer("num_batches_tracked",
                             torch.tensor(0, dtype=torch.long))

    def _batch_count(self, x) -> float:
        return float(x.shape[0] * x.shape[1] * x.shape[2])

    def _compute_stats(self, x):
        with torch.no_grad():
            mean, var = OF.bn_stats(x)
        return mean, var, self._batch_count(x)

    def forward(self, x, residual=None):
        # residual: fused ResNet join y = relu(bn(x) + residual)
        if self.training:
            if x.is_cuda and self.comm is None:
                part = getattr(x, "_pdta_bn_part", None)
                if part is not None:
                    # stats pre-reduced by the producing conv's epilogue
                    rows = x.shape[0] * x.shape[1] * x.shape[2]
                    mean, invstd = OF.bn_stats_train_from_part(
                        part, rows, self.running_mean, self.running_var,
                        self.momentum, self.eps)
                else:
                    # fast path: stats + running update in one kernel pair
                    mean, invstd = OF.bn_stats_train(
                        x, self.running_mean, self.running_var,
                        self.momentum, self.eps)
                if not getattr(self, "_nbt_batched", False):
                    with torch.no_grad():
                        self.num_batches_tracked += 1
                fuse = OF.fork_fuse_enabled()
                return OF.bn_train(x, self.weight, self.bias, mean, invstd,
                                   self.fuse_relu, residual=residual,
                                   dres_sink=(getattr(self, "_dres_sink",
                                                      None)
                                              if fuse else None),
                                   dy2_source=(getattr(self, "_dy2_source",
                                                       None)
                                               if fuse else None),
                                   gslot_w=getattr(self, "_gslot_w", None),
                                   gslot_b=getattr(self, "_gslot_b", None))
            mean, var, count = self._compute_stats(x)
            with torch.no_grad():
                if not getattr(self, "_nbt_batched", False):
                    self.num_batches_tracked += 1
                m = self.momentum
                # running_var uses the unbiased estimator, as torch BN does
                # (count may be a device scalar on the SyncBN path)
                if torch.is_tensor(count):
                    unbiased = var * (count / (count - 1.0).clamp(min=1.0))
                else:
                    unbiased = var * (count / max(count - 1.0, 1.0))
                self.running_mean.mul_(1 - m).add_(mean, alpha=m)
                self.running_var.mul_(1 - m).add_(unbiased, alpha=m)
            invstd = (var + self.eps).rsqrt()
            return OF.bn_train(x, self.weight, self.bias, mean, invstd,
                               self.fuse_relu, self.comm,
                               count if self.comm is not None else None,
                               residual,
                               dres_sink=(getattr(self, "_dres_sink", None)
                                          if x.is_cuda and
                                          OF.fork_fuse_enabled() else None),
                               dy2_source=(getattr(self, "_dy2_source",
                                                   None)
                                           if x.is_cuda and
                                           OF.fork_fuse_enabled()
                                           else None),
                               gslot_w=(getattr(self, "_gslot_w", None)
                                        if x.is_cuda else None),
                               gslot_b=(getattr(self, "_gslot_b", None)
                                        if x.is_cuda else None))
        return OF.bn_infer(x, self.weight, self.bias, self.running_mean,
                           self.running_var, self.eps, self.fuse_relu,
                           residual)

    def extra_repr(self):
        return (f"{self.num_features}, eps={self.eps}, "
                f"momentum={self.momentum}, fuse_relu={self.fuse_relu}")


class Linear(nn.Module):
    def __init__(self, in_features: int, out_features: int, bias: bool = True):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        bound = 1.0 / math.sqrt(in_features)
        nn.init.uniform_(self.weight, -bound, bound)
        if bias:
            self.bias = nn.Parameter(torch.zeros(out_features))
            nn.init.uniform_(self.bias, -bound, bound)
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        packed = getattr(self, "_pack_w", None)
        packed_t = getattr(self, "_pack_wt", None)
        if packed is None or packed.dtype != x.dtype:
            packed = self.weight.to(x.dtype)  # [N,K] = B^T (MFMA layout)
            packed_t = None
        on_direct = x.is_cuda and self.training
        return OF.linear(x, self.weight, packed, self.bias, packed_t,
                         gslot_w=(getattr(self, "_gslot_w", None)
                                  if on_direct else None),
                         gslot_b=(getattr(self, "_gslot_b", None)
                                  if on_direct else None))

    def extra_repr(self):
        return (f"in_features={self.in_features}, "
                f"out_features={self.out_features}, "
                f"bias={self.bias is not None}")


class MaxPool2dNHWC(nn.Module):
    def __init__(self, kernel_size: int = 3, stride: int = 2, padding: int = 1):
        super().__init__()
        self.kernel_size, self.stride, self.padding = kernel_size, stride, padding

    def forward(self, x):
        return OF.maxpool2d(x, self.kernel_size, self.stride, self.padding)


class GlobalAvgPoolNHWC(nn.Module):
    def forward(self, x):
        return OF.global_avgpool(x)


class ReLUNHWC(nn.Module):
    def forward(self, x):
        return OF.relu(x)
