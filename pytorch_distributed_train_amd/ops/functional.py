"""Autograd ops over NHWC activations.

Every op has two implementations:
  - CUDA (ROCm) tensors -> the in-tree gfx950 HIP extension (backend.ext()),
    which raises loudly if not built — no silent eager fallback on GPU;
  - CPU tensors -> a plain PyTorch fp32 reference. This is the numerics
    oracle the GPU kernels are tested against (tests/test_kernels_gpu.py)
    and what the gloo world_size=2 CPU plumbing config runs on.

Layout contract: activations are [N, H, W, C] contiguous ("NHWC"); C is
innermost so channel vectors are contiguous for MFMA K-dim staging and BN
reductions. Conv weights: master fp32 [K, C, R, S] (torch checkpoint layout,
SURVEY.md §5.4 checkpoint parity), packed compute copy [R, S, C, K] in the
compute dtype. Linear weights: master fp32 [N, K] (torch layout), packed =
same layout cast to the compute dtype (the MFMA GEMM consumes B^T = [N, K]).

Reference behavior being rebuilt: MIOpen conv, ATen
batch_norm kernels, fused log-softmax+NLL, maxpool/avgpool — re-designed
as CDNA4 kernels, not ported.
"""
from __future__ import annotations

import torch
import torch.nn.functional as F

from .backend import ext


def _nhwc_to_nchw(x):
    return x.permute(0, 3, 1, 2).contiguous()


def _nchw_to_nhwc(x):
    return x.permute(0, 2, 3, 1).contiguous()


def conv_out_dim(h: int, k: int, stride: int, pad: int) -> int:
    return (h + 2 * pad - k) // stride + 1


# ---------------------------------------------------------------------------
# Convolution (implicit GEMM on GPU)
# ---------------------------------------------------------------------------

class GradHolder:
    """Per-block, per-iteration slot carrying the residual-path gradient
    (GradHolder protocol): the producer Function (the block's
    last BN `dres`, or the downsample conv's `dx`) stores the tensor here
    and returns None to autograd — no fork add exists; the consumer (the
    PREVIOUS block's BN backward) sums it as a second dy inside its
    reduce/elemt kernels and clears the slot. A slot found still-full at
    the next forward means the fold never happened, and the block raises.
    Pure python; the tensors it references are ordinary stream-ordered
    outputs, so this works in eager mode and under hipGraph capture."""
    __slots__ = ("t",)

    def __init__(self):
        self.t = None


def fork_fuse_enabled() -> bool:
    import os
    return not os.environ.get("PDTA_NO_FORK_FUSE")


class Conv2dNHWC(torch.autograd.Function):
    """y[n,p,q,k] = sum_{r,s,c} x[n, p*st-pad+r, q*st-pad+s, c] * w[k,c,r,s].

    GPU: implicit-GEMM MFMA kernels (csrc/conv.hip). Two compute-dtype
    weight packs keep every direction K-contiguous ("nt"-shaped):
      wt [K,R,S,C] for forward, w2 [R,S,C,K] for backward-data (see
      conv.hip header). Gradient is returned for the master fp32 weight
    directly in its [K,C,R,S] layout (the wrw kernel scatters there).
    """

    @staticmethod
    def forward(ctx, x, weight, wt, w2, stride: int, pad: int,
                want_bn_stats: bool = False, dx_sink=None, gslot=None):
        ctx.stride, ctx.pad = stride, pad
        ctx.dx_sink = dx_sink
        ctx.gslot = gslot  # (grad_view, notify) direct-grad destination
        # the bn-stats `part` output is non-differentiable; without this,
        # autograd MATERIALIZES a zero gradient for it every backward
        # (measured: ~53 FillFunctor launches/step on ResNet-50)
        ctx.set_materialize_grads(False)
        if x.is_cuda:
            # stem (C % 64 != 0): cache the im2col image for bwd-weight.
            # NOTE grad mode is always DISABLED inside a custom Function's
            # forward, so gate on the weight's requires_grad instead (a
            # torch.is_grad_enabled() gate here is always False — measured
            # as a full extra im2col pass per step)
            want_xcol = x.shape[3] % 64 != 0 and weight.requires_grad
            out = ext().conv2d_fwd(x, wt, stride, pad, want_bn_stats,
                                   want_xcol)
            xcol = out[-1] if want_xcol else None
            ctx.save_for_backward(x, weight, w2, xcol)
            if want_bn_stats:
                y, part = out[0], out[1]
                ctx.mark_non_differentiable(part)
                return y, part
            return out[0]
        ctx.save_for_backward(x, weight, w2, None)
        # CPU reference (fp32)
        w = weight.float()
        y = F.conv2d(_nhwc_to_nchw(x).float(), w, stride=stride, padding=pad)
        return _nchw_to_nhwc(y).to(x.dtype)

    @staticmethod
    def backward(ctx, dy, _dpart=None):
        x, weight, w2, xcol = ctx.saved_tensors
        stride, pad = ctx.stride, ctx.pad
        dy = dy.contiguous()
        need_dx = ctx.needs_input_grad[0]
        need_dw = ctx.needs_input_grad[1]
        dx = dw = None
        if x.is_cuda:
            if need_dx:
                dx = ext().conv2d_bwd_data(dy, w2, x.shape[1], x.shape[2],
                                           stride, pad)
                if ctx.dx_sink is not None:
                    # residual-path producer (GradHolder protocol): the
                    # previous block's BN backward reads this as its
                    # second dy; autograd sees None — no fork add exists
                    ctx.dx_sink.t = dx
                    dx = None
            if need_dw:
                if ctx.gslot is not None:
                    # direct-grad: write straight into the flat slice and
                    # return None (no AccumulateGrad add); tell the
                    # reducer the param arrived
                    view, notify = ctx.gslot
                    ext().conv2d_bwd_weight(x, dy, weight.shape[2],
                                            weight.shape[3], stride, pad,
                                            xcol_cached=xcol, out=view)
                    if notify is not None:
                        notify()
                    dw = None
                else:
                    dw = ext().conv2d_bwd_weight(x, dy, weight.shape[2],
                                                 weight.shape[3], stride,
                                                 pad, xcol_cached=xcol)
        else:
            x_nchw = _nhwc_to_nchw(x).float()
            dy_nchw = _nhwc_to_nchw(dy).float()
            w = weight.float()
            if need_dx:
                dx = torch.nn.grad.conv2d_input(x_nchw.shape, w, dy_nchw,
                                                stride=stride, padding=pad)
                dx = _nchw_to_nhwc(dx).to(x.dtype)
            if need_dw:
                dw = torch.nn.grad.conv2d_weight(x_nchw, w.shape, dy_nchw,
                                                 stride=stride, padding=pad)
        return dx, dw, None, None, None, None, None, None, None


def conv2d_nhwc(x, weight, wt, w2, stride: int = 1, pad: int = 0,
                want_bn_stats: bool = False, dx_sink=None, gslot=None):
    if want_bn_stats or dx_sink is not None or gslot is not None:
        return Conv2dNHWC.apply(x, weight, wt, w2, stride, pad,
                                want_bn_stats, dx_sink, gslot)
    return Conv2dNHWC.apply(x, weight, wt, w2, stride, pad)


# ---------------------------------------------------------------------------
# BatchNorm (+ fused ReLU)
# ---------------------------------------------------------------------------

def _bn_cpu_stats(x):
    x32 = x.float()
    n = x.shape[0] * x.shape[1] * x.shape[2]
    mean = x32.sum(dim=(0, 1, 2)) / n
    var = x32.pow(2).sum(dim=(0, 1, 2)) / n - mean * mean
    return mean, var.clamp_(min=0.0)


class BatchNorm2dNHWCFn(torch.autograd.Function):
    """Training-mode BN over [N,H,W,C] with optional fused ReLU.

    Decomposed exactly as the kernels are (stats -> apply; bwd_reduce ->
    bwd_elemt) so SyncBatchNorm reuses the same kernels with an all-gather
    between stats and apply.
    """

    @staticmethod
    def forward(ctx, x, weight, bias, mean, invstd, fuse_relu: bool,
                comm=None, count_override=None, residual=None,
                dres_sink=None, dy2_source=None, gslot_w=None,
                gslot_b=None):
        # mean/invstd are per-channel fp32, computed by the module (local
        # stats for BN, cross-rank combined for SyncBN). For SyncBN, `comm`
        # all-reduces the backward sums and `count_override` is the GLOBAL
        # element count (all_reduce of [sum_dy, sum_dy_xmu]).
        # `residual` fuses the ResNet join: y = relu(bn(x) + residual) in
        # one pass (requires fuse_relu).
        mask = None
        if x.is_cuda:
            import os as _os
            if fuse_relu and residual is not None \
                    and _os.environ.get("PDTA_BN_BITMASK"):
                # 1-bit join relu mask: NEGATIVE result at B=256 (-2%
                # step: the byte loads are 8x redundant across lanes and
                # the emit costs more than the saved y pass) — kept
                # opt-in for re-evaluation at other shapes
                y, mask = ext().bn_apply_v(x, weight, bias, mean, invstd,
                                           fuse_relu, residual, True)
            else:
                y = ext().bn_apply(x, weight, bias, mean, invstd,
                                   fuse_relu, residual)
        else:
            x32 = x.float()
            xhat = (x32 - mean) * invstd
            y = xhat * weight.float() + bias.float()
            if residual is not None:
                y = y + residual.float()
            if fuse_relu:
                y = y.clamp_(min=0.0)
            y = y.to(x.dtype)
        ctx.fuse_relu = fuse_relu
        ctx.comm = comm
        ctx.count_override = count_override
        ctx.has_residual = residual is not None
        ctx.dres_sink = dres_sink
        ctx.dy2_source = dy2_source
        ctx.gslot_w, ctx.gslot_b = gslot_w, gslot_b
        ctx.save_for_backward(x, weight, bias, mean, invstd, y, mask)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, bias, mean, invstd, y, mask = ctx.saved_tensors
        fuse_relu = ctx.fuse_relu
        dy = dy.contiguous()
        dy2 = None
        if ctx.dy2_source is not None and ctx.dy2_source.t is not None:
            # second incoming gradient (next block's residual path),
            # summed in-kernel (two-dy backward)
            dy2 = ctx.dy2_source.t
            ctx.dy2_source.t = None
        n = x.shape[0] * x.shape[1] * x.shape[2]
        if ctx.count_override is not None:
            n = ctx.count_override
        dres = None
        if x.is_cuda:
            # ReLU mask: residual joins must read the stored relu output
            # (pre-activation includes the identity branch); plain conv->
            # bn->relu recomputes the mask from xhat in-kernel and never
            # re-reads y (one whole HBM pass saved per kernel)
            relu_y = y if (fuse_relu and ctx.has_residual) else None
            premask = fuse_relu and not ctx.has_residual
            # direct-grad: dbias == sum_dy — have the finalize kernel
            # write it straight into the flat slice (no D2D copy); with a
            # comm the slice still holds the LOCAL sums (correct: DDP
            # averages param grads) and the packed allreduce reads it
            b_view = ctx.gslot_b[0] if ctx.gslot_b is not None else None
            red = ext().bn_bwd_reduce(
                x, dy, relu_y if mask is None else None, mean,
                invstd=invstd if premask else None,
                weight=weight if premask else None,
                bias=bias if premask else None, dy2=dy2, mbits=mask,
                sum_dy_out=b_view)
            sum_dy, sum_dy_xmu = red[0], red[1]
            dy_comb = None
            if dy2 is not None:
                # two-dy: the reduce pass wrote the masked combined dy;
                # the elemt pass now runs mask-free on that one tensor,
                # and it IS the residual-path gradient (dres)
                dy_comb = red[2]
            # param grads from LOCAL sums (DDP averages them afterwards,
            # matching torch SyncBN's split of responsibilities)
            if ctx.gslot_w is not None and x.is_cuda:
                torch.mul(sum_dy_xmu, invstd, out=ctx.gslot_w[0])
                if ctx.gslot_w[1] is not None:
                    ctx.gslot_w[1]()
                dweight = None
            else:
                dweight = sum_dy_xmu * invstd
            if ctx.gslot_b is not None and x.is_cuda:
                # sum_dy IS the flat slice (written by the reduce)
                if ctx.gslot_b[1] is not None:
                    ctx.gslot_b[1]()
                dbias = None
            else:
                dbias = sum_dy.clone() if ctx.comm is not None else sum_dy
            if ctx.comm is not None:
                # [sum_dy ‖ sum_dy_xmu ‖ local_count] in ONE all-reduce;
                # normalizing by the reduced count on-device keeps the
                # path host-sync-free AND exact for ragged rank batches
                # (fused pack/div kernels: no cat/full/slice launches)
                c = sum_dy.numel()
                local_n = float(x.shape[0] * x.shape[1] * x.shape[2])
                packed = ext().syncbn_pack2(sum_dy, sum_dy_xmu, local_n)
                ctx.comm.allreduce_sum(packed).sync_stream()
                ext().syncbn_div(packed, c)
                sum_dy, sum_dy_xmu = packed[:c], packed[c:2 * c]
                n = 1.0
            if dy_comb is not None:
                out = ext().bn_bwd_elemt(x, dy_comb, None, mean, invstd,
                                         weight, sum_dy.contiguous(),
                                         sum_dy_xmu.contiguous(), float(n),
                                         False)
                dx = out[0]
                dres = dy_comb if ctx.has_residual else None
            else:
                out = ext().bn_bwd_elemt(x, dy, relu_y, mean, invstd,
                                         weight, sum_dy.contiguous(),
                                         sum_dy_xmu.contiguous(), float(n),
                                         ctx.has_residual,
                                         bias=bias if premask else None)
                dx = out[0]
                if ctx.has_residual:
                    dres = out[1]
            if dres is not None and ctx.dres_sink is not None:
                # residual-path producer (GradHolder protocol): the
                # consumer BN sums this as its second dy; autograd
                # sees None
                ctx.dres_sink.t = dres
                dres = None
        else:
            dy32 = dy.float()
            if dy2 is not None:
                dy32 = dy32 + dy2.float()
            if fuse_relu:
                dy32 = dy32 * (y.float() > 0)
            x32 = x.float()
            xmu = x32 - mean
            sum_dy = dy32.sum(dim=(0, 1, 2))
            sum_dy_xmu = (dy32 * xmu).sum(dim=(0, 1, 2))
            dweight = sum_dy_xmu * invstd
            dbias = sum_dy.clone() if ctx.comm is not None else sum_dy
            if ctx.comm is not None:
                c = sum_dy.numel()
                local_n = float(x.shape[0] * x.shape[1] * x.shape[2])
                packed = torch.cat([sum_dy, sum_dy_xmu,
                                    torch.full((1,), local_n,
                                               dtype=torch.float32)])
                ctx.comm.allreduce_sum(packed).wait()
                packed[:2 * c] /= packed[2 * c]
                sum_dy, sum_dy_xmu = packed[:c], packed[c:2 * c]
                n = 1.0
            w32 = weight.float()
            if ctx.has_residual:
                dres = dy32.to(x.dtype)
            dx = (dy32 - sum_dy / n - xmu * invstd * invstd * sum_dy_xmu / n)
            dx = (dx * w32 * invstd).to(x.dtype)
        return (dx, dweight, dbias, None, None, None, None, None,
                dres, None, None, None, None)


def bn_stats(x):
    """Per-channel (mean, var) over N*H*W in fp32. GPU: reduction kernel,
    or finalize-only when the producing conv pre-reduced the partials."""
    if x.is_cuda:
        part = getattr(x, "_pdta_bn_part", None)
        if part is not None:
            rows = x.shape[0] * x.shape[1] * x.shape[2]
            return ext().bn_stats_from_part(part, rows, x.shape[3])
        return ext().bn_stats(x)
    return _bn_cpu_stats(x)


def bn_stats_train(x, running_mean, running_var, momentum, eps):
    """(mean, invstd) + in-kernel running-stat update — GPU training fast
    path (one kernel pair instead of stats + 4 python-side buffer ops)."""
    return ext().bn_stats_train(x, running_mean, running_var, momentum, eps)


def bn_stats_train_from_part(part, rows, running_mean, running_var,
                             momentum, eps):
    """(mean, invstd) finalized from a conv2d_nhwc(want_bn_stats=True)
    partial slab — the stage-1 pass over the activation never runs."""
    return ext().bn_stats_train_from_part(part, rows, running_mean,
                                          running_var, momentum, eps)


def bn_train(x, weight, bias, mean, invstd, fuse_relu=False, comm=None,
             count=None, residual=None, dres_sink=None, dy2_source=None,
             gslot_w=None, gslot_b=None):
    return BatchNorm2dNHWCFn.apply(x, weight, bias, mean, invstd, fuse_relu,
                                   comm, count, residual, dres_sink,
                                   dy2_source, gslot_w, gslot_b)


def bn_infer(x, weight, bias, running_mean, running_var, eps, fuse_relu=False,
             residual=None):
    invstd = (running_var.float() + eps).rsqrt()
    if x.is_cuda:
        return ext().bn_apply(x, weight, bias, running_mean.float(), invstd,
                              fuse_relu, residual)
    x32 = x.float()
    y = (x32 - running_mean) * invstd * weight.float() + bias.float()
    if residual is not None:
        y = y + residual.float()
    if fuse_relu:
        y = y.clamp_(min=0.0)
    return y.to(x.dtype)


# ---------------------------------------------------------------------------
# Elementwise: relu, residual add+relu
# ---------------------------------------------------------------------------

class ReLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        y = ext().relu_fwd(x) if x.is_cuda else x.clamp(min=0)
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        if y.is_cuda:
            return ext().relu_bwd_mask(dy.contiguous(), y)
        return dy * (y > 0)


class AddReLU(torch.autograd.Function):
    """y = relu(a + b) — the residual join, one kernel instead of two."""

    @staticmethod
    def forward(ctx, a, b):
        if a.is_cuda:
            y = ext().add_relu_fwd(a, b)
        else:
            y = (a + b).clamp_(min=0)
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        if y.is_cuda:
            d = ext().relu_bwd_mask(dy.contiguous(), y)
        else:
            d = dy * (y > 0)
        return d, d


def relu(x):
    return ReLU.apply(x)


def add_relu(a, b):
    return AddReLU.apply(a, b)


# ---------------------------------------------------------------------------
# Pooling
# ---------------------------------------------------------------------------

class MaxPool2dNHWC(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, kernel: int, stride: int, pad: int):
        ctx.kernel, ctx.stride, ctx.pad = kernel, stride, pad
        ctx.in_shape = x.shape
        if x.is_cuda:
            y, idx = ext().maxpool_fwd(x, kernel, stride, pad)
            ctx.save_for_backward(idx)
            ctx.gpu = True
            return y
        x_nchw = _nhwc_to_nchw(x).float()
        y, idx = F.max_pool2d(x_nchw, kernel, stride, pad, return_indices=True)
        ctx.save_for_backward(idx)
        ctx.gpu = False
        return _nchw_to_nhwc(y).to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        n, h, w, c = ctx.in_shape
        dy = dy.contiguous()
        if ctx.gpu:
            dx = ext().maxpool_bwd(dy, idx, h, w, ctx.kernel, ctx.stride,
                                   ctx.pad)
            return dx, None, None, None
        dy_nchw = _nhwc_to_nchw(dy).float()
        # scatter_ADD (not max_unpool, which overwrites duplicate winners of
        # overlapping windows) to match true max_pool backward semantics
        nn_, cc = dy_nchw.shape[0], dy_nchw.shape[1]
        dx = torch.zeros(nn_, cc, h * w, dtype=dy_nchw.dtype)
        dx.scatter_add_(2, idx.view(nn_, cc, -1), dy_nchw.view(nn_, cc, -1))
        dx = dx.view(nn_, cc, h, w)
        return _nchw_to_nhwc(dx).to(dy.dtype), None, None, None


class GlobalAvgPoolNHWC(torch.autograd.Function):
    """[N,H,W,C] -> [N,C] mean over H,W."""

    @staticmethod
    def forward(ctx, x):
        ctx.in_shape = x.shape
        if x.is_cuda:
            return ext().avgpool_global_fwd(x)
        return x.float().mean(dim=(1, 2)).to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        n, h, w, c = ctx.in_shape
        dy = dy.contiguous()
        if dy.is_cuda:
            return ext().avgpool_global_bwd(dy, h, w)
        dx = (dy.float() / (h * w)).view(n, 1, 1, c).expand(n, h, w, c)
        return dx.to(dy.dtype).contiguous()


def maxpool2d(x, kernel=3, stride=2, pad=1):
    return MaxPool2dNHWC.apply(x, kernel, stride, pad)


def global_avgpool(x):
    return GlobalAvgPoolNHWC.apply(x)


# ---------------------------------------------------------------------------
# Linear (MFMA GEMM)
# ---------------------------------------------------------------------------

def _pad_cols64(t):
    """Zero-pad the last dim to a multiple of 64 (GEMM K-dim contract)."""
    k = t.shape[-1]
    r = (-k) % 64
    return t if r == 0 else F.pad(t, (0, r))

class LinearFn(torch.autograd.Function):
    """y = x @ W^T + b; x [M,K] compute dtype, W master fp32 [N,K]."""

    @staticmethod
    def forward(ctx, x, weight, w_packed, bias, w_packed_t=None,
                gslot_w=None, gslot_b=None):
        ctx.save_for_backward(x, weight, w_packed,
                              w_packed_t if w_packed_t is not None
                              else w_packed)
        ctx.has_t = w_packed_t is not None
        ctx.has_bias = bias is not None
        ctx.gslot_w, ctx.gslot_b = gslot_w, gslot_b
        if x.is_cuda:
            return ext().gemm_nt(x, w_packed, bias)
        y = x.float() @ weight.float().t()
        if bias is not None:
            y = y + bias.float()
        return y.to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        x, weight, w_packed, w_packed_t = ctx.saved_tensors
        dy = dy.contiguous()
        dx = dw = db = None
        if x.is_cuda:
            # every variant is the nt kernel over cheap repacks; K-dims are
            # zero-padded to multiples of 64 (zero rows/cols add nothing)
            if ctx.needs_input_grad[0]:
                dyp = _pad_cols64(dy)
                wtp = _pad_cols64(w_packed_t if ctx.has_t
                                  else w_packed.t().contiguous())
                dx = ext().gemm_nt(dyp, wtp)                  # [M, K_in]
            if ctx.needs_input_grad[1]:
                o_dim, i_dim = weight.shape
                rows = x.shape[0]
                w_out = ctx.gslot_w[0] if ctx.gslot_w is not None else None
                if o_dim % 64 == 0 and i_dim % 64 == 0:
                    # dW = dy^T @ x is exactly the conv-wrw TN GEMM: reuse
                    # its split-K slab kernel (a plain-nt path serialized
                    # K=rows in a handful of blocks — 358us per MLP grad)
                    dw = ext().conv2d_bwd_weight(
                        x.view(1, rows, 1, i_dim), dy.view(1, rows, 1, o_dim),
                        1, 1, 1, 0,
                        out=(w_out.view(o_dim, i_dim, 1, 1)
                             if w_out is not None else None)
                    ).view(o_dim, i_dim)
                else:
                    dyt = _pad_cols64(dy.t().contiguous())    # [O, M]
                    xt = _pad_cols64(x.t().contiguous())      # [K_in, M]
                    dw = ext().gemm_nt(dyt, xt, None, True)   # fp32 [O, K_in]
                    if w_out is not None:
                        w_out.copy_(dw)
                if w_out is not None:
                    if ctx.gslot_w[1] is not None:
                        ctx.gslot_w[1]()
                    dw = None
            if ctx.has_bias and ctx.needs_input_grad[3]:
                db = ext().colsum_f32(
                    dy, out=(ctx.gslot_b[0] if ctx.gslot_b is not None
                             else None))                      # [M,N] -> [N]
                if ctx.gslot_b is not None:
                    if ctx.gslot_b[1] is not None:
                        ctx.gslot_b[1]()
                    db = None
        else:
            dy32, x32, w32 = dy.float(), x.float(), weight.float()
            if ctx.needs_input_grad[0]:
                dx = (dy32 @ w32).to(x.dtype)
            if ctx.needs_input_grad[1]:
                dw = dy32.t() @ x32
            if ctx.has_bias and ctx.needs_input_grad[3]:
                db = dy32.sum(dim=0)
        return dx, dw, None, db, None, None, None


def linear(x, weight, w_packed, bias=None, w_packed_t=None, gslot_w=None,
           gslot_b=None):
    return LinearFn.apply(x, weight, w_packed, bias, w_packed_t, gslot_w,
                          gslot_b)


# ---------------------------------------------------------------------------
# Fused softmax cross-entropy
# ---------------------------------------------------------------------------

class SoftmaxCrossEntropy(torch.autograd.Function):
    """Mean CE over the batch; fused log-softmax + NLL fwd and
    dlogits = (softmax - onehot)/B bwd (SURVEY.md §2d softmax-CE row)."""

    @staticmethod
    def forward(ctx, logits, labels, label_smoothing: float):
        ctx.label_smoothing = label_smoothing
        if logits.is_cuda:
            loss_per, lse = ext().softmax_ce_fwd(logits, labels,
                                                 label_smoothing)
            ctx.save_for_backward(logits, labels, lse)
            return loss_per.mean()
        l32 = logits.float()
        lse = torch.logsumexp(l32, dim=1)
        nll = lse - l32.gather(1, labels.view(-1, 1)).squeeze(1)
        if label_smoothing > 0.0:
            eps = label_smoothing
            smooth = lse - l32.mean(dim=1)
            nll = (1 - eps) * nll + eps * smooth
        ctx.save_for_backward(logits, labels, lse)
        return nll.mean()

    @staticmethod
    def backward(ctx, grad):
        logits, labels, lse = ctx.saved_tensors
        eps = ctx.label_smoothing
        b = logits.shape[0]
        # grad is a device scalar tensor; never .item() it (host sync in the
        # hot loop) — the kernel multiplies by it on device.
        if logits.is_cuda:
            dlogits = ext().softmax_ce_bwd(logits, labels, lse, eps, 1.0 / b,
                                           grad.to(torch.float32))
        else:
            l32 = logits.float()
            p = torch.exp(l32 - lse.unsqueeze(1))
            n = l32.shape[1]
            tgt = F.one_hot(labels, n).float()
            if eps > 0.0:
                tgt = (1 - eps) * tgt + eps / n
            dlogits = ((p - tgt) * (grad.float() / b)).to(logits.dtype)
        return dlogits, None, None


def softmax_cross_entropy(logits, labels, label_smoothing: float = 0.0):
    return SoftmaxCrossEntropy.apply(logits, labels, label_smoothing)


# ---------------------------------------------------------------------------
# Transformer ops (ViT path: LayerNorm, GELU, attention)
# ---------------------------------------------------------------------------

class LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps: float, gslot_w=None,
                gslot_b=None):
        ctx.gslot_w, ctx.gslot_b = gslot_w, gslot_b
        if x.is_cuda:
            y, mean, invstd = ext().ln_fwd(x, weight, bias, eps)
        else:
            x32 = x.float()
            mean = x32.mean(dim=-1)
            var = x32.var(dim=-1, unbiased=False)
            invstd = (var + eps).rsqrt()
            y = ((x32 - mean.unsqueeze(-1)) * invstd.unsqueeze(-1)
                 * weight.float() + bias.float()).to(x.dtype)
        ctx.save_for_backward(x, weight, mean, invstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, invstd = ctx.saved_tensors
        dy = dy.contiguous()
        if x.is_cuda:
            dx, dw, db = ext().ln_bwd(
                x, dy, weight, mean, invstd,
                dw_out=ctx.gslot_w[0] if ctx.gslot_w is not None else None,
                db_out=ctx.gslot_b[0] if ctx.gslot_b is not None else None)
            if ctx.gslot_w is not None:
                if ctx.gslot_w[1] is not None:
                    ctx.gslot_w[1]()
                dw = None
            if ctx.gslot_b is not None:
                if ctx.gslot_b[1] is not None:
                    ctx.gslot_b[1]()
                db = None
        else:
            d = x.shape[-1]
            x32, g32, w32 = x.float(), dy.float(), weight.float()
            xhat = (x32 - mean.unsqueeze(-1)) * invstd.unsqueeze(-1)
            gw = g32 * w32
            s1 = gw.mean(dim=-1, keepdim=True)
            s2 = (gw * xhat).mean(dim=-1, keepdim=True)
            dx = (invstd.unsqueeze(-1) * (gw - s1 - xhat * s2)).to(x.dtype)
            dims = tuple(range(x.dim() - 1))
            dw = (g32 * xhat).sum(dim=dims)
            db = g32.sum(dim=dims)
        return dx, dw, db, None, None, None


def layer_norm(x, weight, bias, eps: float = 1e-6, gslot_w=None,
               gslot_b=None):
    return LayerNormFn.apply(x, weight, bias, eps, gslot_w, gslot_b)



class AddLayerNormFn(torch.autograd.Function):
    """Fused residual join + LayerNorm: (y, s) = (ln(a+b), a+b).

    The ViT block threads (residual, branch) PAIRS between sublayers so
    the sum has exactly ONE consumer — no autograd fork add exists in
    either direction. Backward folds the pass-through gradient ds into
    the dx store (DXADD kernel variant) and returns the SAME combined
    gradient for both inputs."""

    @staticmethod
    def forward(ctx, a, b, weight, bias, eps, gslot_w=None, gslot_b=None):
        ctx.set_materialize_grads(False)
        ctx.gslot_w, ctx.gslot_b = gslot_w, gslot_b
        if a.is_cuda:
            y, mean, invstd, s = ext().ln_fwd(a, weight, bias, eps,
                                              res=b.contiguous())
        else:
            s = (a.float() + b.float()).to(a.dtype)
            s32 = s.float()
            mean = s32.mean(dim=-1)
            var = s32.var(dim=-1, unbiased=False)
            invstd = (var + eps).rsqrt()
            y = ((s32 - mean.unsqueeze(-1)) * invstd.unsqueeze(-1)
                 * weight.float() + bias.float()).to(a.dtype)
        ctx.save_for_backward(s, weight, mean, invstd)
        return y, s

    @staticmethod
    def backward(ctx, dy, ds):
        s, weight, mean, invstd = ctx.saved_tensors
        dy = dy.contiguous()
        if s.is_cuda:
            dx, dw, db = ext().ln_bwd(
                s, dy, weight, mean, invstd,
                dxadd=ds.contiguous() if ds is not None else None,
                dw_out=ctx.gslot_w[0] if ctx.gslot_w is not None else None,
                db_out=ctx.gslot_b[0] if ctx.gslot_b is not None else None)
            if ctx.gslot_w is not None:
                if ctx.gslot_w[1] is not None:
                    ctx.gslot_w[1]()
                dw = None
            if ctx.gslot_b is not None:
                if ctx.gslot_b[1] is not None:
                    ctx.gslot_b[1]()
                db = None
        else:
            d = s.shape[-1]
            s32, g32, w32 = s.float(), dy.float(), weight.float()
            xhat = (s32 - mean.unsqueeze(-1)) * invstd.unsqueeze(-1)
            gw = g32 * w32
            s1 = gw.mean(dim=-1, keepdim=True)
            s2 = (gw * xhat).mean(dim=-1, keepdim=True)
            dx = (invstd.unsqueeze(-1) * (gw - s1 - xhat * s2))
            if ds is not None:
                dx = dx + ds.float()
            dx = dx.to(s.dtype)
            dims = tuple(range(s.dim() - 1))
            dw = (g32 * xhat).sum(dim=dims)
            db = g32.sum(dim=dims)
        return dx, dx, dw, db, None, None, None


def add_layer_norm(a, b, weight, bias, eps: float = 1e-6, gslot_w=None,
                   gslot_b=None):
    return AddLayerNormFn.apply(a, b, weight, bias, eps, gslot_w, gslot_b)


class GELUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.save_for_backward(x)
        if x.is_cuda:
            return ext().gelu_fwd(x)
        return F.gelu(x.float()).to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        dy = dy.contiguous()
        if x.is_cuda:
            return ext().gelu_bwd(x, dy)
        x32 = x.float().detach().requires_grad_()
        with torch.enable_grad():
            y = F.gelu(x32)
        (dx,) = torch.autograd.grad(y, x32, dy.float())
        return dx.to(x.dtype)


def gelu(x):
    return GELUFn.apply(x)


class AttentionFn(torch.autograd.Function):
    """Scaled-dot-product attention over padded sequences.

    q,k,v: [bat, S, hd] (S padded to a multiple of 64; hd % 64 == 0);
    columns >= nvalid are masked out of the softmax, padded v rows are
    zero, so results for rows < nvalid are exact. GPU path: batched MFMA
    GEMMs (QK^T fp32) + fused masked-softmax kernel (SURVEY.md §2d ViT
    attention row).
    """

    @staticmethod
    def forward(ctx, q, k, v, nvalid: int, scale: float):
        if q.is_cuda:
            scores = ext().gemm_nt_batched(q, k, True)        # fp32 [b,S,S]
            p = ext().attn_softmax_fwd(scores, nvalid, scale, q.dtype)
            vt = v.transpose(1, 2).contiguous()               # [b, hd, S]
            out = ext().gemm_nt_batched(p, vt)                # [b, S, hd]
        else:
            s32 = (q.float() @ k.float().transpose(1, 2)) * scale
            mask = torch.arange(q.shape[1]) >= nvalid
            s32 = s32.masked_fill(mask.view(1, 1, -1), float("-inf"))
            p32 = torch.softmax(s32, dim=-1)
            p = p32.to(q.dtype)
            out = (p32 @ v.float()).to(q.dtype)
        ctx.save_for_backward(q, k, v, p)
        ctx.scale = scale
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, p = ctx.saved_tensors
        dout = dout.contiguous()
        scale = ctx.scale
        if q.is_cuda:
            dp = ext().gemm_nt_batched(dout, v)               # [b, S, S]
            ds = ext().attn_softmax_bwd(p, dp, scale).to(q.dtype)
            kt = k.transpose(1, 2).contiguous()               # [b, hd, S]
            dq = ext().gemm_nt_batched(ds, kt)                # [b, S, hd]
            dst = ds.transpose(1, 2).contiguous()
            qt = q.transpose(1, 2).contiguous()
            dk = ext().gemm_nt_batched(dst, qt)
            pt = p.transpose(1, 2).contiguous()
            dot = dout.transpose(1, 2).contiguous()
            dv = ext().gemm_nt_batched(pt, dot)
        else:
            p32, g32 = p.float(), dout.float()
            dp = g32 @ v.float().transpose(1, 2)
            ds = (p32 * (dp - (p32 * dp).sum(-1, keepdim=True)) * scale)
            dq = (ds @ k.float()).to(q.dtype)
            dk = (ds.transpose(1, 2) @ q.float()).to(q.dtype)
            dv = (p32.transpose(1, 2) @ g32).to(q.dtype)
        return dq, dk, dv, None, None


def attention(q, k, v, nvalid: int, scale: float):
    return AttentionFn.apply(q, k, v, nvalid, scale)


class AttentionQKVFn(torch.autograd.Function):
    """Zero-copy multi-head attention over the interleaved qkv projection
    [b*s, 3*h*hd]: q/k/v are consumed IN PLACE via strided-batched MFMA
    GEMMs and all gradients are written IN PLACE into d(qkv) — no
    permute/contiguous copies anywhere (those were 30% of the ViT step).
    GPU-only; the CPU path uses AttentionFn via explicit permutes.
    """

    @staticmethod
    def forward(ctx, qkv2d, b: int, s: int, h: int, hd: int, nvalid: int,
                scale: float):
        qkv3 = qkv2d.view(b, s, 3 * h * hd)
        out, p = ext().attn_fwd(qkv3, b, s, h, hd, nvalid, scale)
        ctx.save_for_backward(qkv2d, p)
        ctx.dims = (b, s, h, hd, nvalid, scale)
        return out  # [b, s, h*hd]

    @staticmethod
    def backward(ctx, dout):
        qkv2d, p = ctx.saved_tensors
        b, s, h, hd, nvalid, scale = ctx.dims
        dqkv = ext().attn_bwd(qkv2d.view(b, s, 3 * h * hd), p,
                              dout.contiguous(), b, s, h, hd, nvalid,
                              scale)
        return dqkv.view(b * s, 3 * h * hd), None, None, None, None, None, \
            None


def attention_qkv(qkv2d, b, s, h, hd, nvalid, scale):
    return AttentionQKVFn.apply(qkv2d, b, s, h, hd, nvalid, scale)
