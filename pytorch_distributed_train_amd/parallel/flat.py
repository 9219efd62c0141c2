"""Flat master-parameter storage.

All trainable parameters live in ONE contiguous fp32 buffer (and their
gradients in one matching buffer). This is the MI355X-native replacement for
torch's per-tensor params + multi-tensor-apply machinery (SURVEY.md §2d
fused-SGD row): with 288 GB of HBM3E there is no reason to fragment state —
  - the optimizer is a single HBM-bound kernel over one array,
  - DDP buckets are contiguous SLICES of the gradient buffer (zero-copy:
    autograd accumulates straight into the bucket via p.grad views, the
    gradient_as_bucket_view idea taken to its endpoint),
  - the init-time parameter broadcast is one RCCL broadcast.

Parameters are laid out in REVERSE registration order so gradient arrival
during backward (last layer first) fills the buffer front-to-back, which
lets bucket i complete before bucket i+1 and overlap its all-reduce with
the rest of backward (the arrival-order bucketing torch's Reducer rebuilds
dynamically — reducer.hpp:135,148 semantics — achieved statically here).
"""
from __future__ import annotations

import torch

ALIGN = 64  # elements; 256 B alignment for vectorized kernels


def _aligned(n: int) -> int:
    return (n + ALIGN - 1) // ALIGN * ALIGN


class FlatParams:
    def __init__(self, named_params, device=None):
        items = [(n, p) for n, p in named_params if p.requires_grad]
        items.reverse()  # reverse registration order (see module docstring)
        self.names = [n for n, _ in items]
        params = [p for _, p in items]
        if device is None:
            device = params[0].device
        self.device = torch.device(device)

        offset = 0
        self.offsets, self.sizes, self.shapes = [], [], []
        for p in params:
            n = p.numel()
            self.offsets.append(offset)
            self.sizes.append(n)
            self.shapes.append(p.shape)
            offset += _aligned(n)
        self.total = offset

        self.master = torch.zeros(self.total, dtype=torch.float32,
                                  device=self.device)
        self.grad = torch.zeros(self.total, dtype=torch.float32,
                                device=self.device)
        # copy current values in and repoint each Parameter at its view
        self.params = params
        for p, off, n, shape in zip(params, self.offsets, self.sizes,
                                    self.shapes):
            view = self.master[off:off + n].view(shape)
            view.copy_(p.data.to(self.device, torch.float32))
            p.data = view
            # autograd accumulates in place into an existing .grad tensor,
            # so this view IS the bucket slice (no copy on the hot path)
            p.grad = self.grad[off:off + n].view(shape)

        self.num_params = len(params)
        self.index_by_param = {id(p): i for i, p in enumerate(params)}

    def grad_view(self, i: int) -> torch.Tensor:
        off, n = self.offsets[i], self.sizes[i]
        return self.grad[off:off + n].view(self.shapes[i])

    def zero_grad(self) -> None:
        self.grad.zero_()

    def reattach_grads(self) -> None:
        """Re-point p.grad at the flat views (torch may have replaced them
        e.g. after optimizer.zero_grad(set_to_none=True) elsewhere)."""
        for p, off, n, shape in zip(self.params, self.offsets, self.sizes,
                                    self.shapes):
            g = p.grad
            if g is None or g.data_ptr() != self.grad[off:off + n].data_ptr():
                p.grad = self.grad[off:off + n].view(shape)
