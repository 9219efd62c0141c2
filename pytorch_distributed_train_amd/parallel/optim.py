"""Fused SGD+momentum over the flat parameter buffer.

Replaces torch's _fused_sgd / multi-tensor-apply path
with ONE kernel over ONE buffer: p <- p - lr * (mu*buf + g + wd*p), torch
SGD semantics (optim/sgd.py behavior, re-implemented):
    d_p = grad + wd * p
    buf = mu * buf + (1 - dampening) * d_p
    d_p = buf                      (or d_p + mu*buf if nesterov)
    p  -= lr * d_p

state_dict()/load_state_dict() expose torch.optim.SGD-compatible structure
(param_groups + per-param momentum_buffer) for checkpoint layout parity.
"""
from __future__ import annotations

import torch

from .flat import FlatParams
from ..ops.backend import ext


class FusedSGD:
    def __init__(self, flat: FlatParams, lr: float, momentum: float = 0.0,
                 weight_decay: float = 0.0, dampening: float = 0.0,
                 nesterov: bool = False):
        if nesterov and (momentum <= 0 or dampening != 0):
            raise ValueError("nesterov needs momentum > 0, dampening = 0")
        self.flat = flat
        # on GPU lr lives in a device scalar the kernel dereferences, so a
        # hipGraph-captured step sees schedule updates (lr setter below)
        self._lr_dev = (torch.zeros(1, dtype=torch.float32,
                                    device=flat.master.device)
                        if flat.master.is_cuda else None)
        self.lr = lr
        self.momentum = momentum
        self.weight_decay = weight_decay
        self.dampening = dampening
        self.nesterov = nesterov
        self.momentum_buf = None
        if momentum != 0.0:
            self.momentum_buf = torch.zeros_like(flat.master)
        self._step_count = 0

    @property
    def lr(self) -> float:
        return self._lr

    @lr.setter
    def lr(self, value: float) -> None:
        self._lr = float(value)
        if self._lr_dev is not None:
            self._lr_dev.fill_(self._lr)

    @torch.no_grad()
    def step(self, grad_scale: float = 1.0) -> None:
        p, g = self.flat.master, self.flat.grad
        # torch.optim.SGD initializes the momentum buffer to a CLONE of d_p
        # on its first use (no (1-dampening) factor); with zero-initialized
        # flat storage that is equivalent to dampening=0 on the first step
        damp = self.dampening if self._step_count > 0 else 0.0
        if p.is_cuda:
            mom = self.momentum_buf if self.momentum_buf is not None \
                else torch.empty(0, device=p.device)
            ext().fused_sgd(p, g, mom, self.lr, self.momentum,
                            self.weight_decay, damp,
                            self.nesterov, grad_scale,
                            lr_dev=self._lr_dev)
        else:
            d_p = g if grad_scale == 1.0 else g * grad_scale
            if self.weight_decay != 0.0:
                d_p = d_p.add(p, alpha=self.weight_decay)
            if self.momentum != 0.0:
                buf = self.momentum_buf
                buf.mul_(self.momentum).add_(d_p, alpha=1 - damp)
                d_p = d_p.add(buf, alpha=self.momentum) if self.nesterov else buf
            p.add_(d_p, alpha=-self.lr)
        self._step_count += 1

    def zero_grad(self, set_to_none: bool = False) -> None:
        # set_to_none is ignored: flat gradient storage is zeroed in place
        self.flat.zero_grad()

    # -- torch.optim.SGD-compatible checkpoint structure ------------------
    def state_dict(self) -> dict:
        state = {}
        if self.momentum_buf is not None:
            for i in range(self.flat.num_params):
                off, n = self.flat.offsets[i], self.flat.sizes[i]
                state[i] = {"momentum_buffer":
                            self.momentum_buf[off:off + n]
                            .view(self.flat.shapes[i]).clone()}
        return {
            "state": state,
            "param_groups": [{
                "lr": self.lr, "momentum": self.momentum,
                "dampening": self.dampening,
                "weight_decay": self.weight_decay,
                "nesterov": self.nesterov,
                "params": list(range(self.flat.num_params)),
            }],
        }

    def load_state_dict(self, sd: dict) -> None:
        g = sd["param_groups"][0]
        self.lr = g["lr"]
        self.momentum = g["momentum"]
        self.dampening = g.get("dampening", 0.0)
        self.weight_decay = g.get("weight_decay", 0.0)
        self.nesterov = g.get("nesterov", False)
        if self.momentum != 0.0 and self.momentum_buf is None:
            self.momentum_buf = torch.zeros_like(self.flat.master)
        for i, s in sd["state"].items():
            i = int(i)
            off, n = self.flat.offsets[i], self.flat.sizes[i]
            buf = s.get("momentum_buffer")
            if buf is not None:
                self.momentum_buf[off:off + n].copy_(
                    buf.reshape(-1).to(self.momentum_buf.device,
                                       torch.float32))
                # restored buffers are past their first use: the next step
                # must apply the real dampening factor
                self._step_count = max(self._step_count, 1)
