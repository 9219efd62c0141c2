"""Loss scaling for the fp16 path (GradScaler semantics, SURVEY.md §2b
row 17 — amp/grad_scaler.py behavior re-implemented over flat gradients).

bf16 needs none of this (the default config); fp16 (ResNet-152 config)
scales the loss, unscales the flat gradient buffer while checking for
inf/nan in ONE kernel pass, all-reduces the found_inf flag so every rank
takes the same skip decision, and adapts the scale (growth_factor 2 every
growth_interval clean steps, backoff 0.5 on overflow).
"""
from __future__ import annotations

import torch

from ..ops.backend import ext


class GradScaler:
    def __init__(self, init_scale: float = 2.0 ** 16, growth_factor: float = 2.0,
                 backoff_factor: float = 0.5, growth_interval: int = 2000,
                 enabled: bool = True):
        self.enabled = enabled
        self._scale = init_scale
        self.growth_factor = growth_factor
        self.backoff_factor = backoff_factor
        self.growth_interval = growth_interval
        self._growth_tracker = 0

    def scale(self, loss):
        if not self.enabled:
            return loss
        return loss * self._scale

    @torch.no_grad()
    def unscale_and_check(self, grad_flat, comm=None) -> bool:
        """Unscale grads in place; returns True if step must be SKIPPED.

        The found_inf decision is all-reduced (SUM) so every rank skips
        together (SURVEY.md §2d GradScaler found-inf row).
        """
        if not self.enabled:
            return False
        inv = 1.0 / self._scale
        if grad_flat.is_cuda:
            found = ext().unscale_check(grad_flat, inv)
        else:
            grad_flat.mul_(inv)
            found = (~torch.isfinite(grad_flat)).any().float().reshape(1)
        if comm is not None and comm.world_size > 1:
            comm.allreduce_sum(found).sync_stream()
        skip = bool(found.item() > 0)
        if skip:
            self._scale *= self.backoff_factor
            self._growth_tracker = 0
        else:
            self._growth_tracker += 1
            if self._growth_tracker >= self.growth_interval:
                self._scale *= self.growth_factor
                self._growth_tracker = 0
        return skip

    def get_scale(self) -> float:
        return self._scale

    def state_dict(self) -> dict:
        return {"scale": self._scale, "growth_tracker": self._growth_tracker,
                "enabled": self.enabled}

    def load_state_dict(self, sd: dict) -> None:
        self._scale = sd["scale"]
        self._growth_tracker = sd.get("growth_tracker", 0)
        self.enabled = sd.get("enabled", self.enabled)
