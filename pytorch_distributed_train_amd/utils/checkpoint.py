"""Checkpoint save/resume.

Layout matches the reference pattern: rank 0 writes
    {"epoch": int, "step": int, "model": model.state_dict(),
     "optimizer": optimizer.state_dict(), "scaler": ... , "args": vars}
with torch.save; resume torch.load(map_location=device) + load_state_dict,
then the DDP init broadcast re-aligns ranks. Model keys carry NO "module."
prefix (we save the inner module, as `model.module.state_dict()` does in
the reference pattern); load_checkpoint strips one if present so reference
checkpoints that saved the DDP wrapper still load.

Writes are atomic (tmp + rename) so a killed rank 0 never leaves a torn
checkpoint for the restart gang to resume from.
"""
from __future__ import annotations

import os

import torch


def save_checkpoint(path: str, epoch: int, step: int, model, optimizer,
                    scaler=None, extra=None) -> None:
    state = {
        "epoch": epoch,
        "step": step,
        "model": model.state_dict(),
        "optimizer": optimizer.state_dict(),
    }
    if scaler is not None:
        state["scaler"] = scaler.state_dict()
    if extra:
        state.update(extra)
    os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
    tmp = path + ".tmp"
    torch.save(state, tmp)
    os.replace(tmp, path)


def _strip_module_prefix(sd: dict) -> dict:
    if any(k.startswith("module.") for k in sd):
        return {k[len("module."):] if k.startswith("module.") else k: v
                for k, v in sd.items()}
    return sd


def load_checkpoint(path: str, model, optimizer=None, scaler=None,
                    map_location="cpu") -> dict:
    state = torch.load(path, map_location=map_location, weights_only=False)
    model.load_state_dict(_strip_module_prefix(state["model"]))
    if optimizer is not None and "optimizer" in state:
        optimizer.load_state_dict(state["optimizer"])
    if scaler is not None and "scaler" in state:
        scaler.load_state_dict(state["scaler"])
    return state
