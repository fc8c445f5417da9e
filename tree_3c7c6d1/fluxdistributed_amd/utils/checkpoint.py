"""Checkpoint / resume.

The reference saves BSON with a single `model` key holding the
CPU-materialized model, every N cycles (/root/reference/src/sync.jl:156-161)
and never serializes optimizer state. Here the layout is
the torch-native equivalent of "model tree + optimizer state tree", and —
improving on the reference — optimizer state IS saved, so resume is exact.
"""

import os
from typing import Optional

import torch


def save_checkpoint(path: str, model: torch.nn.Module,
                    optimizer: Optional[torch.optim.Optimizer] = None,
                    step: int = 0, extra: Optional[dict] = None):
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    state = {
        "model": {k: v.detach().cpu() for k, v in model.state_dict().items()},
        "step": step,
    }
    if optimizer is not None:
        state["optimizer"] = _to_cpu(optimizer.state_dict())
    if extra:
        state["extra"] = extra
    tmp = path + ".tmp"
    torch.save(state, tmp)
    os.replace(tmp, path)  # atomic: a crash never leaves a torn checkpoint
    return path


def load_checkpoint(path: str, model: torch.nn.Module,
                    optimizer: Optional[torch.optim.Optimizer] = None,
                    map_location="cpu"):
    state = torch.load(path, map_location=map_location, weights_only=False)
    model.load_state_dict(state["model"])
    if optimizer is not None:
        if "optimizer" in state:
            optimizer.load_state_dict(state["optimizer"])
        elif hasattr(optimizer, "refresh_master"):
            optimizer.refresh_master()
    return state.get("step", 0), state.get("extra", {})


def _to_cpu(obj):
    if isinstance(obj, torch.Tensor):
        return obj.detach().cpu()
    if isinstance(obj, dict):
        return {k: _to_cpu(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return type(obj)(_to_cpu(v) for v in obj)
    return obj
