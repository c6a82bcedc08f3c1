"""Checkpoint save/load — reference-compatible layout + full train state.

Reference layout: raw model.state_dict() of the
DataParallel wrapper, every key prefixed `module.`, saved as
checkpoints/<name>/<step>_<name>.pth and final_model.pth; weights only.
This framework writes the same `module.`-prefixed weight files (so its
checkpoints interchange with the reference) and ADDITIONALLY a full train
state (model/optimizer/scheduler/step/rng) for real resume — the reference
loses optimizer/scheduler state on restart.
"""

import os
import random

import numpy as np
import torch


def _unwrap(model):
    return model.module if hasattr(model, "module") else model


def reference_state_dict(model):
    """state_dict with the reference's `module.` DataParallel prefix."""
    sd = _unwrap(model).state_dict()
    return {"module." + k: v for k, v in sd.items()}


def load_reference_state_dict(model, state_dict, strict=True):
    """Load a checkpoint that may or may not carry the `module.` prefix."""
    stripped = {k[7:] if k.startswith("module.") else k: v
                for k, v in state_dict.items()}
    return _unwrap(model).load_state_dict(stripped, strict=strict)


def save_weights(model, path):
    os.makedirs(os.path.dirname(path), exist_ok=True)
    torch.save(reference_state_dict(model), path)


def load_weights(model, path, strict=True):
    sd = torch.load(path, map_location="cpu", weights_only=True)
    return load_reference_state_dict(model, sd, strict=strict)


def save_train_state(path, model, optimizer, scheduler, total_steps):
    """Full resumable state (framework extension)."""
    os.makedirs(os.path.dirname(path), exist_ok=True)
    torch.save({
        "model": reference_state_dict(model),
        "optimizer": optimizer.state_dict(),
        "scheduler": scheduler.state_dict(),
        "total_steps": total_steps,
        "rng": {
            "torch": torch.get_rng_state(),
            "cuda": torch.cuda.get_rng_state_all() if torch.cuda.is_available() else None,
            "numpy": np.random.get_state(),
            "python": random.getstate(),
        },
    }, path)


def load_train_state(path, model, optimizer=None, scheduler=None,
                     restore_rng=True):
    state = torch.load(path, map_location="cpu", weights_only=False)
    load_reference_state_dict(model, state["model"])
    if optimizer is not None and "optimizer" in state:
        optimizer.load_state_dict(state["optimizer"])
    if scheduler is not None and "scheduler" in state:
        scheduler.load_state_dict(state["scheduler"])
    if restore_rng and "rng" in state:
        rng = state["rng"]
        torch.set_rng_state(rng["torch"].cpu().to(torch.uint8)
                            if torch.is_tensor(rng["torch"]) else rng["torch"])
        if rng.get("cuda") is not None and torch.cuda.is_available():
            try:
                torch.cuda.set_rng_state_all(rng["cuda"])
            except RuntimeError:
                pass  # different device count — keep fresh cuda rng
        np.random.set_state(rng["numpy"])
        random.setstate(rng["python"])
    return state.get("total_steps", 0)
