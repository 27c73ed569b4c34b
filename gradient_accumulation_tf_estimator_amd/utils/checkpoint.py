"""Checkpoint save/restore with exact mid-accumulation-window resume.

The reference gets this from the Estimator's variable saver: params, adam m/v,
the accum_grads buffers AND global_step are all TF variables captured in
model_dir checkpoints. Here a checkpoint is a
torch.save dict of {model state_dict, engine state (flat master/m/v/accum +
micro-step counter), step}, plus a TF-style ``checkpoint`` pointer file
naming the latest one.
"""

from __future__ import annotations

import json
import os
from typing import Dict, List, Optional

import torch

POINTER = "checkpoint"


def _pointer_path(model_dir: str) -> str:
    return os.path.join(model_dir, POINTER)


def save(model_dir: str, step: int, model_state: Dict, engine_state: Dict,
         keep_max: int = 5) -> str:
    os.makedirs(model_dir, exist_ok=True)
    path = os.path.join(model_dir, f"ckpt-{step}.pt")
    # engine_state tensors reference live flat buffers -> clone to CPU
    eng = {
        k: (v.detach().cpu().clone() if torch.is_tensor(v) else v)
        for k, v in engine_state.items()
    }
    mod = {k: v.detach().cpu().clone() for k, v in model_state.items()}
    torch.save({"step": step, "model": mod, "engine": eng}, path)

    ckpts = _all_checkpoints(model_dir)
    if path not in ckpts:
        ckpts.append(path)
    ckpts.sort(key=lambda p: int(p.rsplit("-", 1)[1].split(".")[0]))
    while len(ckpts) > keep_max:
        old = ckpts.pop(0)
        try:
            os.remove(old)
        except OSError:
            pass
    with open(_pointer_path(model_dir), "w") as f:
        json.dump({"latest": path, "all": ckpts}, f)
    return path


def _all_checkpoints(model_dir: str) -> List[str]:
    try:
        with open(_pointer_path(model_dir)) as f:
            d = json.load(f)
        return [p for p in d.get("all", []) if os.path.exists(p)]
    except (OSError, json.JSONDecodeError):
        return []


def latest(model_dir: str) -> Optional[str]:
    try:
        with open(_pointer_path(model_dir)) as f:
            d = json.load(f)
        p = d.get("latest")
        return p if p and os.path.exists(p) else None
    except (OSError, json.JSONDecodeError):
        return None


def load(path: str, map_location="cpu") -> Dict:
    return torch.load(path, map_location=map_location, weights_only=False)


def export_safetensors(path: str, model_state: Dict) -> str:
    """Export model weights (only) as a .safetensors file -- the
    interchange format for serving stacks. Engine state (flat accum/m/v,
    micro-step) stays in the .pt checkpoints: safetensors holds tensors
    only, and serving needs none of it."""
    from safetensors.torch import save_file

    flat = {k: v.detach().cpu().contiguous() for k, v in model_state.items()
            if torch.is_tensor(v)}
    save_file(flat, path)
    return path


def load_safetensors(path: str) -> Dict:
    from safetensors.torch import load_file

    return load_file(path)
