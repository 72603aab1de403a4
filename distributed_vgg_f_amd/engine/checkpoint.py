"""Checkpoint/resume with the reference's directory protocol.

Layout compatibility with the TF-Saver idiom the reference uses (R8):
  <dir>/checkpoint                      latest-pointer text file
        model.ckpt-<step>.npz           fp32 master weights + momentum + meta

The `checkpoint` file uses the TF protocol text format:
  model_checkpoint_path: "model.ckpt-120"
  all_model_checkpoint_paths: "model.ckpt-120"

Tensor names inside the npz are canonical TF-slim style for VGG-F
(conv1/weights, conv1/biases, ..., fc8/weights — see NAME_COMPAT) with our
module names as fallback, kept as data so a future re-verification of the
real reference can adjust them without code changes.
"""

from __future__ import annotations

import os
import re
from typing import Dict, Optional

import numpy as np
import torch

from .arena import ArenaOptimizer

# our VGG-F module param names -> TF-slim style checkpoint names
NAME_COMPAT: Dict[str, str] = {}
_CONV_IDX = {0: "conv1", 3: "conv2", 6: "conv3", 7: "conv4", 8: "conv5"}
for i, c in _CONV_IDX.items():
    NAME_COMPAT[f"features.{i}.weight"] = f"{c}/weights"
    NAME_COMPAT[f"features.{i}.bias"] = f"{c}/biases"
for fc in ("fc6", "fc7", "fc8"):
    NAME_COMPAT[f"{fc}.weight"] = f"{fc}/weights"
    NAME_COMPAT[f"{fc}.bias"] = f"{fc}/biases"
_COMPAT_INV = {v: k for k, v in NAME_COMPAT.items()}


def save(ckpt_dir: str, arena: ArenaOptimizer, step: int,
         model: Optional[torch.nn.Module] = None) -> str:
    os.makedirs(ckpt_dir, exist_ok=True)
    arrays = {"__step__": np.array(step, dtype=np.int64),
              "__momentum__": arena.m32.cpu().numpy()}
    if model is not None:
        # non-trainable state the arena doesn't hold: BN running stats
        # (needed for resnet50 eval/resume) and dropout RNG call counters
        # (so resume doesn't replay mask offsets)
        for name, buf in model.named_buffers():
            arrays["__buf__" + name] = buf.detach().cpu().numpy()
        calls = [int(getattr(m, "_calls", 0)) for m in model.modules()
                 if type(m).__name__ == "Dropout"]
        if calls:
            arrays["__dropout_calls__"] = np.array(calls, dtype=np.int64)
    for name, w in arena.named_master():
        key = NAME_COMPAT.get(name, name)
        arrays[key] = w.detach().cpu().numpy()
    base = f"model.ckpt-{step}"
    path = os.path.join(ckpt_dir, base + ".npz")
    np.savez(path, **arrays)
    with open(os.path.join(ckpt_dir, "checkpoint"), "w") as fh:
        fh.write(f'model_checkpoint_path: "{base}"\n')
        fh.write(f'all_model_checkpoint_paths: "{base}"\n')
    return path


def latest(ckpt_dir: str) -> Optional[str]:
    ptr = os.path.join(ckpt_dir, "checkpoint")
    if not os.path.exists(ptr):
        return None
    with open(ptr) as fh:
        m = re.search(r'model_checkpoint_path:\s*"([^"]+)"', fh.read())
    if not m:
        return None
    path = os.path.join(ckpt_dir, m.group(1) + ".npz")
    return path if os.path.exists(path) else None


def load(path: str, arena: ArenaOptimizer,
         model: Optional[torch.nn.Module] = None) -> int:
    data = np.load(path)
    bufs = dict(model.named_buffers()) if model is not None else {}
    for key in data.files:
        if key in ("__step__", "__momentum__", "__dropout_calls__"):
            continue
        if key.startswith("__buf__"):
            buf = bufs.get(key[len("__buf__"):])
            if buf is not None:
                buf.copy_(torch.from_numpy(data[key]).to(buf.device,
                                                         buf.dtype))
            continue
        name = _COMPAT_INV.get(key, key)
        arena.load_master(name, torch.from_numpy(data[key]))
    if model is not None and "__dropout_calls__" in data.files:
        drops = [m for m in model.modules()
                 if type(m).__name__ == "Dropout"]
        for m, c in zip(drops, data["__dropout_calls__"]):
            m._calls = int(c)
    if "__momentum__" in data.files:
        m = torch.from_numpy(data["__momentum__"])
        if m.numel() == arena.m32.numel():
            arena.m32.copy_(m.to(arena.device))
    arena.sync_compute_weights()
    return int(data["__step__"]) if "__step__" in data.files else 0


def restore_latest(ckpt_dir: str, arena: ArenaOptimizer,
                   model: Optional[torch.nn.Module] = None) -> int:
    path = latest(ckpt_dir)
    if path is None:
        return 0
    return load(path, arena, model)
