"""Structured metrics.

Replaces TF's LoggingTensorHook/TensorBoard scalars with JSONL rows per
log window (step, loss, img/s, per-phase ms) plus a human stdout mirror.
"""

from __future__ import annotations

import json
import sys
import time
from typing import Optional


class MetricsLogger:
    def __init__(self, path: Optional[str] = None, rank: int = 0,
                 stdout: bool = True):
        self.rank = rank
        self.stdout = stdout and rank == 0
        self.fh = open(path, "a") if (path and rank == 0) else None

    def log(self, **row) -> None:
        row.setdefault("t", time.time())
        if self.fh:
            self.fh.write(json.dumps(row) + "\n")
            self.fh.flush()
        if self.stdout:
            step = row.get("step", "?")
            parts = [f"step {step}"]
            for k in ("loss", "images_per_sec", "ms_per_step", "lr"):
                if k in row:
                    v = row[k]
                    parts.append(f"{k}={v:.4g}" if isinstance(v, float) else
                                 f"{k}={v}")
            print("  ".join(parts), file=sys.stderr, flush=True)

    def close(self) -> None:
        if self.fh:
            self.fh.close()


class PhaseTimers:
    """hipEvent-based per-phase timers (h2d/fwd/bwd/ar/opt), GPU only.

    Cheap enough to keep on; events are recorded per window, elapsed read
    at the window boundary after a sync. `ar` measures EXPOSED all-reduce:
    the compute-stream stall waiting on the bucket reductions after
    backward ends (RCCL runs on its own streams; with good overlap this
    is ~0). `h2d` is the compute-stream wait on the copy-stream event.
    """

    PHASES = ("h2d", "fwd", "bwd", "ar", "opt")

    def __init__(self, device, enabled: bool = True):
        import torch
        self.enabled = enabled and device.type == "cuda"
        if self.enabled:
            self.ev = {p: (torch.cuda.Event(enable_timing=True),
                           torch.cuda.Event(enable_timing=True))
                       for p in self.PHASES}
        self.ms = {p: 0.0 for p in self.PHASES}

    def start(self, phase):
        if self.enabled:
            self.ev[phase][0].record()

    def stop(self, phase):
        if self.enabled:
            self.ev[phase][1].record()

    def collect(self):
        """Call only after a device sync; returns last-step ms per phase."""
        if not self.enabled:
            return {}
        out = {}
        for p, (a, b) in self.ev.items():
            try:
                out["ms_" + p] = a.elapsed_time(b)
            except Exception:
                pass
        return out
