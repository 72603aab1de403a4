#!/usr/bin/env python3
"""Attribute per-step Fill / copyBuffer launches to their host-side call
sites: run a few trainer steps under torch.profiler with python stacks
and print every CUDA kernel whose name matches Fill/Copy, with the
deepest repo frame that launched it. Usage:

    python tools/step_trace_probe.py [--model vggf] [--steps 3]
"""

import argparse
import sys

import torch

sys.path.insert(0, ".")
from distributed_vgg_f_amd.config import Config              # noqa: E402
from distributed_vgg_f_amd.engine.trainer import Trainer     # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="vggf")
    ap.add_argument("--steps", type=int, default=3)
    args = ap.parse_args()
    cfg = Config(model=args.model, train_steps=args.steps, log_every=0)
    tr = Trainer(cfg)
    for _ in range(3):  # warmup: caches populated, allocator settled
        tr.train_step()
    torch.cuda.synchronize()
    acts = [torch.profiler.ProfilerActivity.CPU,
            torch.profiler.ProfilerActivity.CUDA]
    with torch.profiler.profile(activities=acts, with_stack=True,
                                with_modules=True) as prof:
        for _ in range(args.steps):
            tr.train_step()
        torch.cuda.synchronize()
    evs = prof.key_averages(group_by_stack_n=20)
    rows = []
    for e in evs:
        name = e.key
        if not ("fill" in name.lower() or "copy" in name.lower() or
                "Memcpy" in name or "zero" in name.lower()):
            continue
        rows.append((e.count, name[:70], list(e.stack or [])[:6]))
    rows.sort(reverse=True)
    print(f"== fill/copy events over {args.steps} steps "
          f"(count, kernel, stack)")
    for cnt, name, frames in rows[:50]:
        print(f"{cnt:5d}x {name}")
        for f in frames:
            print(f"         {f}")


if __name__ == "__main__":
    main()
