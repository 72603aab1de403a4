#!/usr/bin/env python3
"""Quick HBM bandwidth probe: D2D copy of a 1 GiB bf16 tensor.

Boxes in the pool vary (round-2 observation: identical memory-bound
kernels 2x apart across boxes); run this beside any benchmark so the
box's bandwidth class is recorded with the numbers.
"""
import time

import torch


def main():
    assert torch.cuda.is_available()
    n = 1 << 29  # 512Mi bf16 = 1 GiB
    a = torch.empty(n, dtype=torch.bfloat16, device="cuda")
    b = torch.empty(n, dtype=torch.bfloat16, device="cuda")
    a.normal_()
    for _ in range(3):
        b.copy_(a)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    reps = 10
    for _ in range(reps):
        b.copy_(a)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    gbs = reps * 2 * (n * 2) / dt / 1e9  # read + write bytes
    print(f"d2d_copy_GBps {gbs:.0f}")


if __name__ == "__main__":
    main()
