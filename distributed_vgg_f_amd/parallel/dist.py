"""Data-parallel engine: RCCL over xGMI, bucketed, overlapped with backward.

Replaces the reference's gRPC parameter-server sync (SURVEY.md sections 2.2,
3.1, 5.8: per-step full fp32 param pull + grad push per worker) with the
MI355X-native design mandated by BASELINE.json:

  * one process per GPU; torch.distributed backend "nccl" IS RCCL on ROCm
  * gradients live in one contiguous arena in reverse-layer order
    (engine/arena.py), so each bucket is a contiguous slice
  * a post-accumulate-grad hook fires as backward produces each param's
    grad; when every param of a bucket is ready, an async all-reduce (SUM,
    grad dtype = bf16 by default) is launched immediately — RCCL's internal
    stream overlaps it with the remaining backward
  * the all-reduce IS the sync barrier (C5) — no PS staleness semantics; the
    result equals synchronous SGD over the global batch
  * rank-0 broadcast of the init arena replaces chief var init (C4)

The same code runs the CPU/gloo multi-process test path (world_size > 1,
fp32 arenas) — SURVEY.md T-dist-local.
"""

from __future__ import annotations

import datetime
import os
from dataclasses import dataclass
from typing import TYPE_CHECKING, List, Optional

import torch
import torch.distributed as dist

if TYPE_CHECKING:  # import cycle: engine imports parallel (annotations only)
    from ..engine.arena import ArenaOptimizer


@dataclass
class DistContext:
    rank: int = 0
    world_size: int = 1
    local_rank: int = 0
    backend: Optional[str] = None

    @property
    def is_distributed(self) -> bool:
        return self.world_size > 1

    @property
    def is_chief(self) -> bool:
        return self.rank == 0


def init_distributed(backend: str = "auto",
                     device: Optional[torch.device] = None) -> DistContext:
    """Initialise torch.distributed from torchrun env vars; no-op world of 1
    when not launched distributed."""
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return DistContext()
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if backend == "auto":
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if backend == "nccl":
        # tolerate more ranks than GPUs (T-dist-local smoke: 2 ranks, 1 GPU)
        local_rank = local_rank % max(torch.cuda.device_count(), 1)
        torch.cuda.set_device(local_rank)
    if not dist.is_initialized():
        dist.init_process_group(backend=backend, rank=rank, world_size=world,
                                timeout=datetime.timedelta(seconds=300))
    return DistContext(rank, world, local_rank, backend)


def broadcast_init(arena: "ArenaOptimizer", ctx: DistContext) -> None:
    """C4: rank-0's fp32 master arena becomes everyone's; compute copy synced."""
    if not ctx.is_distributed:
        return
    dist.broadcast(arena.w32, src=0)
    arena.sync_compute_weights()


def ring_all_reduce_(t: torch.Tensor, group=None) -> torch.Tensor:
    """Hand-rolled ring all-reduce (SURVEY.md 5.8 fallback seam): a
    reduce-scatter pass followed by an all-gather pass over P-1 steps of
    point-to-point send/recv each. On ROCm the nccl backend's send/recv
    are RCCL point-to-point over xGMI; on CPU/gloo this is the testable
    reference. SUM semantics, in place, matching dist.all_reduce.

    Kept behind the same interface as the RCCL collective so the engine
    can swap it in (Config.ring_fallback) if the library collective
    misbehaves on a given stack. The library path stays the default: a
    single rcclAllReduce already builds ~7 parallel rings over the
    fully-connected xGMI links, which this single ring does not."""
    world = dist.get_world_size(group)
    if world == 1:
        return t
    rank = dist.get_rank(group)
    flat = t.view(-1)
    # pad-free chunking: torch.chunk gives world pieces (last may be short)
    n = flat.numel()
    csz = (n + world - 1) // world
    chunks = [flat[i * csz:min(n, (i + 1) * csz)] for i in range(world)]
    right = (rank + 1) % world
    left = (rank - 1) % world
    buf = torch.empty(csz, dtype=t.dtype, device=t.device)
    # reduce-scatter: after P-1 steps rank r owns the full sum of chunk
    # (r+1) % world
    for step in range(world - 1):
        s_i = (rank - step) % world          # chunk we send
        r_i = (rank - step - 1) % world      # chunk we receive+accumulate
        rb = buf[:chunks[r_i].numel()]
        ops = [dist.P2POp(dist.isend, chunks[s_i].contiguous(), right,
                          group=group),
               dist.P2POp(dist.irecv, rb, left, group=group)]
        for w in dist.batch_isend_irecv(ops):
            w.wait()
        chunks[r_i].add_(rb)
    # all-gather the owned chunks around the ring
    for step in range(world - 1):
        s_i = (rank + 1 - step) % world
        r_i = (rank - step) % world
        rb = buf[:chunks[r_i].numel()]
        ops = [dist.P2POp(dist.isend, chunks[s_i].contiguous(), right,
                          group=group),
               dist.P2POp(dist.irecv, rb, left, group=group)]
        for w in dist.batch_isend_irecv(ops):
            w.wait()
        chunks[r_i].copy_(rb)
    return t


class GradReducer:
    """Bucketed, overlapped gradient all-reduce over the arena (C1)."""

    def __init__(self, arena: "ArenaOptimizer", ctx: DistContext,
                 overlap: bool = True, ring: bool = False):
        self.arena = arena
        self.ctx = ctx
        self.ring = ring
        self.overlap = overlap and ctx.is_distributed and not ring
        self._works: List[dist.Work] = []
        self._pending: dict = {}
        self._param_bucket = {}
        if self.overlap:
            for bi, b in enumerate(arena.buckets):
                for s in b.slots:
                    self._param_bucket[id(s.param)] = bi
                    s.param.register_post_accumulate_grad_hook(self._hook)
            self._reset_counts()

    def _reset_counts(self) -> None:
        self._pending = {bi: len(b.slots)
                         for bi, b in enumerate(self.arena.buckets)}
        self._works = []

    def _hook(self, param: torch.nn.Parameter) -> None:
        bi = self._param_bucket[id(param)]
        self._pending[bi] -= 1
        if self._pending[bi] == 0:
            b = self.arena.buckets[bi]
            sl = self.arena.g[b.start:b.end]
            self._works.append(dist.all_reduce(sl, async_op=True))

    def prepare_step(self) -> None:
        if self.overlap:
            self._reset_counts()

    def finish(self) -> None:
        """Block the current stream on all outstanding bucket reductions; if
        overlap is off (or a bucket never fired), reduce the whole arena."""
        if not self.ctx.is_distributed:
            return
        if self.overlap:
            leftover = [bi for bi, n in self._pending.items() if n > 0]
            for bi in leftover:
                b = self.arena.buckets[bi]
                self._works.append(
                    dist.all_reduce(self.arena.g[b.start:b.end], async_op=True))
            for w in self._works:
                w.wait()
            self._works = []
        elif self.ring:
            # SURVEY.md 5.8 fallback: hand-rolled p2p ring (no overlap)
            for b in self.arena.buckets:
                ring_all_reduce_(self.arena.g[b.start:b.end])
        else:
            dist.all_reduce(self.arena.g)
