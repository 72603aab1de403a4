"""train()/eval() engine (SURVEY.md sections 3.4, 1.3).

The reference's MonitoredTrainingSession step loop (sess.run(train_op) with
checkpoint/logging hooks) becomes an explicit
MI355X-native loop: static per-step schedule, zero host syncs inside the
step (loss D2H only at log windows), bucketed all-reduce overlapped with
backward, one fused SGD over the arenas.
"""

from __future__ import annotations

import time
from typing import Optional

import torch

from .. import ops
from ..config import Config
from ..data import RecordPipeline, SyntheticPipeline
from ..models import build_model
from ..parallel.dist import DistContext, GradReducer, broadcast_init, \
    init_distributed
from . import checkpoint as ckpt
from .arena import ArenaOptimizer
from .metrics import MetricsLogger, PhaseTimers


def _device(ctx: DistContext) -> torch.device:
    if torch.cuda.is_available():
        # tolerate more ranks than GPUs (single-GPU multi-rank smoke tests)
        return torch.device("cuda",
                            ctx.local_rank % max(torch.cuda.device_count(), 1))
    return torch.device("cpu")


class Trainer:
    def __init__(self, cfg: Config, ctx: Optional[DistContext] = None,
                 device: Optional[torch.device] = None):
        self.cfg = cfg
        self.ctx = ctx if ctx is not None else init_distributed(cfg.backend)
        self.device = device if device is not None else _device(self.ctx)

        torch.manual_seed(cfg.seed)          # same init on every rank; C4
        self.model = build_model(cfg)        # broadcast still run for safety
        self.model.to(self.device)           # buffers (BN stats); params get
        self.model.train()                   # re-homed into arenas below
        for m in self.model.modules():       # decorrelate dropout across DP
            if type(m).__name__ == "Dropout":  # ranks (each rank sees its
                m.seed += self.ctx.rank * 9973  # own shard of the batch)

        compute_dtype = torch.bfloat16 if (cfg.precision == "bf16" and
                                           self.device.type == "cuda") \
            else torch.float32
        grad_dtype = torch.bfloat16 if (cfg.grad_dtype == "bf16" and
                                        self.device.type == "cuda") \
            else torch.float32
        self.arena = ArenaOptimizer(self.model, self.device, compute_dtype,
                                    grad_dtype,
                                    cfg.scaled_lr(self.ctx.world_size),
                                    cfg.momentum, cfg.weight_decay,
                                    cfg.bucket_mb)
        broadcast_init(self.arena, self.ctx)
        self.reducer = GradReducer(self.arena, self.ctx, cfg.overlap_comm,
                                   cfg.ring_fallback)
        if cfg.data == "records":
            self.pipeline = RecordPipeline(cfg.data_dir, cfg.batch_size,
                                           self.device, compute_dtype,
                                           rank=self.ctx.rank,
                                           world_size=self.ctx.world_size,
                                           seed=cfg.seed)
        else:
            self.pipeline = SyntheticPipeline(cfg.batch_size, cfg.image_size,
                                              cfg.channels, cfg.num_classes,
                                              self.device, compute_dtype,
                                              cfg.seed + self.ctx.rank)
        self.metrics = MetricsLogger(cfg.metrics_path, self.ctx.rank)
        self.timers = PhaseTimers(self.device, enabled=cfg.phase_timers)
        self.step_i = 0
        self.global_batch = cfg.batch_size * self.ctx.world_size

        if cfg.resume and cfg.checkpoint_dir:
            self.step_i = ckpt.restore_latest(cfg.checkpoint_dir, self.arena,
                                              self.model)
            if self.ctx.is_chief and self.step_i:
                self.metrics.log(event="resumed", step=self.step_i)

        self._graph = None
        if cfg.use_graph:
            self._try_capture_graph()

    # ------------------------------------------------------------------ #
    def _graph_body(self, i: int):
        """Captured step i: compute on buffer i while the OTHER buffer's
        H2D runs on a forked copy stream (cross-step prefetch: graphs are
        replayed alternately, so replay k's copy feeds replay k+1)."""
        main = torch.cuda.current_stream()
        ev_fork = torch.cuda.Event()
        ev_join = torch.cuda.Event()
        ev_fork.record(main)
        self._gcs.wait_event(ev_fork)
        with torch.cuda.stream(self._gcs):
            self._gimg[1 - i].copy_(self._gimg_h[1 - i], non_blocking=True)
            self._glab[1 - i].copy_(self._glab_h[1 - i], non_blocking=True)
            ev_join.record(self._gcs)
        logits = self.model(self._gimg[i])
        loss = ops.softmax_xent(logits, self._glab[i]).sum()
        self.arena.detach_grads()   # direct-grad: kernels write the arena
        loss.backward()
        self.arena.step(grad_scale=1.0 / self.global_batch)
        main.wait_event(ev_join)
        return loss

    def _try_capture_graph(self) -> bool:
        """Capture the train step into TWO ping-pong hipGraphs (compute on
        buffer i, prefetch buffer 1-i on a copy stream) so H2D overlaps
        compute across replays. Single-GPU only this round: RCCL
        collectives stay out of capture. Falls back to eager on any error."""
        if self.device.type != "cuda" or self.ctx.world_size > 1:
            return False
        if not isinstance(self.pipeline, SyntheticPipeline):
            return False   # capture pins fixed host buffers (synthetic pool)
        if any(getattr(m, "p", 0) > 0 for m in self.model.modules()
               if type(m).__name__ == "Dropout"):
            # capture bakes the (seed, offset) kernel args: every replay
            # would reuse the identical dropout mask — stay eager
            return False
        try:
            self._gimg_h = [self.pipeline.images[0], self.pipeline.images[1]]
            self._glab_h = [self.pipeline.labels[0], self.pipeline.labels[1]]
            self._gimg = [torch.empty(h.shape, dtype=h.dtype,
                                      device=self.device)
                          for h in self._gimg_h]
            self._glab = [torch.empty(h.shape, dtype=torch.long,
                                      device=self.device)
                          for h in self._glab_h]
            self._gcs = torch.cuda.Stream(self.device)
            for i in (0, 1):
                self._gimg[i].copy_(self._gimg_h[i], non_blocking=True)
                self._glab[i].copy_(self._glab_h[i], non_blocking=True)
            for i in (0, 1, 0):                      # allocator warmup
                self._graph_body(i)
            torch.cuda.synchronize(self.device)
            self._graphs, self._glosses = [], []
            pool = torch.cuda.graph_pool_handle()
            for i in (0, 1):
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g, pool=pool):
                    loss = self._graph_body(i)
                self._graphs.append(g)
                self._glosses.append(loss)
            self._graph = True
            self._gi = 0
            if self.ctx.is_chief:
                self.metrics.log(event="hipgraph_captured")
            return True
        except Exception as e:  # pragma: no cover - hardware dependent
            import sys
            print(f"[trainer] hipGraph capture failed, eager fallback: {e}",
                  file=sys.stderr)
            self._graph = None
            return False

    # ------------------------------------------------------------------ #
    def train_step(self) -> torch.Tensor:
        """One synchronous-SGD step over the global batch. Returns the
        per-rank summed loss tensor (device, not synced)."""
        cfg = self.cfg
        if (cfg.fault_inject_step == self.step_i and
                cfg.fault_inject_rank == self.ctx.rank):
            raise RuntimeError(
                f"fault-inject: rank {self.ctx.rank} dies at step {self.step_i}")
        if self._graph is not None:
            i = self._gi
            self._gi ^= 1
            self._graphs[i].replay()
            self.step_i += 1
            return self._glosses[i]
        tm = self.timers
        tm.start("h2d")
        images, labels = self.pipeline.next()
        tm.stop("h2d")
        self.reducer.prepare_step()
        tm.start("fwd")
        logits = self.model(images)
        loss = ops.softmax_xent(logits, labels).sum()
        tm.stop("fwd")
        tm.start("bwd")
        if self.device.type == "cuda":
            # direct-grad: backward kernels overwrite the arena views and
            # AccumulateGrad stores them back as .grad (no adds, no zeroing)
            self.arena.detach_grads()
        loss.backward()
        tm.stop("bwd")
        tm.start("ar")   # exposed comm: stream stall on bucket reductions
        self.reducer.finish()
        tm.stop("ar")
        # loss summed per sample; SUM all-reduce over ranks => scale by 1/GB
        tm.start("opt")
        self.arena.step(grad_scale=1.0 / self.global_batch)
        if self.device.type != "cuda":
            self.arena.zero_grad()
        tm.stop("opt")
        self.step_i += 1
        return loss

    def train(self, steps: Optional[int] = None) -> None:
        cfg = self.cfg
        steps = steps if steps is not None else cfg.train_steps
        window_t0 = time.time()
        window_steps = 0
        last_loss = None
        for _ in range(steps):
            last_loss = self.train_step()
            window_steps += 1
            if cfg.log_every and self.step_i % cfg.log_every == 0:
                if self.device.type == "cuda":
                    torch.cuda.synchronize(self.device)
                dt = time.time() - window_t0
                ips = self.global_batch * window_steps / dt
                self.metrics.log(step=self.step_i,
                                 loss=float(last_loss) / cfg.batch_size,
                                 images_per_sec=ips,
                                 ms_per_step=1000.0 * dt / window_steps,
                                 lr=self.arena.lr, **self.timers.collect())
                window_t0 = time.time()
                window_steps = 0
            if (cfg.checkpoint_every and cfg.checkpoint_dir and
                    self.step_i % cfg.checkpoint_every == 0 and
                    self.ctx.is_chief):
                ckpt.save(cfg.checkpoint_dir, self.arena, self.step_i,
                          self.model)
        if cfg.checkpoint_dir and self.ctx.is_chief:
            ckpt.save(cfg.checkpoint_dir, self.arena, self.step_i, self.model)

    # ------------------------------------------------------------------ #
    def _eval_pipeline(self):
        """R7 held-out data: the val-* record shards when data=records and
        they exist; otherwise the training pipeline (synthetic)."""
        if self.cfg.data == "records":
            import glob as _glob
            import os as _os
            if _glob.glob(_os.path.join(self.cfg.data_dir, "val-*.npz")):
                if not hasattr(self, "_val_pipeline"):
                    dtype = self.arena.compute_dtype
                    self._val_pipeline = RecordPipeline(
                        self.cfg.data_dir, self.cfg.batch_size, self.device,
                        dtype, split="val", rank=self.ctx.rank,
                        world_size=self.ctx.world_size, seed=self.cfg.seed)
                return self._val_pipeline
        return self.pipeline

    @torch.no_grad()
    def evaluate(self, steps: Optional[int] = None) -> dict:
        """R7: accuracy over held-out batches (val record shards when
        available, else the synthetic pipeline) from the current state."""
        cfg = self.cfg
        steps = steps if steps is not None else cfg.eval_steps
        pipeline = self._eval_pipeline()
        self.model.eval()
        correct = torch.zeros((), dtype=torch.long, device=self.device)
        total = 0
        for _ in range(steps):
            images, labels = pipeline.next()
            logits = self.model(images)
            correct += (logits.float().argmax(1) == labels).sum()
            total += labels.numel()
        self.model.train()
        acc = float(correct) / max(total, 1)
        if self.ctx.is_chief:
            self.metrics.log(event="eval", step=self.step_i, accuracy=acc,
                             samples=total)
        return {"accuracy": acc, "samples": total}
