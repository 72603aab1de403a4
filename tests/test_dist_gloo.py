"""T-dist (CPU/gloo, world_size=2): bucketed overlapped all-reduce
correctness and rank-identical training.

Runs multi-process via torch.multiprocessing spawn with gloo over
127.0.0.1 — no GPU needed; the same GradReducer code path drives RCCL on
the MI355X node.
"""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from distributed_vgg_f_amd.config import Config

WORLD = 2


def _init(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)


def _run_bucket_allreduce(rank, world, port, q):
    try:
        _init(rank, world, port)
        from distributed_vgg_f_amd.engine.arena import ArenaOptimizer
        from distributed_vgg_f_amd.models import build_model
        from distributed_vgg_f_amd.parallel.dist import DistContext, GradReducer

        cfg = Config(model="vggf", image_size=64, seed=3, bucket_mb=4)
        torch.manual_seed(cfg.seed)
        model = build_model(cfg)
        arena = ArenaOptimizer(model, torch.device("cpu"), bucket_mb=4)
        ctx = DistContext(rank=rank, world_size=world, backend="gloo")
        red = GradReducer(arena, ctx, overlap=True)
        assert len(arena.buckets) > 3  # bucketing actually happened

        # grads pre-seeded analytically: rank r fills grad arena with (r+1)
        red.prepare_step()
        torch.manual_seed(100 + rank)
        x = torch.randn(2, 64, 64, 3)
        y = model(x)
        y.square().mean().backward()   # fires the hooks
        red.finish()
        # after SUM all-reduce each rank holds sum over ranks
        g0 = arena.g.clone()
        gathered = [torch.zeros_like(g0) for _ in range(world)]
        dist.all_gather(gathered, g0)
        assert torch.allclose(gathered[0], gathered[1], atol=1e-6)

        # step with grad_scale and verify replicas stay bitwise identical
        arena.step(grad_scale=0.5)
        ws = [torch.zeros_like(arena.w32) for _ in range(world)]
        dist.all_gather(ws, arena.w32)
        assert torch.equal(ws[0], ws[1])
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        import traceback
        q.put((rank, traceback.format_exc()))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def _run_trainer_2rank(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["WORLD_SIZE"] = str(world)
        os.environ["RANK"] = str(rank)
        os.environ["LOCAL_RANK"] = str(rank)
        from distributed_vgg_f_amd.engine.trainer import Trainer
        cfg = Config(model="vggf", image_size=64, batch_size=2, train_steps=3,
                     log_every=0, backend="gloo", bucket_mb=8)
        t = Trainer(cfg)
        t.train(3)
        # replicas identical after 3 sync-SGD steps
        ws = [torch.zeros_like(t.arena.w32) for _ in range(world)]
        dist.all_gather(ws, t.arena.w32)
        assert torch.equal(ws[0], ws[1])
        q.put((rank, "ok"))
    except Exception:
        import traceback
        q.put((rank, traceback.format_exc()))
    finally:
        for k in ("WORLD_SIZE", "RANK", "LOCAL_RANK"):
            os.environ.pop(k, None)
        if dist.is_initialized():
            dist.destroy_process_group()


def _run_ring_allreduce(rank, world, port, q):
    try:
        _init(rank, world, port)
        from distributed_vgg_f_amd.parallel.dist import ring_all_reduce_

        torch.manual_seed(50 + rank)
        for n in (7, 64, 1 << 16, (1 << 16) + 13):   # odd sizes incl. tails
            t = torch.randn(n)
            ref = t.clone()
            dist.all_reduce(ref)                     # library SUM
            ring_all_reduce_(t)                      # hand-rolled ring
            assert torch.allclose(t, ref, atol=1e-5), n
        q.put((rank, "ok"))
    except Exception:  # pragma: no cover
        import traceback
        q.put((rank, traceback.format_exc()))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def _run_trainer_ring_2rank(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["WORLD_SIZE"] = str(world)
        os.environ["RANK"] = str(rank)
        os.environ["LOCAL_RANK"] = str(rank)
        from distributed_vgg_f_amd.engine.trainer import Trainer
        cfg = Config(model="vggf", image_size=64, batch_size=2, train_steps=2,
                     log_every=0, backend="gloo", bucket_mb=8,
                     ring_fallback=True)
        t = Trainer(cfg)
        t.train(2)
        ws = [torch.zeros_like(t.arena.w32) for _ in range(world)]
        dist.all_gather(ws, t.arena.w32)
        assert torch.equal(ws[0], ws[1])
        q.put((rank, "ok"))
    except Exception:  # pragma: no cover
        import traceback
        q.put((rank, traceback.format_exc()))
    finally:
        for k in ("WORLD_SIZE", "RANK", "LOCAL_RANK"):
            os.environ.pop(k, None)
        if dist.is_initialized():
            dist.destroy_process_group()


def _spawn(fn, port):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=fn, args=(r, WORLD, port, q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    results = [q.get() for _ in range(WORLD)]
    for p in procs:
        p.join(timeout=120)
    for rank, msg in results:
        assert msg == "ok", f"rank {rank}:\n{msg}"


@pytest.mark.timeout(300)
def test_bucketed_allreduce_gloo():
    _spawn(_run_bucket_allreduce, 29711)


@pytest.mark.timeout(300)
def test_trainer_replicas_identical_gloo():
    _spawn(_run_trainer_2rank, 29713)


@pytest.mark.timeout(300)
def test_ring_allreduce_matches_library():
    """SURVEY.md 5.8 fallback: the hand-rolled ring reduce-scatter +
    all-gather must equal the library all-reduce (SUM) at odd sizes."""
    _spawn(_run_ring_allreduce, 29717)


@pytest.mark.timeout(300)
def test_trainer_ring_fallback_gloo():
    _spawn(_run_trainer_ring_2rank, 29719)
