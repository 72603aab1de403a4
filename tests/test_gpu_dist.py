"""GPU-side distributed-path tests (SURVEY.md §2.2 DP row, §5.8).

The driver's round-end scaling bench runs one rank per GPU over RCCL; the
CPU suite covers the bucket/overlap logic with gloo. These tests close the
gap in between on a single-GPU box:

  * a 2-rank RCCL smoke sharing cuda:0 (RCCL may refuse duplicate GPUs the
    way NCCL does — that outcome SKIPs rather than fails; a hang is cut by
    the join timeout so the box is never left wedged);
  * a full 2-rank Trainer step over gloo with CUDA tensors — the arena,
    post-accumulate hooks, bucketed all-reduce and fused SGD all run the
    real HIP kernels, only the wire transport differs from production.
"""

import os

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu


def _env(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")


def _rccl_worker(rank, world, port, q):
    import datetime
    import torch.distributed as dist
    _env(rank, world, port)
    try:
        torch.cuda.set_device(0)
        dist.init_process_group("nccl", rank=rank, world_size=world,
                                timeout=datetime.timedelta(seconds=60))
        t = torch.full((1024,), float(rank + 1), device="cuda:0")
        dist.all_reduce(t)
        torch.cuda.synchronize()
        ok = bool(torch.all(t == 3.0))
        q.put(("ok", rank, ok))
        dist.destroy_process_group()
    except Exception as e:  # duplicate-GPU refusal etc.
        q.put(("err", rank, repr(e)[:300]))


def _spawn_checked(fn, world, port, timeout_s, extra=()):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=fn, args=(r, world, port, q) + tuple(extra),
                         daemon=True)
             for r in range(world)]
    for p in procs:
        p.start()
    results = []
    import queue as _q
    for _ in range(world):
        try:
            results.append(q.get(timeout=timeout_s))
        except _q.Empty:
            break
    for p in procs:
        p.join(timeout=10)
        if p.is_alive():
            p.terminate()
            p.join(timeout=5)
    return results


def test_rccl_allreduce_2rank_shared_gpu():
    results = _spawn_checked(_rccl_worker, 2, 29871, 120)
    if len(results) < 2 or any(r[0] == "err" for r in results):
        pytest.skip(f"RCCL 2-rank on one GPU unavailable: {results}")
    assert all(r[2] for r in results)


def _trainer_worker(rank, world, port, q):
    _env(rank, world, port)
    from distributed_vgg_f_amd.config import Config
    from distributed_vgg_f_amd.engine.trainer import Trainer
    try:
        cfg = Config(model="vggf", image_size=64, num_classes=16,
                     batch_size=8, train_steps=2, lr=0.001,
                     backend="gloo", seed=11)
        t = Trainer(cfg)
        assert t.ctx.world_size == 2
        for _ in range(2):
            t.train_step()
        torch.cuda.synchronize()
        # after sync steps, every rank's master weights must be identical:
        # ship a checksum of the fp32 arena
        q.put(("ok", rank, float(t.arena.w32.double().sum().cpu())))
    except Exception as e:
        q.put(("err", rank, repr(e)[:300]))


def test_trainer_2rank_gloo_gpu():
    results = _spawn_checked(_trainer_worker, 2, 29873, 300)
    assert len(results) == 2, f"workers died/hung: {results}"
    errs = [r for r in results if r[0] == "err"]
    assert not errs, errs
    a, b = results[0][2], results[1][2]
    assert a == pytest.approx(b, rel=0, abs=0.0), (a, b)
