"""T-ckpt: save -> restore -> bitwise-equal masters; resume continues step;
TF-protocol latest-pointer file (SURVEY.md sections 4.2, 5.4)."""

import os

import torch

from distributed_vgg_f_amd.config import Config
from distributed_vgg_f_amd.engine import checkpoint as ckpt
from distributed_vgg_f_amd.engine.arena import ArenaOptimizer
from distributed_vgg_f_amd.models import build_model


def _mk(tmp_path):
    cfg = Config(model="vggf", image_size=64, seed=7)
    torch.manual_seed(cfg.seed)
    model = build_model(cfg)
    arena = ArenaOptimizer(model, torch.device("cpu"))
    return model, arena


def test_roundtrip_bitwise(tmp_path):
    _, arena = _mk(tmp_path)
    arena.m32.normal_()
    w_before = arena.w32.clone()
    m_before = arena.m32.clone()
    path = ckpt.save(str(tmp_path), arena, step=42)
    assert os.path.basename(path) == "model.ckpt-42.npz"
    # layout: TF-protocol pointer file
    with open(tmp_path / "checkpoint") as fh:
        assert 'model_checkpoint_path: "model.ckpt-42"' in fh.read()
    # clobber and restore
    arena.w32.zero_()
    arena.m32.zero_()
    step = ckpt.restore_latest(str(tmp_path), arena)
    assert step == 42
    assert torch.equal(arena.w32, w_before)
    assert torch.equal(arena.m32, m_before)


def test_tf_slim_names_present(tmp_path):
    import numpy as np
    _, arena = _mk(tmp_path)
    path = ckpt.save(str(tmp_path), arena, step=1)
    keys = set(np.load(path).files)
    for want in ("conv1/weights", "conv1/biases", "conv5/weights",
                 "fc6/weights", "fc8/biases"):
        assert want in keys, keys


def test_latest_pointer_tracks_newest(tmp_path):
    _, arena = _mk(tmp_path)
    ckpt.save(str(tmp_path), arena, step=1)
    ckpt.save(str(tmp_path), arena, step=9)
    assert ckpt.latest(str(tmp_path)).endswith("model.ckpt-9.npz")


def test_bn_buffers_and_dropout_calls_roundtrip(tmp_path):
    """BN running stats + dropout RNG counters must
    survive save/restore (resnet50 eval after resume was wrong without)."""
    cfg = Config(model="resnet50", image_size=64, seed=3)
    torch.manual_seed(cfg.seed)
    model = build_model(cfg)
    arena = ArenaOptimizer(model, torch.device("cpu"))
    # perturb the BN running stats away from init
    for name, buf in model.named_buffers():
        buf.add_(1.5)
    bufs_before = {n: b.clone() for n, b in model.named_buffers()}
    assert bufs_before, "resnet50 should have BN buffers"
    ckpt.save(str(tmp_path), arena, step=5, model=model)
    for _, buf in model.named_buffers():
        buf.zero_()
    step = ckpt.restore_latest(str(tmp_path), arena, model=model)
    assert step == 5
    for n, b in model.named_buffers():
        assert torch.equal(b, bufs_before[n]), n

    # dropout call counters (vggf has drop6/drop7)
    cfg = Config(model="vggf", image_size=64, dropout=0.5, seed=3)
    torch.manual_seed(cfg.seed)
    model = build_model(cfg)
    arena = ArenaOptimizer(model, torch.device("cpu"))
    model.drop6._calls, model.drop7._calls = 17, 19
    ckpt.save(str(tmp_path), arena, step=6, model=model)
    model.drop6._calls = model.drop7._calls = 0
    ckpt.restore_latest(str(tmp_path), arena, model=model)
    assert (model.drop6._calls, model.drop7._calls) == (17, 19)


def test_scaled_lr():
    cfg = Config(lr=0.01)
    assert cfg.scaled_lr(8) == 0.01          # default: no scaling
    cfg = Config(lr=0.01, scale_lr=True)
    assert abs(cfg.scaled_lr(8) - 0.08) < 1e-12


def test_resume_path_via_trainer(tmp_path):
    from distributed_vgg_f_amd.engine.trainer import Trainer
    cfg = Config(model="vggf", image_size=64, batch_size=2, train_steps=2,
                 log_every=0, checkpoint_dir=str(tmp_path))
    t = Trainer(cfg)
    t.train(2)
    assert ckpt.latest(str(tmp_path)) is not None
    cfg2 = Config(model="vggf", image_size=64, batch_size=2, train_steps=1,
                  log_every=0, checkpoint_dir=str(tmp_path), resume=True)
    t2 = Trainer(cfg2)
    assert t2.step_i == 2
    assert torch.equal(t2.arena.w32, t.arena.w32)
