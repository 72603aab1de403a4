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
