"""fault injection -> clean failure -> resume from
checkpoint continues the step count and weights (CPU path)."""

import pytest
import torch

from distributed_vgg_f_amd.config import Config
from distributed_vgg_f_amd.engine import checkpoint as ckpt
from distributed_vgg_f_amd.engine.trainer import Trainer


def test_fault_inject_then_resume(tmp_path):
    cfg = Config(model="vggf", image_size=64, batch_size=2, train_steps=6,
                 log_every=0, checkpoint_dir=str(tmp_path),
                 checkpoint_every=2, fault_inject_step=4, fault_inject_rank=0)
    t = Trainer(cfg)
    with pytest.raises(RuntimeError, match="fault-inject"):
        t.train(6)
    # steps 0..3 ran; checkpoints at steps 2 and 4
    assert ckpt.latest(str(tmp_path)).endswith("model.ckpt-4.npz")

    cfg2 = Config(model="vggf", image_size=64, batch_size=2, train_steps=2,
                  log_every=0, checkpoint_dir=str(tmp_path), resume=True)
    t2 = Trainer(cfg2)
    assert t2.step_i == 4
    t2.train(2)
    assert t2.step_i == 6
    assert ckpt.latest(str(tmp_path)).endswith("model.ckpt-6.npz")
    assert torch.isfinite(t2.arena.w32).all()
