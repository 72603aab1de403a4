"""Config precedence + metrics logger coverage."""

import json

import torch

from distributed_vgg_f_amd.config import Config
from distributed_vgg_f_amd.engine.metrics import MetricsLogger


def test_config_precedence_default_json_cli(tmp_path):
    import argparse
    cfgfile = tmp_path / "cfg.json"
    cfgfile.write_text(json.dumps({"batch_size": 64, "lr": 0.5}))
    p = argparse.ArgumentParser()
    Config.add_args(p)
    # CLI overrides JSON; JSON overrides dataclass default
    args = p.parse_args(["--config", str(cfgfile), "--lr", "0.25"])
    cfg = Config.from_args(args)
    assert cfg.batch_size == 64        # from JSON (default is 256)
    assert cfg.lr == 0.25              # CLI wins over JSON's 0.5
    assert cfg.momentum == 0.9         # untouched default


def test_config_json_roundtrip():
    cfg = Config(model="vgg16", batch_size=32)
    d = json.loads(cfg.to_json())
    assert d["model"] == "vgg16" and d["batch_size"] == 32


def test_metrics_jsonl(tmp_path):
    path = tmp_path / "m.jsonl"
    m = MetricsLogger(str(path), rank=0)
    m.log(step=1, loss=2.5, img_s=1000.0)
    m.log(event="eval", step=1, accuracy=0.5)
    lines = [json.loads(l) for l in path.read_text().splitlines()]
    assert lines[0]["step"] == 1 and lines[0]["loss"] == 2.5
    assert lines[1]["event"] == "eval"


def test_metrics_rank_nonzero_silent(tmp_path):
    path = tmp_path / "m.jsonl"
    m = MetricsLogger(str(path), rank=1)
    m.log(step=1, loss=1.0)
    assert not path.exists() or path.read_text() == ""


def test_sgd_grad_scale_matches_torch():
    # arena fused-SGD semantics vs torch.optim.SGD on CPU fp32
    torch.manual_seed(3)
    import torch.nn as nn
    from distributed_vgg_f_amd.engine.arena import ArenaOptimizer
    model = nn.Sequential(nn.Linear(8, 8), nn.Linear(8, 4))
    ref = nn.Sequential(nn.Linear(8, 8), nn.Linear(8, 4))
    ref.load_state_dict(model.state_dict())
    arena = ArenaOptimizer(model, torch.device("cpu"), lr=0.1, momentum=0.9)
    opt = torch.optim.SGD(ref.parameters(), lr=0.1, momentum=0.9)
    x = torch.randn(4, 8)
    for _ in range(3):
        model.zero_grad(set_to_none=False)
        model(x).square().sum().backward()
        arena.step(grad_scale=1.0)
        arena.zero_grad()
        opt.zero_grad()
        ref(x).square().sum().backward()
        opt.step()
    for (n, p), (n2, p2) in zip(model.named_parameters(),
                                ref.named_parameters()):
        assert torch.allclose(p.detach(), p2.detach(), atol=1e-6), n
