"""Entrypoint smoke tests (reference R1/R6/R7 equivalents) + periodic
checkpointing — run the actual CLI scripts in subprocesses on CPU."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(args, timeout=300):
    return subprocess.run([sys.executable] + args, cwd=REPO, timeout=timeout,
                          capture_output=True, text=True)


def test_train_cli_and_periodic_checkpoints(tmp_path):
    r = _run(["train.py", "--model", "vggf", "--image-size", "64",
              "--batch-size", "2", "--train-steps", "4", "--precision",
              "fp32", "--checkpoint-dir", str(tmp_path),
              "--checkpoint-every", "2", "--log-every", "2"])
    assert r.returncode == 0, r.stderr[-2000:]
    names = sorted(os.listdir(tmp_path))
    # periodic saves at steps 2 and 4 plus the end-of-training save and
    # the TF-protocol latest pointer
    assert "checkpoint" in names
    ckpts = [n for n in names if n.startswith("model.ckpt-")]
    assert any("model.ckpt-2" in n for n in ckpts), names
    assert any("model.ckpt-4" in n for n in ckpts), names


def test_eval_cli_restores(tmp_path):
    r = _run(["train.py", "--model", "vggf", "--image-size", "64",
              "--batch-size", "2", "--train-steps", "2", "--precision",
              "fp32", "--checkpoint-dir", str(tmp_path)])
    assert r.returncode == 0, r.stderr[-2000:]
    r = _run(["eval.py", "--model", "vggf", "--image-size", "64",
              "--batch-size", "2", "--eval-steps", "2", "--precision",
              "fp32", "--checkpoint-dir", str(tmp_path)])
    assert r.returncode == 0, r.stderr[-2000:]
    assert "accuracy" in r.stdout


def test_bench_cli_contract_cpu():
    r = _run(["bench.py", "--gpus", "1", "--steps", "2", "--warmup", "1",
              "--batch-size", "2"])
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["unit"] == "images/sec" and d["n_gpus"] == 1
    assert d["steps"] == 2 and d["higher_is_better"] is True
    assert d["config"]["parallelism"] == "dp1"
