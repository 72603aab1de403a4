"""Config system.

Replaces the reference's tf.app.flags module globals (SURVEY.md section 5.6:
job_name/task_index/ps_hosts/worker_hosts/batch_size/learning_rate/
train_steps/data_dir/log_dir) with one dataclass + argparse/JSON. The PS
cluster flags have no analog here: distribution is one-process-per-GPU
torch.distributed over RCCL, configured by the standard RANK/WORLD_SIZE env
(torchrun), so the CLI surface keeps only the training-relevant knobs plus
per-block LRN/pool toggles (SURVEY.md section 2.3 note on the uniform-blocks
reading of the driver's model description).
"""

from __future__ import annotations

import argparse
import dataclasses
import json
from dataclasses import dataclass
from typing import Optional


@dataclass
class Config:
    # model
    model: str = "vggf"               # vggf | vgg16 | resnet50
    num_classes: int = 1000
    image_size: int = 224
    uniform_blocks: bool = False       # LRN+pool after EVERY conv block (driver's
    #                                    literal reading); default = canonical VGG-F
    #                                    (LRN after conv1/2, pool after conv1/2/5).
    lrn_size: int = 5                  # cross-channel window n
    lrn_alpha: float = 1e-4            # torch semantics: k + alpha/n * sum(x^2)
    lrn_beta: float = 0.75
    lrn_k: float = 2.0
    dropout: float = 0.0               # fc6/fc7 dropout keep-rate complement; 0 = off
    #                                    (training-accuracy feature; off for the
    #                                     throughput benchmark)

    # training
    batch_size: int = 256              # per-GPU
    lr: float = 0.01
    scale_lr: bool = False             # linear LR scaling: lr *= world_size
    #                                    (global batch grows with DP width;
    #                                     off by default = reference behavior)
    momentum: float = 0.9
    weight_decay: float = 0.0
    train_steps: int = 100
    eval_steps: int = 50
    seed: int = 1234

    # precision
    precision: str = "bf16"            # bf16 (GPU) | fp32 (CPU reference path)
    grad_dtype: str = "bf16"           # gradient arena / all-reduce dtype

    # distribution
    bucket_mb: int = 25                # gradient all-reduce bucket size (bf16 MB)
    overlap_comm: bool = True          # overlap all-reduce with backward
    backend: str = "auto"              # auto -> nccl(RCCL) on GPU, gloo on CPU
    ring_fallback: bool = False        # hand-rolled p2p ring all-reduce
    #                                    instead of the library collective
    #                                    (SURVEY.md 5.8 fallback seam;
    #                                     disables overlap)

    # io / checkpointing
    checkpoint_dir: Optional[str] = None
    checkpoint_every: int = 0          # steps; 0 = only at end
    resume: bool = False
    log_every: int = 10
    metrics_path: Optional[str] = None  # JSONL metrics
    phase_timers: bool = False          # hipEvent per-phase ms (fwd/bwd/opt)
    use_graph: bool = False             # hipGraph-capture the train step
    #                                     (single-GPU; falls back to eager)

    # debug / fault injection
    fault_inject_step: int = -1
    fault_inject_rank: int = -1

    # data
    data: str = "synthetic"            # synthetic | records (NPZ shards, R5)
    data_dir: Optional[str] = None     # shard directory for data=records
    channels: int = 3

    def scaled_lr(self, world_size: int) -> float:
        return self.lr * world_size if self.scale_lr else self.lr

    @staticmethod
    def add_args(p: argparse.ArgumentParser) -> None:
        # SUPPRESS defaults: only explicitly-passed flags land in the
        # namespace, so precedence is dataclass default < JSON file < CLI
        for f in dataclasses.fields(Config):
            name = "--" + f.name.replace("_", "-")
            if f.type == "bool" or isinstance(f.default, bool):
                p.add_argument(name, type=lambda s: s.lower() in ("1", "true", "yes"),
                               default=argparse.SUPPRESS)
            elif f.default is None:
                p.add_argument(name, type=str, default=argparse.SUPPRESS)
            else:
                p.add_argument(name, type=type(f.default),
                               default=argparse.SUPPRESS)
        p.add_argument("--config-json", type=str, default=None,
                       help="JSON file of Config overrides (CLI wins)")

    @staticmethod
    def from_args(args: argparse.Namespace) -> "Config":
        base = {}
        if getattr(args, "config_json", None):
            with open(args.config_json) as fh:
                base = json.load(fh)
        names = {f.name for f in dataclasses.fields(Config)}
        cfg = {k: v for k, v in base.items() if k in names}
        for k, v in vars(args).items():
            if k in names:
                cfg[k] = v
        return Config(**cfg)

    def to_json(self) -> str:
        return json.dumps(dataclasses.asdict(self), indent=2)
