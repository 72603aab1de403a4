#!/usr/bin/env python3
"""Distributed training entrypoint (reference R1/R6 equivalent).

Single GPU / CPU:
    python train.py --model vggf --batch-size 256 --train-steps 100

One process per GPU over RCCL (replaces the reference's
--job_name/--ps_hosts/--worker_hosts parameter-server bring-up,
SURVEY.md section 3.1):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 train.py --model vggf --batch-size 256
"""

import argparse

from distributed_vgg_f_amd.config import Config
from distributed_vgg_f_amd.engine.trainer import Trainer


def main():
    p = argparse.ArgumentParser(description=__doc__)
    Config.add_args(p)
    cfg = Config.from_args(p.parse_args())
    t = Trainer(cfg)
    if t.ctx.is_chief:
        print(cfg.to_json())
    t.train()


if __name__ == "__main__":
    main()
