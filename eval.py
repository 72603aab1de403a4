#!/usr/bin/env python3
"""Evaluation entrypoint (reference R7 equivalent): restore the latest
checkpoint and report accuracy over eval batches.

    python eval.py --model vggf --checkpoint-dir ckpt/ --eval-steps 50
"""

import argparse

from distributed_vgg_f_amd.config import Config
from distributed_vgg_f_amd.engine.trainer import Trainer


def main():
    p = argparse.ArgumentParser(description=__doc__)
    Config.add_args(p)
    args = p.parse_args()
    cfg = Config.from_args(args)
    cfg.resume = bool(cfg.checkpoint_dir)
    t = Trainer(cfg)
    res = t.evaluate()
    if t.ctx.is_chief:
        print(res)


if __name__ == "__main__":
    main()
