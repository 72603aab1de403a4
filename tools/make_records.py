#!/usr/bin/env python3
"""Convert arrays into the NPZ record shards RecordPipeline reads (R5).

Inputs it accepts:
  * a single .npz with `images` [N,H,W,C] + `labels` [N]
  * a pair of .npy files (--images / --labels)
  * --random N: generate N random uint8 ImageNet-shaped samples (smoke data)

Example:
    python tools/make_records.py --random 4096 --out data/ --shard-size 1024
    python tools/make_records.py --npz dump.npz --out data/ --split val
"""

import argparse

import numpy as np

from distributed_vgg_f_amd.data.records import write_shards


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--npz")
    p.add_argument("--images")
    p.add_argument("--labels")
    p.add_argument("--random", type=int, default=0)
    p.add_argument("--image-size", type=int, default=224)
    p.add_argument("--num-classes", type=int, default=1000)
    p.add_argument("--out", required=True)
    p.add_argument("--split", default="train")
    p.add_argument("--shard-size", type=int, default=1024)
    p.add_argument("--seed", type=int, default=0)
    a = p.parse_args()
    if a.random:
        rng = np.random.default_rng(a.seed)
        img = rng.integers(0, 256, (a.random, a.image_size, a.image_size, 3),
                           dtype=np.uint8)
        lab = rng.integers(0, a.num_classes, a.random, dtype=np.int64)
    elif a.npz:
        with np.load(a.npz) as z:
            img, lab = z["images"], z["labels"]
    else:
        img, lab = np.load(a.images), np.load(a.labels)
    paths = write_shards(img, lab, a.out, a.shard_size, a.split)
    print(f"wrote {len(paths)} shard(s), {len(img)} samples -> {a.out}")


if __name__ == "__main__":
    main()
