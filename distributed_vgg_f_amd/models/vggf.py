"""VGG-F (CNN-F, Chatfield et al. BMVC 2014) — the reference's one model.

Exact geometry per SURVEY.md section 2.3 (60,834,536 params, 1.45 GFLOP
fwd/img at 224x224):

  conv1 64x3x11x11 s4 p0 -> LRN -> pool 3x3 s2 pad(0,1)   # 224 -> 54 -> 27
  conv2 256x64x5x5 s1 p2 -> LRN -> pool 3x3 s2            # 27 -> 13
  conv3/4/5 256x256x3x3 s1 p1; pool after conv5           # 13 -> 6
  fc6 9216->4096 ReLU (dropout), fc7 4096->4096 ReLU (dropout), fc8 ->1000

`uniform_blocks=True` applies LRN after EVERY conv block — toward the
literal reading of the driver's "five conv blocks (conv+ReLU+LRN+maxpool)"
description (BASELINE.json north_star); default is the canonical paper
placement. (A 3x3/s2 maxpool after all five blocks is geometrically
impossible at 224x224 — 54->27->13->6->2->0 — so pools stay at blocks
1/2/5 in both presets.) Both presets use the same kernels (SURVEY.md
section 2.3 decision note).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..config import Config
from .layers import LRN, Conv2d, Dropout, Linear, MaxPool


class VGGF(nn.Module):
    def __init__(self, cfg: Config):
        super().__init__()
        u = cfg.uniform_blocks
        lrn = dict(size=cfg.lrn_size, alpha=cfg.lrn_alpha, beta=cfg.lrn_beta,
                   k=cfg.lrn_k)
        feats = []
        # block 1: 3x224x224 -> 64x54x54 -> pool -> 64x27x27
        feats += [Conv2d(3, 64, 11, stride=4, pad=0), LRN(**lrn),
                  MaxPool(3, 2, (0, 1))]
        # block 2: -> 256x27x27 -> pool -> 256x13x13
        feats += [Conv2d(64, 256, 5, stride=1, pad=2), LRN(**lrn),
                  MaxPool(3, 2, (0, 0))]
        # blocks 3..5: 256x13x13; pool after conv5 only
        for _ in (3, 4, 5):
            feats += [Conv2d(256, 256, 3, stride=1, pad=1)]
            if u:
                feats += [LRN(**lrn)]
        feats += [MaxPool(3, 2, (0, 0))]
        self.features = nn.Sequential(*feats)

        # flatten size: canonical 256*6*6 = 9216; uniform-blocks shrinks it.
        with torch.no_grad():
            probe = torch.zeros(1, cfg.image_size, cfg.image_size, cfg.channels)
            flat = self.features(probe).numel()
        self.flat = flat
        self.fc6 = Linear(flat, 4096, relu=True)
        self.drop6 = Dropout(cfg.dropout, cfg.seed)
        self.fc7 = Linear(4096, 4096, relu=True)
        self.drop7 = Dropout(cfg.dropout, cfg.seed + 1)
        self.fc8 = Linear(4096, cfg.num_classes, relu=False, std=0.01)

    def forward(self, x):
        x = self.features(x)
        x = x.reshape(x.shape[0], -1)
        x = self.drop6(self.fc6(x))
        x = self.drop7(self.fc7(x))
        return self.fc8(x)
