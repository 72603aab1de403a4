"""VGG-16 (configuration D, Simonyan & Zisserman 2014) on the same engine.

BASELINE.json config[3]: "deeper 3x3 conv stack, stresses implicit-GEMM
tiling". All convs are 3x3 s1 p1 (the conv3-class kernel of SURVEY.md
section 2.4 K1 at many channel widths); 2x2 s2 maxpools; no LRN.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..config import Config
from .layers import Conv2d, Dropout, Linear, MaxPool


_CFG_D = [64, 64, "M", 128, 128, "M", 256, 256, 256, "M",
          512, 512, 512, "M", 512, 512, 512, "M"]


class VGG16(nn.Module):
    def __init__(self, cfg: Config):
        super().__init__()
        feats = []
        cin = cfg.channels
        for v in _CFG_D:
            if v == "M":
                feats += [MaxPool(2, 2, (0, 0))]
            else:
                feats += [Conv2d(cin, v, 3, stride=1, pad=1)]
                cin = v
        self.features = nn.Sequential(*feats)
        with torch.no_grad():
            probe = torch.zeros(1, cfg.image_size, cfg.image_size, cfg.channels)
            flat = self.features(probe).numel()     # 512*7*7 = 25088 at 224
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
