"""ResNet-50 (He et al. 2015) on the same engine.

BASELINE.json config[4]: "BN + residual path on same DP/all-reduce engine".
Adds K17 (BatchNorm) and K18 (residual add+ReLU); 1x1 convs are the K8 GEMM
with M = N*H*W. NHWC throughout; BN fuses ReLU where
the canonical graph has conv->BN->ReLU.
"""

from __future__ import annotations

import torch.nn as nn

from .. import ops
from ..config import Config
from .layers import BatchNorm, Conv2d, Linear, MaxPool


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin: int, planes: int, stride: int = 1,
                 downsample: bool = False):
        super().__init__()
        cout = planes * self.expansion
        # conv->BN pairs: conv has no bias/relu; BN fuses ReLU except the last
        self.conv1 = Conv2d(cin, planes, 1, 1, 0, relu=False, bias=False)
        self.bn1 = BatchNorm(planes, relu=True)
        self.conv2 = Conv2d(planes, planes, 3, stride, 1, relu=False, bias=False)
        self.bn2 = BatchNorm(planes, relu=True)
        self.conv3 = Conv2d(planes, cout, 1, 1, 0, relu=False, bias=False)
        self.bn3 = BatchNorm(cout, relu=False)
        if downsample:
            self.ds_conv = Conv2d(cin, cout, 1, stride, 0, relu=False, bias=False)
            self.ds_bn = BatchNorm(cout, relu=False)
        else:
            self.ds_conv = None

    def forward(self, x):
        idt = x
        if self.ds_conv is not None:
            idt = self.ds_bn(self.ds_conv(x))
        y = self.bn1(self.conv1(x))
        y = self.bn2(self.conv2(y))
        # K17+K18 fused: bn3's apply pass adds the residual and ReLUs
        return self.bn3(self.conv3(y), res=idt)


class ResNet50(nn.Module):
    def __init__(self, cfg: Config):
        super().__init__()
        self.stem_conv = Conv2d(cfg.channels, 64, 7, 2, 3, relu=False, bias=False)
        self.stem_bn = BatchNorm(64, relu=True)
        self.stem_pool = MaxPool(3, 2, (1, 1))
        layers = []
        cin = 64
        for planes, blocks, stride in ((64, 3, 1), (128, 4, 2),
                                       (256, 6, 2), (512, 3, 2)):
            for b in range(blocks):
                s = stride if b == 0 else 1
                layers.append(Bottleneck(cin, planes, s, downsample=(b == 0)))
                cin = planes * Bottleneck.expansion
        self.layers = nn.Sequential(*layers)
        self.fc = Linear(2048, cfg.num_classes, relu=False, std=0.01)

    def forward(self, x):
        x = self.stem_pool(self.stem_bn(self.stem_conv(x)))
        x = self.layers(x)
        x = x.mean(dim=(1, 2))          # global average pool (NHWC)
        return self.fc(x)
