"""Layer modules over the functional op seam (NHWC / HWIO).

Thin nn.Module wrappers so the model zoo composes naturally; every compute
call goes through distributed_vgg_f_amd.ops (CPU torch reference or HIP
kernels). Parameters are created fp32 on CPU at init; the GPU engine
re-homes them into contiguous bf16 arenas (engine/arena.py).
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn as nn

from .. import ops


class Conv2d(nn.Module):
    """NHWC conv + optional fused bias/ReLU. Weight HWIO [kh,kw,Ci,Co]."""

    def __init__(self, cin: int, cout: int, kernel: int, stride: int = 1,
                 pad: int = 0, relu: bool = True, bias: bool = True):
        super().__init__()
        self.stride, self.pad, self.relu = stride, pad, relu
        fan_in = kernel * kernel * cin
        std = math.sqrt(2.0 / fan_in)
        self.weight = nn.Parameter(torch.randn(kernel, kernel, cin, cout) * std)
        self.bias = nn.Parameter(torch.zeros(cout)) if bias else None

    def forward(self, x):
        return ops.conv2d(x, self.weight, self.bias, self.stride, self.pad,
                          self.relu)


class Linear(nn.Module):
    """x [M,K] @ w [K,N] + optional fused bias/ReLU."""

    def __init__(self, k: int, n: int, relu: bool = False, bias: bool = True,
                 std: Optional[float] = None):
        super().__init__()
        self.relu = relu
        std = std if std is not None else math.sqrt(2.0 / k)
        self.weight = nn.Parameter(torch.randn(k, n) * std)
        self.bias = nn.Parameter(torch.zeros(n)) if bias else None

    def forward(self, x):
        return ops.linear(x, self.weight, self.bias, self.relu)


class LRN(nn.Module):
    def __init__(self, size: int = 5, alpha: float = 1e-4, beta: float = 0.75,
                 k: float = 2.0):
        super().__init__()
        self.size, self.alpha, self.beta, self.k = size, alpha, beta, k

    def forward(self, x):
        return ops.lrn(x, self.size, self.alpha, self.beta, self.k)


class MaxPool(nn.Module):
    def __init__(self, kernel: int = 3, stride: int = 2,
                 pad: Tuple[int, int] = (0, 0)):
        super().__init__()
        self.kernel, self.stride, self.pad = kernel, stride, pad

    def forward(self, x):
        return ops.maxpool(x, self.kernel, self.stride, self.pad)


class Dropout(nn.Module):
    """Counter-based dropout: offset advances per call so steps decorrelate."""

    def __init__(self, p: float, seed: int = 0):
        super().__init__()
        self.p, self.seed, self._calls = p, seed, 0

    def forward(self, x):
        if not self.training or self.p <= 0.0:
            return x
        self._calls += 1
        return ops.dropout(x, self.p, self.seed, self._calls)


class BatchNorm(nn.Module):
    """Training-mode BN (NHWC, per-channel) with optional fused ReLU.

    Running stats are tracked for eval parity with the reference's
    inference path but are not used in training mode.
    """

    def __init__(self, c: int, eps: float = 1e-5, momentum: float = 0.9,
                 relu: bool = False):
        super().__init__()
        self.eps, self.momentum, self.relu = eps, momentum, relu
        self.weight = nn.Parameter(torch.ones(c))
        self.bias = nn.Parameter(torch.zeros(c))
        self.register_buffer("running_mean", torch.zeros(c))
        self.register_buffer("running_var", torch.ones(c))

    def forward(self, x, res=None):
        if self.training:
            return ops.batchnorm(x, self.weight, self.bias, self.eps,
                                 self.relu, self.running_mean,
                                 self.running_var, self.momentum, res=res)
        rstd = torch.rsqrt(self.running_var.to(x.dtype) + self.eps)
        y = (x - self.running_mean.to(x.dtype)) * rstd * self.weight + self.bias
        if res is not None:
            return torch.relu(y + res)
        return torch.relu(y) if self.relu else y
