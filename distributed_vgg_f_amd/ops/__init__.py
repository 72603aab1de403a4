"""Functional op layer: dispatches CPU -> cpu_ref (torch fp32, differentiable)
and CUDA -> hand-written HIP kernels via custom autograd Functions.

This is the single seam between the model zoo and the two execution paths.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from . import cpu_ref
from . import native  # noqa: F401  (re-export for engine / tests)


def conv2d(x, w, bias=None, stride: int = 1, pad: int = 0, relu: bool = False):
    if x.is_cuda:
        from .autograd import ConvFunction
        return ConvFunction.apply(x, w, bias, stride, pad, relu)
    return cpu_ref.conv2d(x, w, bias, stride, pad, relu)


def linear(x, w, bias=None, relu: bool = False):
    if x.is_cuda:
        from .autograd import LinearFunction
        return LinearFunction.apply(x, w, bias, relu)
    return cpu_ref.linear(x, w, bias, relu)


def lrn(x, size: int = 5, alpha: float = 1e-4, beta: float = 0.75, k: float = 2.0):
    if x.is_cuda:
        from .autograd import LRNFunction
        return LRNFunction.apply(x, size, alpha, beta, k)
    return cpu_ref.lrn(x, size, alpha, beta, k)


def maxpool(x, kernel: int = 3, stride: int = 2, pad: Tuple[int, int] = (0, 0)):
    if x.is_cuda:
        from .autograd import MaxPoolFunction
        return MaxPoolFunction.apply(x, kernel, stride, pad[0], pad[1])
    return cpu_ref.maxpool(x, kernel, stride, pad)


def dropout(x, p: float, seed: int = 0, offset: int = 0,
            mask: Optional[torch.Tensor] = None):
    if p <= 0.0:
        return x
    if x.is_cuda:
        from .autograd import DropoutFunction
        return DropoutFunction.apply(x, p, seed, offset)
    if mask is None:
        gen = torch.Generator().manual_seed(seed + offset)
        mask = (torch.rand(x.shape, generator=gen) >= p).to(x.dtype)
    return cpu_ref.dropout(x, p, mask)


def softmax_xent(logits, labels):
    """Returns per-row loss [M] (fp32)."""
    if logits.is_cuda:
        from .autograd import SoftmaxXentFunction
        return SoftmaxXentFunction.apply(logits, labels)
    # CPU: differentiable composition
    import torch.nn.functional as F
    return F.cross_entropy(logits.float(), labels, reduction="none")


def batchnorm(x, gamma, beta, eps: float = 1e-5, relu: bool = False,
              run_mean=None, run_var=None, momentum: float = 0.9, res=None):
    if x.is_cuda:
        if res is not None:
            from .autograd import BNAddReluFunction
            return BNAddReluFunction.apply(x, gamma, beta, res, eps,
                                           run_mean, run_var, momentum)
        from .autograd import BatchNormFunction
        return BatchNormFunction.apply(x, gamma, beta, eps, relu,
                                       run_mean, run_var, momentum)
    y = cpu_ref.batchnorm(x, gamma, beta, eps, relu and res is None)
    if res is not None:
        y = cpu_ref.add_relu(y, res)
    if run_mean is not None:
        with torch.no_grad():
            m = x.float().mean(dim=(0, 1, 2))
            v = x.float().var(dim=(0, 1, 2), unbiased=False)
            run_mean.mul_(momentum).add_(m, alpha=1 - momentum)
            run_var.mul_(momentum).add_(v, alpha=1 - momentum)
    return y


def add_relu(a, b):
    if a.is_cuda:
        from .autograd import AddReluFunction
        return AddReluFunction.apply(a, b)
    return cpu_ref.add_relu(a, b)


def sgd_momentum_step(w32, w_model, g, m32, lr, momentum, weight_decay=0.0,
                      grad_scale=1.0):
    """Fused flat-arena SGD+momentum (K14). On GPU one kernel updates the fp32
    master, momentum, and rewrites the model-dtype param arena; on CPU w_model
    IS w32 and the torch composition applies."""
    if w32.is_cuda:
        native.ext().sgd_step(w32, w_model, g, m32, float(lr), float(momentum),
                              float(weight_decay), float(grad_scale))
    else:
        cpu_ref.sgd_momentum(w32, g, m32, lr, momentum, weight_decay, grad_scale)
        if w_model.data_ptr() != w32.data_ptr():
            w_model.copy_(w32)
