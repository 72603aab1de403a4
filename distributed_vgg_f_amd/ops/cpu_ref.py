"""CPU fp32 reference implementations (the numerics oracle).

Every HIP kernel in csrc/ is tested against the op here with the same NHWC /
HWIO layout contract (test tier T-unit/T-model).
These are plain differentiable torch compositions, so backward references
come from torch autograd. They are also the CPU execution path of the
framework (config[0] of BASELINE.json: VGG-F fwd+bwd on CPU, batch=4).

Layout contract (package-wide):
  activations x: [N, H, W, C]
  conv weights : [kh, kw, Cin, Cout]   (HWIO — implicit-GEMM B operand is
                                        row-major [K= kh*kw*Cin, Cout])
  fc weights   : [K, N]
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn.functional as F


def conv2d(x: torch.Tensor, w: torch.Tensor, bias: Optional[torch.Tensor],
           stride: int, pad: int, relu: bool = False) -> torch.Tensor:
    """NHWC conv fwd (+bias, +ReLU). K1 reference."""
    xc = x.permute(0, 3, 1, 2)                    # NCHW
    wc = w.permute(3, 2, 0, 1)                    # OIHW
    y = F.conv2d(xc, wc, bias, stride=stride, padding=pad)
    y = y.permute(0, 2, 3, 1)
    if relu:
        y = F.relu(y)
    return y.contiguous()


def maxpool(x: torch.Tensor, kernel: int = 3, stride: int = 2,
            pad: Tuple[int, int] = (0, 0)) -> torch.Tensor:
    """NHWC 3x3/s2 maxpool with (possibly asymmetric) pad = (left/top, right/bottom).

    K6 reference. VGG-F pool1 needs pad (0, 1).
    """
    xc = x.permute(0, 3, 1, 2)
    pl, pr = pad
    if pl or pr:
        xc = F.pad(xc, (pl, pr, pl, pr), value=float("-inf"))
    y = F.max_pool2d(xc, kernel, stride)
    return y.permute(0, 2, 3, 1).contiguous()


def lrn(x: torch.Tensor, size: int = 5, alpha: float = 1e-4,
        beta: float = 0.75, k: float = 2.0) -> torch.Tensor:
    """Cross-channel LRN, torch semantics: y = x * (k + alpha/size * sum x^2)^-beta.

    K4 reference. NHWC in/out; torch's op normalizes over dim 1, so permute.
    """
    xc = x.permute(0, 3, 1, 2)
    y = F.local_response_norm(xc, size=size, alpha=alpha, beta=beta, k=k)
    return y.permute(0, 2, 3, 1).contiguous()


def linear(x: torch.Tensor, w: torch.Tensor, bias: Optional[torch.Tensor],
           relu: bool = False) -> torch.Tensor:
    """y = x @ w (+bias)(+ReLU); x [M,K], w [K,N]. K8 reference."""
    y = x @ w
    if bias is not None:
        y = y + bias
    if relu:
        y = F.relu(y)
    return y


def dropout(x: torch.Tensor, p: float, mask: torch.Tensor) -> torch.Tensor:
    """Deterministic-mask dropout (mask supplied so CPU/GPU can be compared).

    K12 reference; mask is {0,1} with P(1)=1-p; scaling 1/(1-p).
    """
    return x * mask / (1.0 - p)


def softmax_xent(logits: torch.Tensor, labels: torch.Tensor
                 ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Fused softmax + cross-entropy: (per-row loss [M], dlogits [M,C]).

    K13 reference. dlogits = (softmax - onehot) — NOT divided by M; the mean
    is applied by the caller through the SGD grad_scale (matches the kernel).
    """
    ls = F.log_softmax(logits.float(), dim=1)
    loss = -ls.gather(1, labels.view(-1, 1)).squeeze(1)
    dlogits = torch.softmax(logits.float(), dim=1)
    dlogits.scatter_add_(1, labels.view(-1, 1),
                         -torch.ones_like(labels, dtype=dlogits.dtype).view(-1, 1))
    return loss, dlogits


def batchnorm(x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor,
              eps: float = 1e-5, relu: bool = False) -> torch.Tensor:
    """Training-mode per-channel BN over N,H,W (NHWC). K17 reference."""
    mean = x.mean(dim=(0, 1, 2))
    var = x.var(dim=(0, 1, 2), unbiased=False)
    y = (x - mean) / torch.sqrt(var + eps) * gamma + beta
    if relu:
        y = F.relu(y)
    return y


def add_relu(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Residual add + ReLU (K18 reference)."""
    return F.relu(a + b)


def sgd_momentum(w: torch.Tensor, g: torch.Tensor, m: torch.Tensor,
                 lr: float, momentum: float, weight_decay: float = 0.0,
                 grad_scale: float = 1.0) -> None:
    """Flat-arena fused SGD+momentum semantics (K14 reference). In-place.

    m = momentum*m + grad_scale*g (+ wd*w);  w -= lr*m
    """
    g32 = g.float() * grad_scale
    if weight_decay:
        g32 = g32 + weight_decay * w
    m.mul_(momentum).add_(g32)
    w.add_(m, alpha=-lr)
