"""CPU reference-op sanity: shapes, known values, and gradcheck-lite
(T-unit/T-grad of SURVEY.md section 4.2). These define the contract the HIP
kernels are tested against in the gpu-marked tests."""

import torch

from distributed_vgg_f_amd.ops import cpu_ref


def test_conv2d_shapes_vggf():
    # conv1: 3x224x224 -> 64x54x54 (s4 p0 k11), SURVEY.md section 2.3
    x = torch.randn(2, 224, 224, 3)
    w = torch.randn(11, 11, 3, 64) * 0.01
    y = cpu_ref.conv2d(x, w, None, stride=4, pad=0)
    assert y.shape == (2, 54, 54, 64)
    # conv2: 64x27x27 -> 256x27x27 (s1 p2 k5)
    x = torch.randn(2, 27, 27, 64)
    w = torch.randn(5, 5, 64, 256) * 0.01
    y = cpu_ref.conv2d(x, w, None, stride=1, pad=2)
    assert y.shape == (2, 27, 27, 256)


def test_conv2d_identity_kernel():
    # 1x1 identity conv reproduces input
    x = torch.randn(1, 5, 5, 3)
    w = torch.zeros(1, 1, 3, 3)
    for c in range(3):
        w[0, 0, c, c] = 1.0
    y = cpu_ref.conv2d(x, w, None, 1, 0)
    assert torch.allclose(y, x, atol=1e-6)


def test_conv2d_relu_bias():
    x = torch.randn(1, 8, 8, 4)
    w = torch.randn(3, 3, 4, 8) * 0.1
    b = torch.randn(8)
    y = cpu_ref.conv2d(x, w, b, 1, 1, relu=True)
    assert (y >= 0).all()


def test_maxpool_asymmetric_pad():
    # pool1: 54 -> 27 with pad (0,1)
    x = torch.randn(2, 54, 54, 8)
    y = cpu_ref.maxpool(x, 3, 2, (0, 1))
    assert y.shape == (2, 27, 27, 8)
    # pool2: 27 -> 13 no pad
    y = cpu_ref.maxpool(torch.randn(1, 27, 27, 8), 3, 2, (0, 0))
    assert y.shape == (1, 13, 13, 8)
    # padding must never win the max (uses -inf)
    x = -torch.ones(1, 4, 4, 1)
    y = cpu_ref.maxpool(x, 3, 2, (0, 1))
    assert (y == -1).all()


def test_lrn_matches_manual():
    torch.manual_seed(0)
    x = torch.randn(2, 4, 4, 8)
    y = cpu_ref.lrn(x, size=5, alpha=1e-4, beta=0.75, k=2.0)
    # manual cross-channel window sum at one position
    n, h, w, c = 1, 2, 3, 4
    lo, hi = max(0, c - 2), min(8, c + 3)
    s = (x[n, h, w, lo:hi] ** 2).sum()
    expect = x[n, h, w, c] * (2.0 + 1e-4 / 5 * s) ** -0.75
    assert torch.allclose(y[n, h, w, c], expect, atol=1e-5)


def test_softmax_xent_loss_and_grad():
    torch.manual_seed(0)
    logits = torch.randn(4, 10)
    labels = torch.tensor([1, 3, 0, 9])
    loss, dlogits = cpu_ref.softmax_xent(logits, labels)
    # vs torch composite
    ref = torch.nn.functional.cross_entropy(logits, labels, reduction="none")
    assert torch.allclose(loss, ref, atol=1e-6)
    # grad via autograd of summed loss
    lg = logits.clone().requires_grad_(True)
    torch.nn.functional.cross_entropy(lg, labels, reduction="sum").backward()
    assert torch.allclose(dlogits, lg.grad, atol=1e-6)


def test_sgd_momentum_semantics():
    w = torch.ones(10)
    g = torch.full((10,), 2.0)
    m = torch.zeros(10)
    cpu_ref.sgd_momentum(w, g, m, lr=0.1, momentum=0.9, grad_scale=0.5)
    assert torch.allclose(m, torch.full((10,), 1.0))
    assert torch.allclose(w, torch.full((10,), 0.9))
    cpu_ref.sgd_momentum(w, g, m, lr=0.1, momentum=0.9, grad_scale=0.5)
    assert torch.allclose(m, torch.full((10,), 1.9))
    assert torch.allclose(w, torch.full((10,), 0.71))


def test_batchnorm_normalizes():
    torch.manual_seed(0)
    x = torch.randn(8, 4, 4, 16) * 3 + 1
    y = cpu_ref.batchnorm(x, torch.ones(16), torch.zeros(16))
    assert torch.allclose(y.mean(dim=(0, 1, 2)), torch.zeros(16), atol=1e-5)
    assert torch.allclose(y.var(dim=(0, 1, 2), unbiased=False),
                          torch.ones(16), atol=1e-4)


def test_gradcheck_small_conv():
    # T-grad: finite differences via torch.autograd.gradcheck, fp64
    x = torch.randn(1, 6, 6, 2, dtype=torch.double, requires_grad=True)
    w = torch.randn(3, 3, 2, 3, dtype=torch.double, requires_grad=True)
    b = torch.randn(3, dtype=torch.double, requires_grad=True)
    assert torch.autograd.gradcheck(
        lambda x, w, b: cpu_ref.conv2d(x, w, b, 2, 1, relu=False),
        (x, w, b), eps=1e-6, atol=1e-4)


def test_gradcheck_lrn():
    x = torch.randn(1, 3, 3, 6, dtype=torch.double, requires_grad=True)
    assert torch.autograd.gradcheck(
        lambda x: cpu_ref.lrn(x, 5, 1e-2, 0.75, 2.0), (x,),
        eps=1e-6, atol=1e-4)
