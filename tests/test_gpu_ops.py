"""T-unit on MI355X: every HIP kernel vs the CPU fp32 reference
(SURVEY.md section 4.2). Inputs are bf16-rounded before the reference runs,
so tolerances cover only accumulation-order + bf16 output rounding.

Transpose-detection rule (cdna_hip_programming.md section 5.4 rule 16):
all operands are random (asymmetric) tensors.
"""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from distributed_vgg_f_amd.ops import native
    ext = native.ext()
from distributed_vgg_f_amd.ops import cpu_ref

DEV = "cuda:0"


def rnd(*shape, scale=1.0, seed=None):
    if seed is not None:
        torch.manual_seed(seed)
    return (torch.randn(*shape) * scale).bfloat16()


def check(got, ref, tol=0.05, name=""):
    got = got.float().cpu()
    ref = ref.float()
    assert got.shape == ref.shape, f"{name}: {got.shape} vs {ref.shape}"
    scale = max(float(ref.abs().max()), 1e-3)
    err = float((got - ref).abs().max())
    assert err <= tol * scale, f"{name}: max err {err:.4g} vs scale {scale:.4g}"


# --------------------------------------------------------------- linear
def test_linear_fwd_asymmetric():
    torch.manual_seed(0)
    x, w, b = rnd(256, 512), rnd(512, 384, scale=0.05), rnd(384)
    y = ext.linear_fwd(x.to(DEV), w.to(DEV), b.to(DEV), True)
    ref = cpu_ref.linear(x.float(), w.float(), b.float(), relu=True)
    check(y, ref, 0.05, "linear_fwd")


def test_linear_fwd_fc6_shape_splitk():
    torch.manual_seed(1)
    x, w, b = rnd(256, 9216, scale=0.1), rnd(9216, 4096, scale=0.02), rnd(4096)
    y = ext.linear_fwd(x.to(DEV), w.to(DEV), b.to(DEV), False)
    ref = cpu_ref.linear(x.float(), w.float(), b.float())
    check(y, ref, 0.05, "fc6")


def test_splitk_scratch_rezero_contract():
    """The split-K atomic accumulators are CACHED per (site, shape) and
    re-zeroed by their finalize kernel (gemm_ops.hip scratch_c32), so a
    broken re-zero shows up as the SECOND call of the same shape
    accumulating on top of the first. Run the split-K fc6 shape twice
    with different inputs and check the second result independently."""
    torch.manual_seed(7)
    w, b = rnd(9216, 4096, scale=0.02), rnd(4096)
    wd, bd = w.to(DEV), b.to(DEV)
    for trial in range(2):
        x = rnd(256, 9216, scale=0.1, seed=100 + trial)
        y = ext.linear_fwd(x.to(DEV), wd, bd, False)
        check(y, cpu_ref.linear(x.float(), w.float(), b.float()), 0.05,
              f"scratch-rezero trial {trial}")


def test_linear_fwd_fc8_nonmult128():
    torch.manual_seed(2)
    x, w, b = rnd(12, 4096, scale=0.1), rnd(4096, 1000, scale=0.02), rnd(1000)
    y = ext.linear_fwd(x.to(DEV), w.to(DEV), b.to(DEV), False)
    check(y, cpu_ref.linear(x.float(), w.float(), b.float()), 0.05, "fc8")


def test_linear_dgrad_wgrad():
    torch.manual_seed(3)
    x, w = rnd(256, 1024, scale=0.1), rnd(1024, 512, scale=0.05)
    dy = rnd(256, 512, scale=0.1)
    dx = ext.linear_dgrad(dy.to(DEV), w.to(DEV))
    dw = ext.linear_wgrad(x.to(DEV), dy.to(DEV))
    check(dx, dy.float() @ w.float().t(), 0.05, "linear_dgrad")
    check(dw, x.float().t() @ dy.float(), 0.05, "linear_wgrad")


# ----------------------------------------------------------------- conv
CONV_CASES = [
    # (N, H, W, Cin, Cout, k, stride, pad, name)
    (2, 27, 27, 64, 256, 5, 1, 2, "conv2"),
    (2, 13, 13, 256, 256, 3, 1, 1, "conv3"),
    (2, 224, 224, 3, 64, 11, 4, 0, "conv1-im2col"),
    (2, 16, 16, 64, 128, 1, 1, 0, "conv1x1"),
    (2, 16, 16, 64, 128, 3, 2, 1, "conv3x3s2"),
    (2, 16, 16, 64, 128, 1, 2, 0, "conv1x1s2"),
    # tap-fused 3x3 wgrad coverage (wgrad_fused.h): CT=32 whole-segment,
    # OWp2 over-pad (33->64), CT=8 with padded k-rows, CT=16 pad=0 + co64
    (2, 28, 28, 64, 64, 3, 1, 1, "conv3x3-ow28"),
    (2, 33, 33, 64, 128, 3, 1, 1, "conv3x3-ow33pad64"),
    (2, 7, 7, 128, 128, 3, 1, 1, "conv3x3-ow7"),
    (2, 14, 14, 128, 64, 3, 1, 0, "conv3x3-p0-co64"),
    # phase-plane first layer with PC % 8 != 0 (ResNet stem: PC 12 -> 16)
    (2, 112, 112, 3, 64, 7, 2, 3, "stem-phase"),
]


@pytest.mark.parametrize("N,H,W,Ci,Co,k,s,p,name", CONV_CASES)
def test_conv_fwd(N, H, W, Ci, Co, k, s, p, name):
    torch.manual_seed(10)
    x = rnd(N, H, W, Ci, scale=0.5)
    w = rnd(k, k, Ci, Co, scale=1.0 / math.sqrt(k * k * Ci))
    b = rnd(Co, scale=0.1)
    y = ext.conv2d_fwd(x.to(DEV), w.to(DEV), b.to(DEV), s, p, True)
    ref = cpu_ref.conv2d(x.float(), w.float(), b.float(), s, p, relu=True)
    check(y, ref, 0.05, f"conv_fwd/{name}")


@pytest.mark.parametrize("N,H,W,Ci,Co,k,s,p,name",
                         [c for c in CONV_CASES if c[3] % 8 == 0])
def test_conv_dgrad(N, H, W, Ci, Co, k, s, p, name):
    torch.manual_seed(11)
    w = rnd(k, k, Ci, Co, scale=1.0 / math.sqrt(k * k * Ci))
    OH = (H + 2 * p - k) // s + 1
    dy = rnd(N, OH, OH, Co, scale=0.5)
    dx = ext.conv2d_dgrad(dy.to(DEV), w.to(DEV), H, W, s, p)
    # reference via autograd on the CPU composition
    x32 = torch.zeros(N, H, W, Ci, requires_grad=True)
    y = cpu_ref.conv2d(x32, w.float(), None, s, p)
    y.backward(dy.float())
    check(dx, x32.grad, 0.05, f"conv_dgrad/{name}")


@pytest.mark.parametrize("N,H,W,Ci,Co,k,s,p,name", CONV_CASES)
def test_conv_wgrad(N, H, W, Ci, Co, k, s, p, name):
    torch.manual_seed(12)
    x = rnd(N, H, W, Ci, scale=0.5)
    OH = (H + 2 * p - k) // s + 1
    dy = rnd(N, OH, OH, Co, scale=0.5)
    dw = ext.conv2d_wgrad(dy.to(DEV), x.to(DEV), k, k, s, p)
    w32 = torch.zeros(k, k, Ci, Co, requires_grad=True)
    y = cpu_ref.conv2d(x.float(), w32, None, s, p)
    y.backward(dy.float())
    check(dw, w32.grad, 0.06, f"conv_wgrad/{name}")


# ------------------------------------------------------------- LRN/pool
def test_lrn_fwd_bwd():
    torch.manual_seed(20)
    x = rnd(2, 9, 9, 64, scale=2.0)
    y = ext.lrn_fwd(x.to(DEV), 5, 1e-4, 0.75, 2.0)
    ref = cpu_ref.lrn(x.float(), 5, 1e-4, 0.75, 2.0)
    check(y, ref, 0.03, "lrn_fwd")
    dy = rnd(2, 9, 9, 64, seed=21)
    dx = ext.lrn_bwd(dy.to(DEV), x.to(DEV), 5, 1e-4, 0.75, 2.0)
    x32 = x.float().requires_grad_(True)
    cpu_ref.lrn(x32, 5, 1e-4, 0.75, 2.0).backward(dy.float())
    check(dx, x32.grad, 0.05, "lrn_bwd")


def test_lrn_large_alpha():
    # make the normalizer actually bite (s >> k) to exercise the poly path
    torch.manual_seed(22)
    x = rnd(1, 5, 5, 128, scale=4.0)
    y = ext.lrn_fwd(x.to(DEV), 5, 0.5, 0.75, 1.0)
    check(y, cpu_ref.lrn(x.float(), 5, 0.5, 0.75, 1.0), 0.04, "lrn_bite")


def test_maxpool_fwd_bwd_asympad():
    torch.manual_seed(23)
    x = rnd(2, 54, 54, 64)
    y, idx = ext.maxpool_fwd(x.to(DEV), 3, 2, 0, 1)
    ref = cpu_ref.maxpool(x.float(), 3, 2, (0, 1))
    check(y, ref, 0.01, "maxpool_fwd")
    dy = rnd(2, 27, 27, 64, seed=24)
    dx = ext.maxpool_bwd(dy.to(DEV), idx, 54, 54, 3, 2, 0, 1)
    x32 = x.float().requires_grad_(True)
    cpu_ref.maxpool(x32, 3, 2, (0, 1)).backward(dy.float())
    check(dx, x32.grad, 0.01, "maxpool_bwd")


def test_maxpool_nopad():
    torch.manual_seed(25)
    x = rnd(2, 27, 27, 32)
    y, idx = ext.maxpool_fwd(x.to(DEV), 3, 2, 0, 0)
    check(y, cpu_ref.maxpool(x.float(), 3, 2, (0, 0)), 0.01, "maxpool_np")


# ------------------------------------------------------- softmax / misc
def test_softmax_xent():
    torch.manual_seed(30)
    logits = rnd(64, 1000, scale=2.0)
    labels = torch.randint(0, 1000, (64,))
    loss, dlogits = ext.softmax_xent_fwd(logits.to(DEV), labels.to(DEV))
    ref_loss, ref_d = cpu_ref.softmax_xent(logits.float(), labels)
    check(loss, ref_loss, 0.01, "xent_loss")
    check(dlogits, ref_d, 0.02, "xent_dlogits")


def test_relu_bwd_add_relu_bias_grad():
    torch.manual_seed(31)
    y = rnd(1000), rnd(1000)
    dy, yv = rnd(4, 250, seed=32), rnd(4, 250, seed=33)
    got = ext.relu_bwd(dy.to(DEV), yv.to(DEV))
    ref = dy.float() * (yv.float() > 0)
    check(got, ref, 0.01, "relu_bwd")
    a, b = rnd(4, 256, seed=34), rnd(4, 256, seed=35)
    check(ext.add_relu_fwd(a.to(DEV), b.to(DEV)),
          torch.relu(a.float() + b.float()), 0.02, "add_relu")
    dyb = rnd(512, 64, seed=36)
    check(ext.bias_grad(dyb.to(DEV)), dyb.float().sum(0), 0.02, "bias_grad")


def test_dropout_stats_and_bwd():
    torch.manual_seed(37)
    x = torch.ones(100000).bfloat16().to(DEV)
    y, mask = ext.dropout_fwd(x, 0.3, 1234, 7)
    keep = float(mask.float().mean())
    assert abs(keep - 0.7) < 0.02, keep
    # kept elements scaled by 1/(1-p)
    got = y.float()
    assert torch.allclose(got[mask.bool()],
                          torch.full_like(got[mask.bool()], 1.0 / 0.7),
                          atol=0.01)
    dy = torch.ones_like(x)
    dx = ext.dropout_bwd(dy, mask, 0.3)
    assert torch.allclose(dx.float()[mask.bool()],
                          torch.full_like(got[mask.bool()], 1.0 / 0.7),
                          atol=0.01)
    assert float(dx.float()[~mask.bool()].abs().sum()) == 0.0


def test_sgd_step_matches_cpu():
    torch.manual_seed(38)
    n = 4096
    w32 = torch.randn(n)
    g = torch.randn(n).bfloat16()
    m32 = torch.randn(n)
    w32_d = w32.clone().to(DEV)
    wc_d = w32.bfloat16().to(DEV)
    m32_d = m32.clone().to(DEV)
    ext.sgd_step(w32_d, wc_d, g.to(DEV), m32_d, 0.05, 0.9, 1e-4, 0.25)
    wr, mr = w32.clone(), m32.clone()
    cpu_ref.sgd_momentum(wr, g.float(), mr, 0.05, 0.9, 1e-4, 0.25)
    check(w32_d.bfloat16(), wr, 0.01, "sgd_w")
    check(m32_d.bfloat16(), mr, 0.01, "sgd_m")
    check(wc_d, wr.bfloat16().float(), 0.01, "sgd_wc")


def test_bn_fwd_bwd():
    torch.manual_seed(39)
    x = rnd(8, 7, 7, 64, scale=2.0)
    g, b = rnd(64, seed=40), rnd(64, seed=41)
    y, mean, rstd = ext.bn_fwd(x.to(DEV), g.to(DEV), b.to(DEV), 1e-5, False,
                               None, None, 0.9)
    ref = cpu_ref.batchnorm(x.float(), g.float(), b.float())
    check(y, ref, 0.05, "bn_fwd")
    dy = rnd(8, 7, 7, 64, seed=42)
    dx, dgamma, dbeta = ext.bn_bwd(dy.to(DEV), x.to(DEV), g.to(DEV), mean,
                                   rstd, y, False)
    x32 = x.float().requires_grad_(True)
    g32 = g.float().requires_grad_(True)
    b32 = b.float().requires_grad_(True)
    cpu_ref.batchnorm(x32, g32, b32).backward(dy.float())
    check(dx, x32.grad, 0.06, "bn_dx")
    check(dgamma, g32.grad, 0.06, "bn_dgamma")
    check(dbeta, b32.grad, 0.06, "bn_dbeta")


def test_conv2d_bwd_fused_matches_parts():
    torch.manual_seed(50)
    N, H, W, Ci, Co, k, s, p = 2, 13, 13, 256, 256, 3, 1, 1
    x = rnd(N, H, W, Ci, scale=0.5)
    w = rnd(k, k, Ci, Co, scale=1.0 / math.sqrt(k * k * Ci))
    b = rnd(Co, scale=0.1)
    xg, wg, bg = x.to(DEV), w.to(DEV), b.to(DEV)
    y = ext.conv2d_fwd(xg, wg, bg, s, p, True)
    dy = rnd(N, H, W, Co, seed=51).to(DEV)
    dx, dw, db = ext.conv2d_bwd(dy, y, xg, wg, s, p, True, True, True)
    # reference: mask then separate ops
    dym = ext.relu_bwd(dy, y)
    dx2 = ext.conv2d_dgrad(dym, wg, H, W, s, p)
    dw2 = ext.conv2d_wgrad(dym, xg, k, k, s, p)
    db2 = ext.bias_grad(dym)
    check(dx, dx2.float().cpu(), 0.01, "fused_dx")
    check(dw, dw2.float().cpu(), 0.02, "fused_dw")
    check(db, db2.float().cpu(), 0.02, "fused_db")


def test_bn_bwd_fused_relu():
    torch.manual_seed(43)
    x = rnd(4, 5, 5, 32, scale=2.0)
    g, b = rnd(32, seed=44), rnd(32, seed=45)
    y, mean, rstd = ext.bn_fwd(x.to(DEV), g.to(DEV), b.to(DEV), 1e-5, True,
                               None, None, 0.9)
    dy = rnd(4, 5, 5, 32, seed=46)
    dx, dgamma, dbeta = ext.bn_bwd(dy.to(DEV), x.to(DEV), g.to(DEV), mean,
                                   rstd, y, True)
    x32 = x.float().requires_grad_(True)
    g32 = g.float().requires_grad_(True)
    b32 = b.float().requires_grad_(True)
    cpu_ref.batchnorm(x32, g32, b32, relu=True).backward(dy.float())
    check(dx, x32.grad, 0.06, "bnrelu_dx")
    check(dgamma, g32.grad, 0.06, "bnrelu_dgamma")
    check(dbeta, b32.grad, 0.06, "bnrelu_dbeta")


def test_bn_running_stats_update():
    torch.manual_seed(60)
    x = (torch.randn(16, 3, 3, 32) * 2 + 1).bfloat16().to(DEV)
    g = torch.ones(32).bfloat16().to(DEV)
    b = torch.zeros(32).bfloat16().to(DEV)
    rm = torch.zeros(32, device=DEV)
    rv = torch.ones(32, device=DEV)
    ext.bn_fwd(x, g, b, 1e-5, False, rm, rv, 0.5)
    m_ref = x.float().mean(dim=(0, 1, 2))
    v_ref = x.float().var(dim=(0, 1, 2), unbiased=False)
    check(rm.bfloat16(), (0.5 * m_ref).cpu(), 0.03, "bn_rm")
    check(rv.bfloat16(), (0.5 + 0.5 * v_ref).cpu(), 0.03, "bn_rv")


def test_wgrad_splitk_stress_repeat():
    """split-K atomic accumulation under varying block
    execution order — repeated runs must all stay within tolerance of the
    reference (catches order-dependent bugs in the k-slicing)."""
    torch.manual_seed(70)
    N, H, W, Ci, Co, k, s, p = 2, 13, 13, 256, 256, 3, 1, 1
    x = rnd(N, H, W, Ci, scale=0.5)
    dy = rnd(N, H, W, Co, scale=0.5)
    w32 = torch.zeros(k, k, Ci, Co, requires_grad=True)
    cpu_ref.conv2d(x.float(), w32, None, s, p).backward(dy.float())
    xg, dyg = x.to(DEV), dy.to(DEV)
    for rep in range(5):
        dw = ext.conv2d_wgrad(dyg, xg, k, k, s, p)
        check(dw, w32.grad, 0.06, f"wgrad_stress_rep{rep}")


# ---- kernel-variant regression guards: the routing env knobs select
# between gemm_glds_kernel (2-buffer), gemm_glds3_kernel (counted ring,
# default) and gemm8_kernel (deep tile); every variant must stay
# numerically correct even while the default routing evolves.
VARIANT_ENVS = [
    {"DVGG_GLDS3": "0"},                                   # legacy 2-buffer
    {"DVGG_GLDS3": "1"},                                   # counted ring
    {"DVGG_GEMM8": "1", "DVGG_G8_MINTILES": "0",
     "DVGG_G8_MINCONTIG": "0"},                            # deep tile, forced
    {"DVGG_GEMM8": "0"},                                   # deep tile off
    {"DVGG_GEMM8N": "1", "DVGG_G8_MINTILES": "0",
     "DVGG_G8_MINCONTIG": "0"},                            # BN=128 deep tile
]


@pytest.mark.parametrize("env", VARIANT_ENVS,
                         ids=["glds2", "glds3", "gemm8-on", "gemm8-off",
                              "gemm8n"])
def test_conv_fwd_dgrad_variants(env, monkeypatch):
    import os
    for k_, v_ in env.items():
        monkeypatch.setenv(k_, v_)
    torch.manual_seed(21)
    if env.get("DVGG_GEMM8N") == "1":
        N, H, W, Ci, Co, k, s, p = 2, 16, 16, 128, 128, 3, 1, 1
    else:
        N, H, W, Ci, Co, k, s, p = 2, 13, 13, 256, 256, 3, 1, 1
    x = rnd(N, H, W, Ci, scale=0.5)
    w = rnd(k, k, Ci, Co, scale=1.0 / math.sqrt(k * k * Ci))
    b = rnd(Co, scale=0.1)
    y = ext.conv2d_fwd(x.to(DEV), w.to(DEV), b.to(DEV), s, p, True)
    ref = cpu_ref.conv2d(x.float(), w.float(), b.float(), s, p, relu=True)
    check(y, ref, 0.05, "variant_fwd")
    dy = rnd(N, H, W, Co, scale=0.5)
    dx = ext.conv2d_dgrad(dy.to(DEV), w.to(DEV), H, W, s, p)
    xr = x.float().requires_grad_(True)
    yr = cpu_ref.conv2d(xr, w.float(), None, s, p)
    yr.backward(dy.float())
    check(dx, xr.grad, 0.05, "variant_dgrad")


@pytest.mark.parametrize("shape", [
    (2, 13, 13, 256, 256, 1),   # CT=16, WAVES_N=4, pad row (OH=13 -> 14)
    (2, 28, 28, 64, 64, 1),     # CT=32 single segment, WAVES_N=2
    (2, 33, 33, 64, 128, 1),    # OWp2 over-pad 40 -> 64, 2 segments
    (2, 7, 7, 128, 128, 1),     # CT=8, R=4, padded k-rows
    (2, 14, 14, 128, 64, 0),    # CT=16, pad=0, Cout=64
], ids=["ct16", "ct32", "ct32pad", "ct8", "ct16p0"])
def test_wgrad_fused_vs_classic(shape, monkeypatch):
    """The tap-fused 3x3 wgrad (wgrad_fused.h) must agree with both the
    classic implicit-GEMM path and the CPU reference on every CT regime."""
    N, H, W, Ci, Co, p = shape
    k, s = 3, 1
    _wgrad_fused_check(N, H, W, Ci, Co, p, k, s, monkeypatch,
                       "DVGG_WGRAD_FUSED")


@pytest.mark.parametrize("shape", [
    (2, 27, 27, 64, 256, 2),    # VGG-F conv2 geometry (CT=32)
    (2, 13, 13, 64, 64, 2),     # CT=16 5x5
], ids=["k5ct32", "k5ct16"])
def test_wgrad_fused5_vs_classic(shape, monkeypatch):
    """5x5 instantiation (WAVES_M=4, 25 accumulator sets)."""
    N, H, W, Ci, Co, p = shape
    k, s = 5, 1
    _wgrad_fused_check(N, H, W, Ci, Co, p, k, s, monkeypatch,
                       "DVGG_WGRAD_FUSED5")


def _wgrad_fused_check(N, H, W, Ci, Co, p, k, s, monkeypatch, env_name):
    torch.manual_seed(81)
    x = rnd(N, H, W, Ci, scale=0.5)
    oh = (H + 2 * p - k) // s + 1
    dy = rnd(N, oh, oh, Co, scale=0.5)
    w32 = torch.zeros(k, k, Ci, Co, requires_grad=True)
    cpu_ref.conv2d(x.float(), w32, None, s, p).backward(dy.float())
    xg, dyg = x.to(DEV), dy.to(DEV)
    monkeypatch.setenv(env_name, "1")
    dw_f = ext.conv2d_wgrad(dyg, xg, k, k, s, p)
    monkeypatch.setenv(env_name, "0")
    dw_c = ext.conv2d_wgrad(dyg, xg, k, k, s, p)
    check(dw_f, w32.grad, 0.06, "wgrad_fused_vs_cpu")
    check(dw_c, w32.grad, 0.06, "wgrad_classic_vs_cpu")
    check(dw_f, dw_c.float().cpu(), 0.02, "wgrad_fused_vs_classic")


def test_cached_wd_refresh_1x1s2():
    """ResNet downsample (1x1 stride-2) with the weight-form cache on:
    contiguous() of the [Cin,1,Cout] permute aliases w, so the cache must
    clone at creation and refresh across generations without aliasing."""
    torch.manual_seed(91)
    N, H, W, Ci, Co, k, s, p = 2, 16, 16, 64, 128, 1, 2, 0
    x = rnd(N, H, W, Ci, scale=0.5).to(DEV)
    w = rnd(k, k, Ci, Co, scale=0.1).to(DEV)
    oh = (H + 2 * p - k) // s + 1
    dy = rnd(N, oh, oh, Co, scale=0.5).to(DEV)
    y = ext.conv2d_fwd(x, w, None, s, p, False, True)
    dx1, dw1, _ = ext.conv2d_bwd(dy, y, x, w, s, p, False, True, False,
                                 None, None, True)
    ext.bump_weight_gen()
    w.copy_(rnd(k, k, Ci, Co, scale=0.1))   # new generation's weights
    dx2, dw2, _ = ext.conv2d_bwd(dy, y, x, w, s, p, False, True, False,
                                 None, None, True)
    # dgrad must reflect the NEW weights (stale cache would reuse old wD)
    dx_ref = ext.conv2d_dgrad(dy, w, H, W, s, p)
    check(dx2, dx_ref.float().cpu(), 1e-3, "cached_wd_refresh")


def test_bn_add_relu_fused():
    """K17+K18 fusion: z = relu(bn(x) + res) in one apply pass must match
    the composed CPU reference in forward AND all gradients."""
    from distributed_vgg_f_amd import ops
    torch.manual_seed(93)
    N, H, W, C = 4, 8, 8, 64
    x = rnd(N, H, W, C, scale=0.7)
    res = rnd(N, H, W, C, scale=0.7)
    g = rnd(C, scale=0.3) + 1.0
    b = rnd(C, scale=0.3)

    xg = x.to(DEV).requires_grad_(True)
    rg = res.to(DEV).requires_grad_(True)
    gg = g.to(DEV).requires_grad_(True)
    bg = b.to(DEV).requires_grad_(True)
    z = ops.batchnorm(xg, gg, bg, 1e-5, False, res=rg)
    z.float().square().sum().backward()

    xc = x.float().requires_grad_(True)
    rc = res.float().requires_grad_(True)
    gc = g.float().requires_grad_(True)
    bc = b.float().requires_grad_(True)
    zc = ops.batchnorm(xc, gc, bc, 1e-5, False, res=rc)
    zc.square().sum().backward()

    check(z, zc.detach(), 0.05, "bnaddrelu_fwd")
    check(xg.grad, xc.grad, 0.08, "bnaddrelu_dx")
    check(rg.grad, rc.grad, 0.05, "bnaddrelu_dres")
    check(gg.grad, gc.grad, 0.05, "bnaddrelu_dgamma")
    check(bg.grad, bc.grad, 0.05, "bnaddrelu_dbeta")
