"""Gather-GEMM conv kernel numerics vs torch fp32 oracles (MI355X)."""

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

from dsin_amd.ops import conv as dconv


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    return torch.device("cuda:0")


def _cmp_conv(dev, Ci, Co, H, W, k, stride=1, padding=0, dilation=1,
              bias=False, act=0, atol=0.15, seed=0, bias_shift=0.0):
    torch.manual_seed(seed)
    x = torch.randn(2, Ci, H, W, device=dev)
    w = torch.randn(Co, Ci, k, k, device=dev) / (k * Ci ** 0.5)
    b = (torch.randn(Co, device=dev) + bias_shift) if bias else None
    x1 = x.clone().requires_grad_(True)
    w1 = w.clone().requires_grad_(True)
    b1 = b.clone().requires_grad_(True) if bias else None
    y = dconv.conv2d(x1, w1, b1, stride, padding, dilation, act)

    x2 = x.clone().requires_grad_(True)
    w2 = w.clone().requires_grad_(True)
    b2 = b.clone().requires_grad_(True) if bias else None
    yr = F.conv2d(x2, w2, b2, stride=stride, padding=padding, dilation=dilation)
    if act == 1:
        yr = torch.relu(yr)
    elif act == 2:
        yr = F.leaky_relu(yr, 0.2)

    assert y.shape == yr.shape, (y.shape, yr.shape)
    torch.testing.assert_close(y.float(), yr, rtol=0.05, atol=atol)

    g = torch.randn_like(yr)
    y.backward(g.to(y.dtype))
    yr.backward(g)
    torch.testing.assert_close(x1.grad.float(), x2.grad, rtol=0.08,
                               atol=atol * 3)
    # dw/db are sums over all pixels: scale the absolute tolerance with the
    # gradient magnitude (bf16 relative error accumulates on large sums)
    atol_w = max(atol * 3, 4e-3 * float(w2.grad.abs().max()))
    torch.testing.assert_close(w1.grad.float(), w2.grad, rtol=0.08,
                               atol=atol_w)
    if bias:
        atol_b = max(atol, 4e-3 * float(b2.grad.abs().max()))
        torch.testing.assert_close(b1.grad.float(), b2.grad, rtol=0.05,
                                   atol=atol_b)


def test_conv3x3_s1(dev):
    _cmp_conv(dev, 32, 48, 24, 40, 3, stride=1, padding=1)


def test_conv3x3_s1_128ch(dev):
    _cmp_conv(dev, 128, 128, 20, 48, 3, stride=1, padding=1, atol=0.4)


def test_conv5x5_s2(dev):
    _cmp_conv(dev, 3, 64, 64, 96, 5, stride=2, padding=2)


def test_conv5x5_s2_33(dev):
    _cmp_conv(dev, 64, 33, 40, 48, 5, stride=2, padding=2, atol=0.4)


def test_conv_dilated(dev):
    # act=0 for the gradient check: near-zero pre-activations make the
    # leaky-relu derivative branch (1 vs 0.2) flip under bf16 rounding,
    # which is a tolerance artifact, not a kernel bug. Activation-gradient
    # branches are covered deterministically below.
    _cmp_conv(dev, 6, 32, 40, 56, 3, stride=1, padding=4, dilation=4,
              bias=True, act=0)


def test_conv_dilated_large_rate(dev):
    _cmp_conv(dev, 32, 32, 48, 72, 3, stride=1, padding=16, dilation=16,
              bias=True, act=0)


def test_conv_lrelu_positive_branch(dev):
    # big positive bias -> every pre-activation > 0 -> lrelu' == 1 everywhere
    _cmp_conv(dev, 8, 16, 24, 32, 3, stride=1, padding=1, bias=True, act=2,
              bias_shift=30.0, atol=0.4)


def test_conv_lrelu_negative_branch(dev):
    # big negative bias -> every pre-activation < 0 -> lrelu' == 0.2
    _cmp_conv(dev, 8, 16, 24, 32, 3, stride=1, padding=1, bias=True, act=2,
              bias_shift=-30.0, atol=0.4)


def test_conv_relu_branches(dev):
    _cmp_conv(dev, 8, 16, 24, 32, 3, stride=1, padding=1, bias=True, act=1,
              bias_shift=30.0, atol=0.4)
    _cmp_conv(dev, 8, 16, 24, 32, 3, stride=1, padding=1, bias=True, act=1,
              bias_shift=-30.0, atol=0.4)


def test_conv1x1(dev):
    _cmp_conv(dev, 32, 3, 24, 40, 1, stride=1, padding=0, bias=True)


def _cmp_convT(dev, Ci, Co, H, W, k, stride=2, padding=None, seed=1,
               atol=0.15):
    if padding is None:
        padding = (k - 1) // 2
    torch.manual_seed(seed)
    x = torch.randn(1, Ci, H, W, device=dev)
    w = torch.randn(Ci, Co, k, k, device=dev) / (k * Ci ** 0.5)
    x1 = x.clone().requires_grad_(True)
    w1 = w.clone().requires_grad_(True)
    y = dconv.conv_transpose2d(x1, w1, None, stride, padding, stride - 1)

    x2 = x.clone().requires_grad_(True)
    w2 = w.clone().requires_grad_(True)
    yr = F.conv_transpose2d(x2, w2, None, stride=stride, padding=padding,
                            output_padding=stride - 1)
    assert y.shape == yr.shape, (y.shape, yr.shape)
    torch.testing.assert_close(y.float(), yr, rtol=0.05, atol=atol)
    g = torch.randn_like(yr)
    y.backward(g.to(y.dtype))
    yr.backward(g)
    torch.testing.assert_close(x1.grad.float(), x2.grad, rtol=0.08, atol=atol * 3)
    torch.testing.assert_close(w1.grad.float(), w2.grad, rtol=0.08, atol=atol * 3)


def test_convT3x3_s2(dev):
    _cmp_convT(dev, 32, 64, 8, 12, 3)


def test_convT5x5_s2(dev):
    _cmp_convT(dev, 64, 32, 16, 24, 5)


def test_convT5x5_s2_to3(dev):
    _cmp_convT(dev, 32, 3, 16, 24, 5)


def test_conv_grad_isolation_first_layer(dev):
    """x without requires_grad must skip bwd-data cleanly."""
    torch.manual_seed(2)
    x = torch.randn(1, 3, 32, 48, device=dev)
    w = torch.randn(16, 3, 5, 5, device=dev, requires_grad=True)
    y = dconv.conv2d(x, w, None, 2, 2, 1)
    y.sum().backward()
    assert w.grad is not None and torch.isfinite(w.grad).all()


def test_conv_stress_repeated(dev):
    """Repeated launches across the model's conv shapes: catches rare
    race/fault behavior that single-shot tests miss."""
    torch.manual_seed(0)
    shapes = [(32, 48, 24, 40, 3, 1, 1), (128, 128, 40, 48, 3, 1, 1),
              (3, 64, 64, 96, 5, 2, 2), (64, 33, 40, 48, 5, 2, 2)]
    tensors = [(torch.randn(1, ci, h, w, device=dev),
                torch.randn(co, ci, k, k, device=dev) * 0.05, st, p)
               for ci, co, h, w, k, st, p in shapes]
    for rep in range(50):
        for x, w, st, p in tensors:
            y = dconv.conv2d(x, w, None, st, p, 1)
    torch.cuda.synchronize()
    assert torch.isfinite(y.float()).all()


def test_conv3d_valid_vs_torch(dev):
    torch.manual_seed(3)
    x = torch.randn(1, 24, 34, 28, 36, device=dev)  # (B,Ci,D,H,W) padded-ish
    w = torch.randn(24, 24, 2, 3, 3, device=dev) * 0.05
    b = torch.randn(24, device=dev)
    x1 = x.clone().requires_grad_(True)
    w1 = w.clone().requires_grad_(True)
    b1 = b.clone().requires_grad_(True)
    # act=0 for the gradient check (bf16 ReLU-threshold flips near zero are
    # a tolerance artifact); the fused act branches are covered by the
    # shifted-bias 2D tests and the relu-forward check below
    y = dconv.conv3d_valid(x1, w1, b1, act=0)
    yrelu = dconv.conv3d_valid(x.clone(), w.clone(), b.clone(), act=1)
    torch.testing.assert_close(yrelu.float(),
                               torch.relu(F.conv3d(x, w, b)), rtol=0.05,
                               atol=0.2)
    x2 = x.clone().requires_grad_(True)
    w2 = w.clone().requires_grad_(True)
    b2 = b.clone().requires_grad_(True)
    yr = F.conv3d(x2, w2, b2)
    assert y.shape == yr.shape
    torch.testing.assert_close(y.float(), yr, rtol=0.05, atol=0.2)
    g = torch.randn_like(yr)
    y.backward(g.to(y.dtype))
    yr.backward(g)
    torch.testing.assert_close(x1.grad.float(), x2.grad, rtol=0.08, atol=0.5)
    atol_w = max(0.5, 4e-3 * float(w2.grad.abs().max()))
    torch.testing.assert_close(w1.grad.float(), w2.grad, rtol=0.08, atol=atol_w)


def test_probclass_gpu_matches_cpu(dev):
    """Whole entropy model: GPU gather-conv3d path vs the CPU torch oracle."""
    import os
    from dsin_amd import config as cm
    from dsin_amd.models.probclass import ProbClass
    here = os.path.dirname(os.path.abspath(__file__))
    pc_cfg, _ = cm.parse(os.path.join(here, "..", "run_configs", "pc_run_configs"))
    torch.manual_seed(0)
    pc = ProbClass(pc_cfg, num_centers=6)
    q = torch.randn(1, 32, 12, 20)
    sym = torch.randint(0, 6, (1, 32, 12, 20))
    bc_cpu = pc.bitcost(q, sym, torch.tensor(0.5))
    pc_g = pc.to(dev)
    bc_gpu = pc_g.bitcost(q.to(dev), sym.to(dev), torch.tensor(0.5, device=dev))
    torch.testing.assert_close(bc_gpu.cpu(), bc_cpu, rtol=0.05, atol=0.05)


def test_conv_fp8_path(dev):
    """fp8 (e4m3) MFMA conv path: forward within fp8 tolerance of fp32
    torch conv; gradients flow and are finite."""
    from dsin_amd.ops import conv as dconv
    dconv.set_compute_dtype("fp8")
    try:
        torch.manual_seed(0)
        x = torch.randn(1, 32, 24, 40, device=dev)
        w = torch.randn(48, 32, 3, 3, device=dev) / (3 * 32 ** 0.5)
        x1 = x.clone().requires_grad_(True)
        w1 = w.clone().requires_grad_(True)
        y = dconv.conv2d(x1, w1, None, 1, 1, 1)
        yr = F.conv2d(x, w, None, stride=1, padding=1)
        # e4m3 has a 3-bit mantissa: ~6% relative element error
        rel = (y.float() - yr).abs().max() / yr.abs().max()
        assert float(rel) < 0.15, float(rel)
        y.sum().backward()
        assert torch.isfinite(x1.grad.float()).all()
        assert torch.isfinite(w1.grad.float()).all()
        # transposed conv on the fp8 path too
        wt = torch.randn(32, 16, 3, 3, device=dev) * 0.05
        yt = dconv.conv_transpose2d(x.clone(), wt, None, 2, 1, 1)
        ytr = F.conv_transpose2d(x, wt, None, stride=2, padding=1,
                                 output_padding=1)
        relt = (yt.float() - ytr).abs().max() / ytr.abs().max()
        assert float(relt) < 0.15, float(relt)
    finally:
        dconv.set_compute_dtype("bf16")


@pytest.mark.gpu
def test_conv3x3_direct_path(dev):
    """Direct LDS-halo 3x3 kernel (stride 1, Ci%64==0) vs torch fp32:
    forward + input/weight grads, odd sizes (edge-tile masking), and the
    conv-transpose route (stuffed stride-1 3x3 with Co%64==0 backward)."""
    from dsin_amd.ops import conv as dconv
    torch.manual_seed(1)
    for Ci, Co, H, W in [(64, 48, 24, 40), (128, 128, 19, 29)]:
        x = torch.randn(2, Ci, H, W, device=dev)
        w = torch.randn(Co, Ci, 3, 3, device=dev) / (3 * Ci ** 0.5)
        x1 = x.clone().requires_grad_(True)
        w1 = w.clone().requires_grad_(True)
        y = dconv.conv2d(x1, w1, None, 1, 1, 1)
        x2 = x.clone().requires_grad_(True)
        w2 = w.clone().requires_grad_(True)
        yr = F.conv2d(x2, w2, None, stride=1, padding=1)
        assert torch.allclose(y.float(), yr, atol=0.05, rtol=0.05), \
            (Ci, Co, (y.float() - yr).abs().max().item())
        g = torch.randn_like(yr)
        y.backward(g.to(y.dtype))
        yr.backward(g)
        assert torch.allclose(x1.grad.float(), x2.grad, atol=0.08, rtol=0.08)
        dwmax = w2.grad.abs().max().item()
        assert torch.allclose(w1.grad.float(), w2.grad,
                              atol=0.02 * max(dwmax, 1.0), rtol=0.05)
    # conv-transpose: backward-data of the stuffed conv takes the direct
    # path when Co%64==0
    xt = torch.randn(1, 128, 10, 14, device=dev).requires_grad_(True)
    wt = torch.randn(128, 64, 3, 3, device=dev) * 0.05
    wt1 = wt.clone().requires_grad_(True)
    yt = dconv.conv_transpose2d(xt, wt1, None, 2, 1, 1)
    xr = xt.detach().clone().requires_grad_(True)
    wr = wt.clone().requires_grad_(True)
    ytr = F.conv_transpose2d(xr, wr, None, stride=2, padding=1,
                             output_padding=1)
    assert torch.allclose(yt.float(), ytr, atol=0.05, rtol=0.05)
    gt = torch.randn_like(ytr)
    yt.backward(gt.to(yt.dtype))
    ytr.backward(gt)
    assert torch.allclose(xt.grad.float(), xr.grad, atol=0.08, rtol=0.08)
