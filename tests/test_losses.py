import math

import numpy as np
import torch
import pytest

from dsin_amd.losses import (Distortions, bitcost_to_bpp, get_loss,
                             multiscale_ssim, multiscale_ssim_np)


def test_bpp(small_ae_config):
    x = torch.zeros(2, 3, 8, 8)
    bc = torch.ones(2, 32, 1, 1)
    bpp = bitcost_to_bpp(bc, x)
    assert torch.allclose(bpp, torch.tensor(64.0 / (2 * 64)))


def test_distortion_int_cast(small_ae_config):
    x = torch.full((1, 3, 4, 4), 10.0)
    y = torch.full((1, 3, 4, 4), 10.6)
    # minimize mae at train: NO cast -> mae 0.6
    d = Distortions(small_ae_config, x, y, is_training=True)
    assert torch.allclose(d.mae, torch.tensor(0.6), atol=1e-6)
    # mse is not the minimized metric -> int cast (trunc) -> error 0
    assert torch.allclose(d.mse, torch.tensor(0.0))
    # at eval everything is int cast
    d2 = Distortions(small_ae_config, x, y, is_training=False)
    assert torch.allclose(d2.mae, torch.tensor(0.0))


def test_get_loss_hinge(small_ae_config):
    d_loss = torch.tensor(2.0)
    reg = torch.tensor(0.5)
    heat = torch.ones(1, 32, 2, 2)
    # H below target -> pc_loss 0
    bc_low = torch.full((1, 32, 2, 2), 0.01)
    total, H_real, pc_loss = get_loss(small_ae_config, d_loss, bc_low, heat, reg)
    assert float(pc_loss) == 0.0
    assert torch.allclose(total, d_loss + reg)
    # H above target -> beta * (H_soft - target)
    bc_high = torch.full((1, 32, 2, 2), 1.0)
    total, H_real, pc_loss = get_loss(small_ae_config, d_loss, bc_high, heat, reg)
    expect = 500.0 * (1.0 - small_ae_config.H_target)
    assert torch.allclose(pc_loss, torch.tensor(expect), rtol=1e-5)


def test_get_loss_hmask_halves(small_ae_config):
    """With heatmap == 0, H_soft = H_real / 2."""
    d_loss = torch.tensor(0.0)
    reg = torch.tensor(0.0)
    bc = torch.full((1, 32, 2, 2), 1.0)
    heat0 = torch.zeros_like(bc)
    _, _, pc0 = get_loss(small_ae_config, d_loss, bc, heat0, reg)
    expect = 500.0 * (0.5 - small_ae_config.H_target)
    assert torch.allclose(pc0, torch.tensor(expect), rtol=1e-5)


def test_psnr_formula(small_ae_config):
    x = torch.zeros(1, 3, 8, 8)
    y = torch.full((1, 3, 8, 8), 16.0)
    d = Distortions(small_ae_config, x, y, is_training=False)
    expect = 10 * math.log10(255.0 ** 2 / 256.0)
    assert torch.allclose(d.psnr, torch.tensor(expect), rtol=1e-5)


def test_msssim_identical_is_one():
    torch.manual_seed(0)
    img = torch.rand(1, 3, 176, 176) * 255
    v = multiscale_ssim(img, img)
    assert abs(float(v) - 1.0) < 1e-5


def test_msssim_torch_vs_numpy_oracle():
    torch.manual_seed(1)
    base = torch.rand(1, 3, 23, 23)
    img1 = torch.nn.functional.interpolate(base, size=(176, 176), mode="bilinear") * 255
    img2 = (img1 + torch.randn_like(img1) * 10).clamp(0, 255)
    v_t = float(multiscale_ssim(img1, img2))
    nhwc = lambda t: t[0].permute(1, 2, 0).numpy()[None]
    v_np = multiscale_ssim_np(nhwc(img1), nhwc(img2))
    # same construction up to the downsample border convention
    assert abs(v_t - v_np) < 5e-3, (v_t, v_np)


def test_msssim_decreases_with_noise():
    torch.manual_seed(2)
    img = torch.rand(1, 3, 176, 176) * 255
    v1 = float(multiscale_ssim(img, (img + torch.randn_like(img) * 5).clamp(0, 255)))
    v2 = float(multiscale_ssim(img, (img + torch.randn_like(img) * 40).clamp(0, 255)))
    assert v1 > v2


def test_msssim_differentiable():
    torch.manual_seed(3)
    img = torch.rand(1, 3, 176, 176, requires_grad=True)
    v = multiscale_ssim(img * 255, (img * 255).detach() + 5)
    v.backward()
    assert img.grad is not None and torch.isfinite(img.grad).all()
