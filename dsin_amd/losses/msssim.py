"""Differentiable multi-scale SSIM (training-loss variant).

Mirror of the reference TF implementation (/root/reference/src/ms_ssim_imgcomp.py):
separable 1D Gaussian blur applied per channel with VALID boundary handling
(:16-43: no padding when the image is at least kernel-sized), per-scale
SSIM/CS means (:81-112), 2-tap box [1/2,1/2] separable downsample padded
REFLECT (0 before, 1 after) then ::2 decimation (:46-64,179-181), 5 scales
with the paper weights [0.0448, 0.2856, 0.3001, 0.2363, 0.1333] combined as
prod(cs[:4]**w[:4]) * ssim[4]**w[4] (:164-186). Inputs NCHW in 0..255.
"""

from __future__ import annotations

from typing import Tuple

import torch
import torch.nn.functional as F

_WEIGHTS = (0.0448, 0.2856, 0.3001, 0.2363, 0.1333)


def _gauss_kernel1d(sigma: float, size: int, device, dtype) -> torch.Tensor:
    n = size // 2
    x = torch.arange(-n, n + 1, device=device, dtype=dtype)
    g = torch.exp(-x * x / (2.0 * sigma * sigma))
    return g / g.abs().sum()


def _conv_valid(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """VALID conv used by the blur passes. On GPU the per-channel grouped
    conv is flattened to batch B*C with Ci=Co=1 and routed through the
    custom gather-GEMM kernel (keeps MIOpen out of the ms_ssim loss path)."""
    if x.is_cuda:
        from ..ops import hip_available
        from ..ops import conv as dconv
        if hip_available():
            b, c, h, wd = x.shape
            y = dconv.conv2d(x.reshape(b * c, 1, h, wd), w, None, 1, 0, 1)
            return y.reshape(b, c, y.shape[2], y.shape[3]).float()
    c = x.shape[1]
    return F.conv2d(x, w.expand(c, 1, w.shape[2], w.shape[3]), groups=c)


def _blur_sep(x: torch.Tensor, k1d: torch.Tensor) -> torch.Tensor:
    """Per-channel separable VALID conv, rows then cols (reference :31-42)."""
    x = _conv_valid(x, k1d.view(1, 1, 1, -1))
    return _conv_valid(x, k1d.view(1, 1, -1, 1))


def _ssim_scale(img1: torch.Tensor, img2: torch.Tensor, max_val: float,
                filter_size: int, filter_sigma: float, k1: float, k2: float
                ) -> Tuple[torch.Tensor, torch.Tensor]:
    _, _, h, w = img1.shape
    size = min(filter_size, h, w)
    sigma = size * filter_sigma / filter_size
    k = _gauss_kernel1d(sigma, size, img1.device, img1.dtype)
    mu1 = _blur_sep(img1, k)
    mu2 = _blur_sep(img2, k)
    s11 = _blur_sep(img1 * img1, k) - mu1 * mu1
    s22 = _blur_sep(img2 * img2, k) - mu2 * mu2
    s12 = _blur_sep(img1 * img2, k) - mu1 * mu2
    c1 = (k1 * max_val) ** 2
    c2 = (k2 * max_val) ** 2
    v1 = 2.0 * s12 + c2
    v2 = s11 + s22 + c2
    ssim = (((2.0 * mu1 * mu2 + c1) * v1) / ((mu1 * mu1 + mu2 * mu2 + c1) * v2)).mean()
    cs = (v1 / v2).mean()
    return ssim, cs


def _downsample2(x: torch.Tensor) -> torch.Tensor:
    """2-tap box filter, REFLECT pad (0 front, 1 back), separable, then ::2
    (reference kernel_blur at :46-64 with pad_w1=0, pad_w2=1)."""
    x = F.pad(x, (0, 1, 0, 1), mode="reflect")
    x = _conv_valid(x, x.new_full((1, 1, 1, 2), 0.5))
    x = _conv_valid(x, x.new_full((1, 1, 2, 1), 0.5))
    return x[:, :, ::2, ::2]


def multiscale_ssim(img1: torch.Tensor, img2: torch.Tensor, max_val: float = 255.0,
                    filter_size: int = 11, filter_sigma: float = 1.5,
                    k1: float = 0.01, k2: float = 0.03) -> torch.Tensor:
    assert img1.shape == img2.shape and img1.dim() == 4
    im1, im2 = img1, img2
    mssim = []
    mcs = []
    for _ in range(len(_WEIGHTS)):
        ssim, cs = _ssim_scale(im1, im2, max_val, filter_size, filter_sigma, k1, k2)
        mssim.append(ssim)
        mcs.append(cs)
        im1, im2 = _downsample2(im1), _downsample2(im2)
    out = img1.new_ones(())
    for w, cs in zip(_WEIGHTS[:-1], mcs[:-1]):
        out = out * cs ** w
    return out * mssim[-1] ** _WEIGHTS[-1]
