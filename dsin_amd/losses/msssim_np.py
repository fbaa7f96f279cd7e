"""Numpy MS-SSIM evaluation oracle.

Independent reference implementation of multi-scale SSIM used for test-time
metric lists, mirroring /root/reference/src/ms_ssim_np_imgcomp.py:51-200
(itself the TensorFlow-Authors reference): per-scale valid-mode Gaussian
filtering via scipy fftconvolve, 2x2/4 box downsample with reflect-mode
scipy convolve, weights [0.0448, 0.2856, 0.3001, 0.2363, 0.1333].

Inputs are NHWC numpy arrays in 0..255. Used by tests as the oracle for the
differentiable torch implementation (losses/msssim.py) — the two differ only
at image borders (the downsample pad convention), within ~1e-3 on natural
crops.
"""

from __future__ import annotations

import numpy as np
from scipy import signal
from scipy.ndimage import convolve


def _fspecial_gauss(size: int, sigma: float) -> np.ndarray:
    radius = size // 2
    offset = 0.0
    start, stop = -radius, radius + 1
    if size % 2 == 0:
        offset = 0.5
        stop -= 1
    x, y = np.mgrid[offset + start:stop, offset + start:stop]
    assert len(x) == size
    g = np.exp(-((x ** 2 + y ** 2) / (2.0 * sigma ** 2)))
    return g / g.sum()


def _ssim_for_scale(img1, img2, max_val=255, filter_size=11, filter_sigma=1.5,
                    k1=0.01, k2=0.03):
    img1 = img1.astype(np.float64)
    img2 = img2.astype(np.float64)
    _, height, width, _ = img1.shape
    size = min(filter_size, height, width)
    sigma = size * filter_sigma / filter_size if filter_size else 0
    if filter_size:
        window = np.reshape(_fspecial_gauss(size, sigma), (1, size, size, 1))
        mu1 = signal.fftconvolve(img1, window, mode="valid")
        mu2 = signal.fftconvolve(img2, window, mode="valid")
        s11 = signal.fftconvolve(img1 * img1, window, mode="valid")
        s22 = signal.fftconvolve(img2 * img2, window, mode="valid")
        s12 = signal.fftconvolve(img1 * img2, window, mode="valid")
    else:
        mu1, mu2 = img1, img2
        s11, s22, s12 = img1 * img1, img2 * img2, img1 * img2
    mu11, mu22, mu12 = mu1 * mu1, mu2 * mu2, mu1 * mu2
    s11 -= mu11
    s22 -= mu22
    s12 -= mu12
    c1 = (k1 * max_val) ** 2
    c2 = (k2 * max_val) ** 2
    v1 = 2.0 * s12 + c2
    v2 = s11 + s22 + c2
    ssim = np.mean(((2.0 * mu12 + c1) * v1) / ((mu11 + mu22 + c1) * v2))
    cs = np.mean(v1 / v2)
    return ssim, cs


def multiscale_ssim_np(img1: np.ndarray, img2: np.ndarray, max_val: float = 255,
                       filter_size: int = 11, filter_sigma: float = 1.5,
                       k1: float = 0.01, k2: float = 0.03,
                       weights=None) -> float:
    if img1.shape != img2.shape:
        raise RuntimeError(f"shape mismatch {img1.shape} vs {img2.shape}")
    if img1.ndim != 4:
        raise RuntimeError(f"expected NHWC, got ndim={img1.ndim}")
    weights = np.array(weights if weights else [0.0448, 0.2856, 0.3001, 0.2363, 0.1333])
    levels = weights.size
    down = np.ones((1, 2, 2, 1)) / 4.0
    im1, im2 = img1.astype(np.float64), img2.astype(np.float64)
    mssim, mcs = [], []
    for _ in range(levels):
        ssim, cs = _ssim_for_scale(im1, im2, max_val, filter_size, filter_sigma, k1, k2)
        mssim.append(ssim)
        mcs.append(cs)
        im1, im2 = [convolve(im, down, mode="reflect")[:, ::2, ::2, :]
                    for im in (im1, im2)]
    mssim, mcs = np.array(mssim), np.array(mcs)
    return float(np.prod(mcs[:levels - 1] ** weights[:levels - 1])
                 * (mssim[levels - 1] ** weights[levels - 1]))
