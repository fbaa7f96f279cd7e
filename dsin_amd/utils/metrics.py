"""Eval metrics, artifact IO and structured run logging.

Numpy metric utilities mirroring /root/reference/src/utils.py:82-180 (L1,
PSNR, MS-SSIM list files, per-patch Pearson correlation), the test-image
writer (:102-111, named `<index>_<bpp>bpp.png`), plus a JSONL metrics logger
(the rebuild's replacement for the reference's console-only observability,
SURVEY.md section 5.5).
"""

from __future__ import annotations

import json
import os
import time
from typing import Optional

import numpy as np

from ..data.png import write_png
from ..losses.msssim_np import multiscale_ssim_np


def l1_np(a: np.ndarray, b: np.ndarray) -> float:
    return float(np.mean(np.abs(a.astype(np.float64) - b.astype(np.float64))))


def mse_np(a: np.ndarray, b: np.ndarray) -> float:
    return float(np.mean((a.astype(np.float64) - b.astype(np.float64)) ** 2))


def psnr_np(a: np.ndarray, b: np.ndarray) -> float:
    return float(10.0 * np.log10(255.0 ** 2 / mse_np(a, b)))


def pearson_per_patch(a: np.ndarray, b: np.ndarray, ph: int, pw: int) -> float:
    """Mean Pearson correlation over the non-overlapping (ph, pw) patch grid
    of two (H, W, C) or (C, H, W) images (reference src/utils.py:161-180 via
    skimage view_as_windows; reimplemented with reshape)."""
    if a.shape[0] in (1, 3) and a.ndim == 3:  # CHW -> HWC
        a, b = np.transpose(a, (1, 2, 0)), np.transpose(b, (1, 2, 0))
    h, w, c = a.shape
    gh, gw = h // ph, w // pw
    a = a[:gh * ph, :gw * pw].reshape(gh, ph, gw, pw, c)
    b = b[:gh * ph, :gw * pw].reshape(gh, ph, gw, pw, c)
    a = a.transpose(0, 2, 1, 3, 4).reshape(gh * gw, -1).astype(np.float64)
    b = b.transpose(0, 2, 1, 3, 4).reshape(gh * gw, -1).astype(np.float64)
    a = a - a.mean(1, keepdims=True)
    b = b - b.mean(1, keepdims=True)
    denom = np.sqrt((a * a).sum(1) * (b * b).sum(1))
    denom = np.where(denom == 0, 1.0, denom)
    return float(np.mean((a * b).sum(1) / denom))


def save_test_img(root_save_img: str, model_name: str, img_chw: np.ndarray,
                  index: int, bpp: float) -> str:
    """Write the reconstructed image as `<index>_<bpp>bpp.png`
    (reference src/utils.py:102-111)."""
    out_dir = os.path.join(root_save_img, "images", model_name)
    os.makedirs(out_dir, exist_ok=True)
    path = os.path.join(out_dir, f"{index}_{bpp:.4f}bpp.png")
    write_png(path, np.transpose(np.clip(img_chw, 0, 255), (1, 2, 0)).astype(np.uint8))
    return path


def loss_list_saver(x: np.ndarray, y: np.ndarray, x_rec: np.ndarray,
                    y_syn: Optional[np.ndarray], model_name: str, bpp: float,
                    root_save_img: str, ph: int = 20, pw: int = 24) -> None:
    """Append per-image metric lines to list files (reference
    src/utils.py:114-158): bpp, L1, PSNR, MS-SSIM, MSE, mean patch Pearson."""
    out_dir = os.path.join(root_save_img, "loss_lists")
    os.makedirs(out_dir, exist_ok=True)

    def appendf(name, value):
        with open(os.path.join(out_dir, f"{name}_{model_name}.txt"), "a") as f:
            f.write(f"{value}\n")

    x0, r0 = x[0], x_rec[0]
    appendf("bpp", bpp)
    appendf("l1", l1_np(x0, r0))
    appendf("psnr", psnr_np(x0, r0))
    appendf("mse", mse_np(x0, r0))
    nhwc = lambda img: np.transpose(img, (1, 2, 0))[None]
    appendf("msssim", multiscale_ssim_np(nhwc(x0), nhwc(np.clip(r0, 0, 255))))
    if y_syn is not None:
        appendf("pearson", pearson_per_patch(x0, y_syn[0], ph, pw))


def eval_msssim_bpp(model, x, y):
    """MS-SSIM (numpy 5-scale oracle) + bpp of the model's reconstruction of
    one batch — the quality half of the headline metric ("MS-SSIM @ 0.02
    bpp"). Uses x_with_si when side information is active,
    x_dec otherwise. Returns (msssim, bpp) floats; msssim is NaN when the
    crop is too small for 5 dyadic scales (min side < 176)."""
    import torch
    with torch.no_grad():
        _, _, x_dec, x_with_si, bpp = model.reconstruct(x, y)
        x_rec = (x_with_si if x_with_si is not None and
                 float(x_with_si.abs().mean()) > 0 else x_dec)
        x_np = x.float().cpu().numpy()
        r_np = x_rec.float().clamp(0, 255).cpu().numpy()
    if min(x_np.shape[-2], x_np.shape[-1]) < 176:
        return float("nan"), float(bpp)
    nhwc = lambda img: np.transpose(img, (0, 2, 3, 1))
    return float(multiscale_ssim_np(nhwc(x_np), nhwc(r_np))), float(bpp)


class MetricsLogger:
    """JSONL metrics stream: one line per event, flushed immediately."""

    def __init__(self, path: Optional[str]):
        self.path = path
        if path:
            os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
            self._f = open(path, "a")
        else:
            self._f = None

    def log(self, event: str, **fields):
        if self._f is None:
            return
        rec = {"t": time.time(), "event": event}
        rec.update(fields)
        self._f.write(json.dumps(rec) + "\n")
        self._f.flush()

    def close(self):
        if self._f:
            self._f.close()
