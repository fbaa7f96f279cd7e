"""Loss/inference figures (reference src/utils.py:12-79 equivalents).

The reference calls plt.show(block=True) in an interactive session; this
environment is headless, so both functions render with the Agg backend and
save a PNG (returning its path). matplotlib is optional: when it is not
importable the raw figure data is dumped as .npz next to the requested
path so nothing is silently lost.
"""

from __future__ import annotations

import os
import warnings
from typing import Optional, Sequence

import numpy as np

from .metrics import l1_np, psnr_np
from ..losses.msssim_np import multiscale_ssim_np


def _get_plt():
    try:
        import matplotlib
        matplotlib.use("Agg")
        import matplotlib.pyplot as plt
        return plt
    except Exception:  # pragma: no cover - env without matplotlib
        return None


def _chw_to_hwc_u8(img: np.ndarray) -> np.ndarray:
    if img.ndim == 3 and img.shape[0] in (1, 3):
        img = np.transpose(img, (1, 2, 0))
    return np.clip(img, 0, 255).astype(np.uint8)


def plot_loss(train_loss_history: Sequence[float],
              val_loss_history: Sequence[float],
              val_iters: Sequence[int], train_iters: Sequence[int],
              total_iterations: int, best_val: float, best_iter: int,
              model_name: str, out_path: Optional[str] = None) -> str:
    """Train + validation loss scatter (reference src/utils.py:12-32).
    Saves a 16x9 PNG; returns the written path."""
    out_path = out_path or f"loss_{model_name}.png"
    plt = _get_plt()
    if plt is None:
        alt = os.path.splitext(out_path)[0] + ".npz"
        warnings.warn("matplotlib unavailable - dumping loss history to " + alt)
        np.savez(alt, train_loss=np.asarray(train_loss_history),
                 val_loss=np.asarray(val_loss_history),
                 train_iters=np.asarray(train_iters),
                 val_iters=np.asarray(val_iters))
        return alt
    fig, ax = plt.subplots(figsize=(16, 9))
    ax.plot(list(train_iters), list(train_loss_history), ".", label="train")
    ax.plot(list(val_iters), list(val_loss_history), ".", label="val")
    ax.set_xlim(0, total_iterations)
    ax.set_title("Train and Validation - average loss per iteration")
    ax.legend(loc="upper left")
    ax.set_xlabel("iteration")
    ax.set_ylabel("loss")
    fig.suptitle(f"Best validation loss = {best_val}, "
                 f"Best validation iterations = {best_iter}/{total_iterations}"
                 f"\nModel name = {model_name}")
    os.makedirs(os.path.dirname(out_path) or ".", exist_ok=True)
    fig.savefig(out_path, dpi=100)
    plt.close(fig)
    return out_path


def plot_inference(x: np.ndarray, x_dec: np.ndarray, y: np.ndarray,
                   y_syn: np.ndarray, x_with_si: np.ndarray,
                   model_name: str, total_iterations: int = 0,
                   cnt="NA", lr=("NA", "NA"), bpp="NA",
                   out_path: Optional[str] = None) -> str:
    """5-panel inference figure (reference src/utils.py:35-79): original x,
    synthetic y, original y on the left; x_dec and x_with_si on the right;
    L1/PSNR/MS-SSIM with and without side information in the suptitle.
    All images CHW (or HWC) in 0..255. Saves a PNG; returns the path."""
    out_path = out_path or f"inference_{model_name}.png"
    panels = {name: _chw_to_hwc_u8(img) for name, img in
              (("x", x), ("x_dec", x_dec), ("y", y), ("y_syn", y_syn),
               ("x_with_si", x_with_si))}

    def _msssim(a, b):
        return multiscale_ssim_np(a[None].astype(np.float64),
                                  b[None].astype(np.float64))

    l1_no_si = l1_np(panels["x"], panels["x_dec"])
    l1_si = l1_np(panels["x"], panels["x_with_si"])
    psnr_no_si = psnr_np(panels["x"], panels["x_dec"])
    psnr_si = psnr_np(panels["x"], panels["x_with_si"])
    ms_no_si = _msssim(panels["x"], panels["x_dec"])
    ms_si = _msssim(panels["x"], panels["x_with_si"])

    plt = _get_plt()
    if plt is None:
        alt = os.path.splitext(out_path)[0] + ".npz"
        warnings.warn("matplotlib unavailable - dumping panels to " + alt)
        np.savez(alt, l1_no_si=l1_no_si, l1_si=l1_si, psnr_no_si=psnr_no_si,
                 psnr_si=psnr_si, msssim_no_si=ms_no_si, msssim_si=ms_si,
                 **panels)
        return alt
    fig = plt.figure(figsize=(18, 11))
    for pos, key, title in ((321, "x", "original x"),
                            (323, "y_syn", "synthetic y"),
                            (325, "y", "original y"),
                            (222, "x_dec", "x decoded"),
                            (224, "x_with_si", "x_with_si")):
        ax = fig.add_subplot(pos)
        ax.imshow(panels[key])
        ax.set_title(title)
        ax.set_axis_off()
    fig.suptitle(
        f"x_no_si: l1 = {l1_no_si:.3f}, psnr = {psnr_no_si:.2f}, "
        f"ms-ssim = {ms_no_si:.4f}\n"
        f"x_with_si: l1 = {l1_si:.3f}, psnr = {psnr_si:.2f}, "
        f"ms-ssim = {ms_si:.4f}\n"
        f"ae_lr = {lr[0]}, pc_lr = {lr[1]}, iters = {cnt}/{total_iterations}, "
        f"bpp = {bpp}\nModel Name = {model_name}")
    fig.subplots_adjust(top=0.8)
    os.makedirs(os.path.dirname(out_path) or ".", exist_ok=True)
    fig.savefig(out_path, dpi=100)
    plt.close(fig)
    return out_path
