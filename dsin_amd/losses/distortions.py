"""Distortion metrics and the rate-distortion loss assembly.

Mirror of /root/reference/src/Distortions_imgcomp.py. Semantics kept exactly:

* MAE / MSE / PSNR are per-image means over CHW then a batch mean (:57-106).
* When a metric is NOT the minimized distortion (or at eval), both images are
  cast to int32 (truncation) first, "to ensure real world errors" (:17-22).
* d_loss_scaled: mae | mse | K_psnr - psnr | K_ms_ssim * (1 - ms_ssim) (:43-55).
* get_loss (:113-146): H_real = mean(bc); H_mask = mean(bc * heatmap3D);
  H_soft = (H_mask + H_real)/2; pc_loss = beta * max(H_soft - H_target, 0);
  total = d_loss_scaled + pc_loss + regularizers.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from .msssim import multiscale_ssim


def _maybe_int(x: torch.Tensor, cast: bool) -> torch.Tensor:
    return x.to(torch.int32).to(torch.float32) if cast else x


def mae_per_image(x: torch.Tensor, y: torch.Tensor, cast_to_int: bool) -> torch.Tensor:
    if not cast_to_int and x.is_cuda:
        from .. import ops
        return ops.l1_mean_per_image(x, y)  # fused single-pass kernel
    x, y = _maybe_int(x, cast_to_int), _maybe_int(y, cast_to_int)
    return (y - x).abs().float().mean(dim=(1, 2, 3))


def mse_per_image(x: torch.Tensor, y: torch.Tensor, cast_to_int: bool) -> torch.Tensor:
    x, y = _maybe_int(x, cast_to_int), _maybe_int(y, cast_to_int)
    return ((y - x) ** 2).float().mean(dim=(1, 2, 3))


def psnr_per_image(x: torch.Tensor, y: torch.Tensor, cast_to_int: bool) -> torch.Tensor:
    mse = mse_per_image(x, y, cast_to_int)
    return 10.0 * torch.log10(255.0 * 255.0 / mse)


class Distortions:
    """Lazy metric evaluation: TF builds all metric nodes but only computes
    the fetched ones (the reference's sess.run prunes the graph); eager
    mirrors that by evaluating each metric on first attribute access.
    `d_loss_scaled` (the minimized distortion) is computed eagerly."""

    def __init__(self, config, x: torch.Tensor, x_out: torch.Tensor, is_training: bool):
        minimize_for = config.distortion_to_minimize
        assert minimize_for in ("mae", "mse", "psnr", "ms_ssim")
        self._x, self._xo = x, x_out
        self._int_psnr = (not is_training) or minimize_for != "psnr"
        self._int_mse = (not is_training) or minimize_for != "mse"
        self._int_mae = (not is_training) or minimize_for != "mae"
        self._cache = {}
        self._minimize_for = minimize_for

        if minimize_for == "mae":
            self.d_loss_scaled = self.mae
        elif minimize_for == "mse":
            self.d_loss_scaled = self.mse
        elif minimize_for == "psnr":
            self.d_loss_scaled = config.K_psnr - self.psnr
        else:
            self.d_loss_scaled = config.K_ms_ssim * (1.0 - self.ms_ssim)

    @property
    def mae(self):
        if "mae" not in self._cache:
            self._cache["mae"] = mae_per_image(self._x, self._xo, self._int_mae).mean()
        return self._cache["mae"]

    @property
    def mse(self):
        if "mse" not in self._cache:
            self._cache["mse"] = mse_per_image(self._x, self._xo, self._int_mse).mean()
        return self._cache["mse"]

    @property
    def psnr(self):
        if "psnr" not in self._cache:
            self._cache["psnr"] = psnr_per_image(self._x, self._xo, self._int_psnr).mean()
        return self._cache["psnr"]

    @property
    def ms_ssim(self):
        if "ms_ssim" not in self._cache:
            self._cache["ms_ssim"] = multiscale_ssim(self._x, self._xo)
        return self._cache["ms_ssim"]


def get_loss(config, d_loss_scaled: torch.Tensor, bc: torch.Tensor,
             heatmap: Optional[torch.Tensor], reg_loss: torch.Tensor
             ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Returns (total_loss, H_real, pc_loss). reg_loss is the summed L2
    regularizers (encoder+decoder conv weights, centers, PC when enabled) —
    the reference collects these through tf.losses REGULARIZATION_LOSSES
    (:129-136); we sum them explicitly in the model."""
    assert config.H_target
    if heatmap is not None:
        from .. import ops
        H_real, H_mask = ops.rate_terms(bc, heatmap)  # fused single pass
    else:
        H_real = bc.mean()
        H_mask = H_real
    H_soft = 0.5 * (H_mask + H_real)
    pc_loss = float(config.beta) * torch.clamp(H_soft - float(config.H_target), min=0.0)
    total = d_loss_scaled + pc_loss + reg_loss
    return total, H_real, pc_loss


def bitcost_to_bpp(bc: torch.Tensor, x: torch.Tensor) -> torch.Tensor:
    """bpp = sum(bc) / (N*H*W) of the INPUT image x
    (reference src/bits_imgcomp.py:4-20)."""
    n, _, h, w = x.shape
    return bc.sum() / float(n * h * w)
