"""Side-information patch search driver.

Batch wrapper over the single-image NCC search op (ops.ncc_search): for each
image in the batch, match every non-overlapping (ph, pw) patch of x_dec
against all locations of the DECODED side image y_dec (Pearson correlation on
fixed-normalized, H1H2H3-decorrelated values, weighted by a Gaussian location
prior), then gather the winning patches FROM THE ORIGINAL y and scatter them
back into an image. Mirror of /root/reference/src/siFull_img.py:5-68 +
src/siFinder.py:7-135 + the mask construction at src/AE.py:193-220.

The whole search is non-trainable (reference src/siFinder.py:3-5) and y_syn
is consumed under stop_gradient (src/AE.py:67), so this path runs without
autograd entirely.
"""

from __future__ import annotations


import torch

from .. import ops


class SiFinder(torch.nn.Module):
    def __init__(self, config):
        super().__init__()
        self.ph, self.pw = (int(v) for v in config.y_patch_size)
        self.use_mask = bool(config.use_gauss_mask)
        self.l2lab = bool(config.use_L2andLAB)
        if self.l2lab:
            import warnings
            warnings.warn(
                "use_L2andLAB runs on the torch path (correct but slower "
                "than the streaming HIP NCC kernel used for the shipped "
                "Pearson+H1H2H3 mode)")

    @torch.no_grad()
    def forward(self, x_dec: torch.Tensor, y_orig: torch.Tensor,
                y_dec: torch.Tensor) -> torch.Tensor:
        """x_dec, y_orig, y_dec: (N, 3, H, W) -> y_syn (N, 3, H, W)."""
        outs = []
        for n in range(x_dec.shape[0]):
            y_syn, _, _ = ops.ncc_search(x_dec[n], y_dec[n], y_orig[n],
                                         self.ph, self.pw, self.use_mask,
                                         l2lab=self.l2lab)
            outs.append(y_syn)
        return torch.stack(outs).to(x_dec.dtype)
