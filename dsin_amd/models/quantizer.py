"""Soft-to-hard scalar quantizer module.

Holds the trainable centers variable, shape (L,), uniform init in
centers_initial_range (reference src/quantizer_imgcomp.py:11-31, seed 666),
and applies the fused quantize op (HIP kernel on GPU). The straight-through
estimator qbar = qsoft + sg(qhard - qsoft) lives inside the op (reference
src/autoencoder_imgcomp.py:131-134). The L2 regularization term on centers
(factor regularization_factor_centers, reference src/quantizer_imgcomp.py:18-24)
is exposed via `regularization_loss()` and summed by the loss assembly.
"""

from __future__ import annotations

from typing import Tuple

import torch
import torch.nn as nn

from .. import ops


class Quantizer(nn.Module):
    def __init__(self, config):
        super().__init__()
        L = int(config.num_centers)
        lo, hi = (int(v) for v in config.centers_initial_range)  # ref casts to int (:29)
        gen = torch.Generator().manual_seed(666)
        init = torch.empty(L).uniform_(float(lo), float(hi), generator=gen)
        self.centers = nn.Parameter(init)
        self.sigma = 1.0  # reference src/autoencoder_imgcomp.py:131 (sigma=1)
        self.reg_factor = float(config.regularization_factor_centers or 0.0)

    def forward(self, z: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        return ops.quantize(z, self.centers, self.sigma)

    def regularization_loss(self) -> torch.Tensor:
        # tf.nn.l2_loss = sum(w^2)/2 (reference src/quantizer_imgcomp.py:23)
        if self.reg_factor == 0.0:
            return self.centers.new_zeros(())
        return self.reg_factor * 0.5 * (self.centers ** 2).sum()
