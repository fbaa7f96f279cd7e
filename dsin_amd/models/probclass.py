"""Autoregressive entropy model ("probability classifier", PC).

Causal masked 3D-CNN over the quantized symbol volume — mirror of the
reference's `_ResShallow` (/root/reference/src/probclass_imgcomp.py:199-221):

  conv0 (first_mask, k channels) -> 1 residual block of 2 masked convs
  (input cropped [2:, 2:-2, 2:-2] to track VALID shrinkage, :185-196)
  -> conv2 to L logits.

Filter shape DHW = (K//2+1, K, K) = (2,3,3) (:145-148). Causality masks
(:150-176): in the *current* depth plane (last filter plane) zero everything
right of the center (inclusive for the first layer, exclusive after) and all
rows below. Padding: front-only in depth, symmetric pad = context//2 = 4 in
H/W, filled with centers[0] when use_centers_for_padding (:59-61,268-292).

Input to bitcost is stop_gradient(qbar) (reference src/AE.py:74) — the PC
trains its own weights; rate gradients reach the encoder only through the
heatmap mask term in the loss.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops


def first_mask(K: int) -> torch.Tensor:
    """(1,1,D,K,K) strict-causal mask for the first layer (center excluded).
    Reference src/probclass_imgcomp.py:150-162."""
    D = K // 2 + 1
    m = torch.ones(D, K, K)
    m[-1, K // 2, K // 2:] = 0
    m[-1, K // 2 + 1:, :] = 0
    return m.view(1, 1, D, K, K)


def other_mask(K: int) -> torch.Tensor:
    """(1,1,D,K,K) causal mask with the center kept (later layers).
    Reference src/probclass_imgcomp.py:164-176."""
    D = K // 2 + 1
    m = torch.ones(D, K, K)
    m[-1, K // 2, K // 2 + 1:] = 0
    m[-1, K // 2 + 1:, :] = 0
    return m.view(1, 1, D, K, K)


class MaskedConv3d(nn.Module):
    """VALID 3D conv whose weight is multiplied by a fixed causality mask
    every forward (reference src/probclass_imgcomp.py:227-261). Weight layout
    (out, in, D, kH, kW); input (N, ch, D, H, W)."""

    def __init__(self, cin: int, cout: int, K: int, mask: torch.Tensor):
        super().__init__()
        D = K // 2 + 1
        w = torch.empty(cout, cin, D, K, K)
        # xavier/glorot uniform (reference uses xavier_initializer, :235)
        nn.init.xavier_uniform_(w)
        self.weight = nn.Parameter(w)
        self.bias = nn.Parameter(torch.zeros(cout))
        self.register_buffer("mask", mask.to(w.dtype), persistent=False)

    def forward(self, x: torch.Tensor, act: int = 0) -> torch.Tensor:
        from ..ops import conv as dconv
        return dconv.conv3d_valid(x, self.weight * self.mask, self.bias, act)


class ProbClass(nn.Module):
    NUM_RESIDUAL = 1  # reference _ResShallow._NUM_RESIDUAL

    def __init__(self, pc_config, num_centers: int):
        super().__init__()
        self.config = pc_config
        self.L = int(num_centers)
        K = int(pc_config.kernel_size)
        k = int(pc_config.arch_param__k)
        self.K = K
        fm, om = first_mask(K), other_mask(K)
        self.conv0 = MaskedConv3d(1, k, K, fm)
        self.res_conv1 = MaskedConv3d(k, k, K, om)
        self.res_conv2 = MaskedConv3d(k, k, K, om)
        self.conv2 = MaskedConv3d(k, self.L, K, om)
        self.reg_factor = pc_config.regularization_factor  # None in shipped config

    @classmethod
    def num_layers(cls) -> int:
        return 2 + cls.NUM_RESIDUAL * 2  # reference :207-212

    def context_size(self) -> int:
        return self.num_layers() * (self.K - 1) + 1  # reference :47-52

    def logits(self, q_pad: torch.Tensor) -> torch.Tensor:
        """q_pad: (N, 1, D+pad, H+2pad, W+2pad) -> logits (N, L, D, H, W).
        ReLUs fused into the masked-conv epilogue on GPU."""
        net = self.conv0(q_pad, act=1)
        inner = net
        net = self.res_conv1(net, act=1)
        net = self.res_conv2(net)
        net = net + inner[:, :, 2:, 2:-2, 2:-2]  # VALID shrink tracking (:196)
        return self.conv2(net)

    def bitcost(self, q: torch.Tensor, symbols: torch.Tensor,
                pad_value: torch.Tensor) -> torch.Tensor:
        """q: (N, C, H, W) detached qbar; symbols: (N, C, H, W) int64.
        Returns bits per symbol (N, C, H, W) (reference :63-106)."""
        pad = self.context_size() // 2
        q_pad = ops.pad_for_probclass(q, pad, pad_value)
        logits = self.logits(q_pad.unsqueeze(1))
        return ops.bitcost_ce(logits, symbols)

    def regularization_loss(self) -> torch.Tensor:
        dev = self.conv0.weight.device
        if self.reg_factor is None:
            return torch.zeros((), device=dev)
        s = torch.zeros((), device=dev)
        for m in (self.conv0, self.res_conv1, self.res_conv2, self.conv2):
            s = s + 0.5 * (m.weight ** 2).sum()
        return float(self.reg_factor) * s
