"""Convolutional autoencoder with soft quantization bottleneck.

Architecture mirror of the reference's `_CVPR` network
(/root/reference/src/autoencoder_imgcomp.py:214-288): 8x subsampling encoder
(5x5/s2 conv(64) -> 5x5/s2 conv(128) -> B=5 groups of 3 residual blocks with
inner and outer skips -> final linear residual block -> 5x5/s2 conv to
C+1 channels), heatmap masking (:172-201), quantization, and a mirrored
decoder on transpose convs (:247-269). Every conv carries fused BatchNorm
(decay 0.9 == torch momentum 0.1, eps 1e-5, scale=True; reference
:115-125) and ReLU unless noted.

Deviation from the reference (documented, intentional): spatial padding is
symmetric (PyTorch convention) instead of TF's SAME asymmetric padding —
output sizes are identical for /8-divisible inputs; filters see a one-pixel
shifted window at the borders. We are not bit-compatible with TF checkpoints
(the checkpoint contract we keep is the scope layout, see training/checkpoint.py).
"""

from __future__ import annotations

from typing import NamedTuple, Optional

import torch
import torch.nn as nn

from .. import ops
from .quantizer import Quantizer

ARCH_PARAM_N = 128  # reference src/autoencoder_imgcomp.py:211


class EncoderOutput(NamedTuple):
    """Mirror of the reference's EncoderOutput (src/autoencoder_imgcomp.py:15)."""
    qbar: torch.Tensor
    symbols: torch.Tensor
    z: torch.Tensor
    heatmap: Optional[torch.Tensor]


def _bn(ch: int) -> nn.BatchNorm2d:
    # TF slim fused BN: decay .9 -> momentum .1, eps 1e-5, scale=True
    return nn.BatchNorm2d(ch, eps=1e-5, momentum=0.1, affine=True)


class ConvBNAct(nn.Module):
    """conv (or transpose conv) + BN + optional ReLU. The conv runs on the
    custom gather-GEMM MFMA kernel on GPU (ops/conv.py) and torch eager on
    CPU; nn.Conv2d modules are kept purely as parameter holders so the
    state-dict layout stays conventional."""

    def __init__(self, cin: int, cout: int, k: int, stride: int = 1,
                 relu: bool = True, transpose: bool = False):
        super().__init__()
        self.transpose = transpose
        self.stride = stride
        self.padding = (k - 1) // 2
        self.output_padding = stride - 1 if transpose else 0
        if transpose:
            # out = 2*in for stride 2: pad=(k-1)//2, output_padding=1
            self.conv = nn.ConvTranspose2d(cin, cout, k, stride=stride,
                                           padding=self.padding,
                                           output_padding=self.output_padding,
                                           bias=False)
        else:
            self.conv = nn.Conv2d(cin, cout, k, stride=stride,
                                  padding=self.padding, bias=False)
        self.bn = _bn(cout)
        self.relu = relu

    def forward(self, x, residual=None):
        from ..ops import conv as dconv
        from ..ops.bn import batch_norm_act
        if self.transpose:
            y = dconv.conv_transpose2d(x, self.conv.weight, None, self.stride,
                                       self.padding, self.output_padding)
        else:
            y = dconv.conv2d(x, self.conv.weight, None, self.stride,
                             self.padding, 1)
        return batch_norm_act(y, self.bn, self.training,
                              act=1 if self.relu else 0, residual=residual)


class ResidualBlock(nn.Module):
    """Two 3x3 convs (first ReLU'd unless act=False, second always linear)
    plus identity skip (reference src/autoencoder_imgcomp.py:275-288; the
    `activation_fn=None` call sites at :233-234,261-262 zero the first conv's
    activation too)."""

    def __init__(self, ch: int, act: bool = True):
        super().__init__()
        self.c1 = ConvBNAct(ch, ch, 3, relu=act)
        self.c2 = ConvBNAct(ch, ch, 3, relu=False)

    def forward(self, x):
        # skip-add fused into the second conv's BN apply pass
        return self.c2(self.c1(x), residual=x)


class ResidualStack(nn.Module):
    """B outer groups of 3 residual blocks with outer skips, then a final
    linear residual block and the outermost skip
    (reference src/autoencoder_imgcomp.py:225-235 == :253-263)."""

    def __init__(self, ch: int, B: int):
        super().__init__()
        self.groups = nn.ModuleList(
            nn.ModuleList(ResidualBlock(ch) for _ in range(3)) for _ in range(B))
        self.final = ResidualBlock(ch, act=False)

    def forward(self, x):
        outer = x
        for group in self.groups:
            inner = x
            for block in group:
                x = block(x)
            x = x + inner
        x = self.final(x)
        return x + outer


class Encoder(nn.Module):
    def __init__(self, config):
        super().__init__()
        n = ARCH_PARAM_N
        self.config = config
        self.use_heatmap = bool(config.heatmap)
        self.normalization = config.normalization
        cbn = config.num_chan_bn + (1 if self.use_heatmap else 0)
        self.h1 = ConvBNAct(3, n // 2, 5, stride=2)
        self.h2 = ConvBNAct(n // 2, n, 5, stride=2)
        self.res = ResidualStack(n, config.arch_param_B)
        self.to_bn = ConvBNAct(n, cbn, 5, stride=2, relu=False)  # BN, no act (ref :238)
        self.quantizer = Quantizer(config)

    def forward(self, x: torch.Tensor) -> EncoderOutput:
        if self.normalization == "FIXED":
            x = ops.kitti_normalize(x)
        net = self.h1(x)
        net = self.h2(net)
        net = self.res(net)
        net = self.to_bn(net)
        if self.use_heatmap:
            z, heatmap = ops.heatmap_mask(net)
        else:
            z, heatmap = net, None
        qbar, symbols = self.quantizer(z)
        return EncoderOutput(qbar=qbar, symbols=symbols, z=z, heatmap=heatmap)


class Decoder(nn.Module):
    def __init__(self, config):
        super().__init__()
        n = ARCH_PARAM_N
        self.normalization = config.normalization
        self.from_bn = ConvBNAct(config.num_chan_bn, n, 3, stride=2, transpose=True)
        self.res = ResidualStack(n, config.arch_param_B)
        self.h12 = ConvBNAct(n, n // 2, 5, stride=2, transpose=True)
        # reference h13: activation_fn=None but BN still applied by arg_scope
        self.h13 = ConvBNAct(n // 2, 3, 5, stride=2, transpose=True, relu=False)

    def forward(self, q: torch.Tensor) -> torch.Tensor:
        net = self.from_bn(q)
        net = self.res(net)
        net = self.h12(net)
        net = self.h13(net)
        if self.normalization == "FIXED":
            net = ops.kitti_denormalize(net)
        return net.clamp(0.0, 255.0)
