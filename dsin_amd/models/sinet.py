"""siNet: dilated context-aggregation fusion CNN.

Mirror of the reference (/root/reference/src/siNet.py:29-41): input is
concat(norm(x_dec), sg(norm(y_syn))) with 6 channels; 9x 3x3 conv(32) with
dilation rates 1,2,4,8,16,32,64,128,1 and leaky-ReLU slope 0.2 (:9-10), no
batch-norm, identity-initialized weights (:13-20), then a 1x1 conv to 3
channels (xavier init, linear).
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

DILATIONS = (1, 2, 4, 8, 16, 32, 64, 128, 1)


def identity_init_(w: torch.Tensor) -> None:
    """weight (out, in, kh, kw): delta at the kernel center mapping channel
    i -> i for i < in (reference src/siNet.py:13-20, which writes
    array[cx, cy, i, i] in HWIO layout)."""
    with torch.no_grad():
        w.zero_()
        cout, cin, kh, kw = w.shape
        for i in range(min(cin, cout)):
            w[i, i, kh // 2, kw // 2] = 1.0


class SiNet(nn.Module):
    def __init__(self, cin: int = 6, width: int = 32):
        super().__init__()
        convs = []
        ch = cin
        for d in DILATIONS:
            conv = nn.Conv2d(ch, width, 3, padding=d, dilation=d)
            identity_init_(conv.weight)
            nn.init.zeros_(conv.bias)
            convs.append(conv)
            ch = width
        self.convs = nn.ModuleList(convs)
        self.last = nn.Conv2d(width, 3, 1)
        nn.init.xavier_uniform_(self.last.weight)
        nn.init.zeros_(self.last.bias)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from ..ops import conv as dconv
        for conv in self.convs:
            d = conv.dilation[0]
            # fused bias + leaky-relu epilogue on the gather-GEMM kernel
            x = dconv.conv2d(x, conv.weight, conv.bias, stride=1, padding=d,
                             dilation=d, act=2)
        return dconv.conv2d(x, self.last.weight, self.last.bias, 1, 0, 1, act=0)
