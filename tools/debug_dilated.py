"""Debug the large-dilation conv mismatch (tests/test_gpu_conv.py)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F
from dsin_amd.ops import conv as dconv

dev = torch.device("cuda:0")
torch.manual_seed(0)
Ci, Co, H, W, k, st, pad, d = 32, 32, 48, 72, 3, 1, 16, 16
x = torch.randn(2, Ci, H, W, device=dev)
w = torch.randn(Co, Ci, k, k, device=dev) / (k * Ci ** 0.5)
b = torch.randn(Co, device=dev)

x1 = x.clone().requires_grad_(True)
w1 = w.clone().requires_grad_(True)
b1 = b.clone().requires_grad_(True)
y = dconv.conv2d(x1, w1, b1, st, pad, d, 0)
x2 = x.clone().requires_grad_(True)
w2 = w.clone().requires_grad_(True)
b2 = b.clone().requires_grad_(True)
yr = F.conv2d(x2, w2, b2, stride=st, padding=pad, dilation=d)
print("fwd maxerr:", (y.float() - yr).abs().max().item(),
      "ref absmax:", yr.abs().max().item())
g = torch.randn_like(yr)
y.backward(g.to(y.dtype))
yr.backward(g)
dxe = (x1.grad.float() - x2.grad).abs()
dwe = (w1.grad.float() - w2.grad).abs()
print("dx maxerr:", dxe.max().item(), "at", (dxe == dxe.max()).nonzero()[0].tolist(),
      "| dx absmax:", x2.grad.abs().max().item())
print("dw maxerr:", dwe.max().item(), "| dw absmax:", w2.grad.abs().max().item())
print("db maxerr:", (b1.grad.float() - b2.grad).abs().max().item())
# error structure: per-row max of dx err
print("dx err per channel (first 8):", dxe.amax(dim=(0, 2, 3))[:8].tolist())
print("dx err frac>0.1:", (dxe > 0.1).float().mean().item())
