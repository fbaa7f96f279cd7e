import sys, os, torch
sys.path.insert(0, "/root/repo")
from dsin_amd.ops import conv as dconv
dev = torch.device("cuda:0")
x = torch.randn(1, 128, 80, 240, device=dev)
w = torch.randn(128, 128, 3, 3, device=dev) * 0.05
for _ in range(30):
    y = dconv.conv2d(x, w, None, 1, 1, 1)
torch.cuda.synchronize()
