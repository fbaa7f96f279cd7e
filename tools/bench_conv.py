"""Micro-benchmark of the custom gather-GEMM conv kernels vs torch/MIOpen,
on the DSIN layer shapes. Run on the GPU box:
    python tools/bench_conv.py [--wrw] [--bwd]
Prints per-shape times (ms) and effective TFLOP/s for both paths.
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from dsin_amd.ops import conv as dconv

SHAPES = [
    # name, Ci, Co, H, W, k, stride, dil  (B=1)
    ("res3x3_128 @80x240", 128, 128, 80, 240, 3, 1, 1),
    ("h1 5x5s2 3->64 @320x960", 3, 64, 320, 960, 5, 2, 1),
    ("h2 5x5s2 64->128 @160x480", 64, 128, 160, 480, 5, 2, 1),
    ("to_bn 5x5s2 128->33 @80x240", 128, 33, 80, 240, 5, 2, 1),
    ("sinet d1 6->32 @320x960", 6, 32, 320, 960, 3, 1, 1),
    ("sinet d32 32->32 @320x960", 32, 32, 320, 960, 3, 1, 32),
]


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000.0


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=20)
    args = ap.parse_args()
    dev = torch.device("cuda:0")
    print(f"{'shape':34s} {'ours-f':>8s} {'miopen-f':>9s} {'ours-w':>8s} "
          f"{'miopen-w':>9s} {'ours-dx':>8s} {'mi-dx':>8s}  TF(f)")
    for name, Ci, Co, H, W, k, st, d in SHAPES:
        pad = (k - 1) // 2 * d if d > 1 else (k - 1) // 2
        pad = d if (k == 3 and d > 1) else (k - 1) // 2
        x = torch.randn(1, Ci, H, W, device=dev)
        w = torch.randn(Co, Ci, k, k, device=dev) * 0.05
        xb = x.to(torch.bfloat16)
        wb = w.to(torch.bfloat16)
        HO = (H + 2 * pad - (k - 1) * d - 1) // st + 1
        WO = (W + 2 * pad - (k - 1) * d - 1) // st + 1
        flops = 2.0 * HO * WO * Co * Ci * k * k

        t_ours_f = timeit(lambda: dconv.conv2d(x, w, None, st, pad, d),
                          args.iters)
        t_mi_f = timeit(lambda: F.conv2d(xb, wb, None, stride=st, padding=pad,
                                         dilation=d), args.iters)

        # wrw via autograd (same route the training step takes)
        dy = torch.randn(1, Co, HO, WO, device=dev).to(torch.bfloat16)
        w2 = w.clone().requires_grad_(True)
        def ours_w():
            yv = dconv.conv2d(x.detach(), w2, None, st, pad, d)
            torch.autograd.grad(yv, w2, dy.float())
        t_ours_w = timeit(ours_w, args.iters)
        wg = wb.requires_grad_(True)
        def mi_w():
            y = F.conv2d(xb, wg, None, stride=st, padding=pad, dilation=d)
            torch.autograd.grad(y, wg, dy)
        t_mi_w = timeit(mi_w, args.iters)

        # bwd-data alone via autograd on x
        x2 = x.clone().requires_grad_(True)
        def ours_dx():
            y = dconv.conv2d(x2, w.detach(), None, st, pad, d)
            torch.autograd.grad(y, x2, dy.float(), retain_graph=False)
        t_ours_dx = timeit(ours_dx, max(args.iters // 2, 5))
        xb2 = xb.clone().requires_grad_(True)
        def mi_dx():
            y = F.conv2d(xb2, wb.detach(), None, stride=st, padding=pad, dilation=d)
            torch.autograd.grad(y, xb2, dy)
        t_mi_dx = timeit(mi_dx, max(args.iters // 2, 5))

        print(f"{name:34s} {t_ours_f:8.3f} {t_mi_f:9.3f} {t_ours_w:8.3f} "
              f"{t_mi_w:9.3f} {t_ours_dx:8.3f} {t_mi_dx:8.3f}  "
              f"{flops / (t_ours_f * 1e9):6.0f}")


if __name__ == "__main__":
    main()
