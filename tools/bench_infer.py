"""Inference / serving benchmark: DSIN reconstruct (encoder -> quantize ->
decoder -> siFinder -> siNet) on the reference's eval crop (320x1224,
batch 1 — run_configs/ae_run_configs:4). No gradients, eager (the
inference path is not graph-captured). Run on the GPU box:

    python tools/bench_infer.py [--height 320] [--width 1224] [--steps 50]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from dsin_amd import config as config_mod
from dsin_amd.data import SyntheticStereo
from dsin_amd.models import DSIN

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--height", type=int, default=320)
    ap.add_argument("--width", type=int, default=1224)
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--warmup", type=int, default=10)
    args = ap.parse_args()

    ae, _ = config_mod.parse(os.path.join(ROOT, "run_configs", "ae_run_configs"))
    pc, _ = config_mod.parse(os.path.join(ROOT, "run_configs", "pc_run_configs"))
    ae.crop_size = (args.height, args.width)
    ph, pw = ae.y_patch_size
    if args.height % ph or args.width % pw:
        ph = next(p for p in (20, 16, 32, 8, 4, 2, 1) if args.height % p == 0)
        pw = next(p for p in (24, 32, 16, 8, 4, 2, 1) if args.width % p == 0)
        ae.y_patch_size = (ph, pw)

    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    model = DSIN(ae, pc).to(dev).eval()
    gen = SyntheticStereo(args.height, args.width, batch_size=1, device=dev)
    x, y = gen.next_batch()
    with torch.no_grad(), torch.autocast("cuda", dtype=torch.bfloat16,
                                         cache_enabled=False):
        for _ in range(args.warmup):
            out = model.reconstruct(x, y)
        torch.cuda.synchronize()
        t0 = time.time()
        for _ in range(args.steps):
            out = model.reconstruct(x, y)
        torch.cuda.synchronize()
        assert all(o is not None for o in out)
    dt = (time.time() - t0) / args.steps
    print({"metric": "inference imgs/sec @%dx%d" % (args.height, args.width),
           "value": round(1.0 / dt, 2), "ms_per_img": round(dt * 1e3, 2)})


if __name__ == "__main__":
    main()
