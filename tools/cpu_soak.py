"""Long-horizon training-stability soak on the CPU oracle path.

Runs full-DSIN training (synthetic stereo, fused-Adam CPU fallback — the
same flat-buffer math as the GPU step) for N steps and aborts on the first
non-finite loss. This is the control experiment for the open RD-run
stability item (profiles/r02_rd_curve.md): the oracle path isolates the
training MATH from the GPU ingredients (custom kernels, hipGraph replay).

    python tools/cpu_soak.py [--steps 5000] [--bf16] [--size 96 192]

Recorded results (profiles/r02_rd_curve.md): 6000 steps fp32 and 5000
steps --bf16 at 96x192 both finite, the bf16 run passing through a sharp
rate-collapse phase (bpp 0.22 -> 0.046) across the step region where the
GPU run went non-finite.
"""

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from dsin_amd import config as cm                   # noqa: E402
from dsin_amd.data import SyntheticStereo           # noqa: E402
from dsin_amd.models import DSIN                    # noqa: E402
from dsin_amd.training import Trainer               # noqa: E402


def main(argv=None):
    here = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=5000)
    ap.add_argument("--bf16", action="store_true",
                    help="bf16 autocast on CPU (probes precision dynamics)")
    ap.add_argument("--size", type=int, nargs=2, default=(96, 192))
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--threads", type=int, default=16)
    args = ap.parse_args(argv)

    torch.set_num_threads(args.threads)
    ae, _ = cm.parse(os.path.join(here, "run_configs", "ae_run_configs"))
    pc, _ = cm.parse(os.path.join(here, "run_configs", "pc_run_configs"))
    H, W = args.size
    ae.crop_size = (H, W)
    ae.y_patch_size = (16, 16)
    torch.manual_seed(args.seed)
    model = DSIN(ae, pc)
    tr = Trainer(model, ae, pc, 1576, nan_guard=True)
    if args.bf16:
        tr._autocast = lambda: torch.autocast(device_type="cpu",
                                              dtype=torch.bfloat16)
    gen = SyntheticStereo(H, W, seed=3)
    t0 = time.time()
    for i in range(args.steps):
        x, y = gen.next_batch()
        loss, bpp = tr.train_step(x, y)   # nan_guard raises + dumps on NaN
        if (i + 1) % 100 == 0:
            print(f"{i + 1:6d} loss {float(loss):10.2f} bpp {float(bpp):.4f}"
                  f" ({(time.time() - t0) / (i + 1):.2f}s/step)", flush=True)
    print("done: all steps finite", flush=True)


if __name__ == "__main__":
    main()
