"""Replay a nan_blackbox dump (training/trainer.py --nan-guard) offline.

Rebuilds the model from the run configs, loads the dumped (post-failure)
model state and failing batch, and re-runs forward + backward with
per-module hooks that report the FIRST module whose output (forward) or
input-gradient (backward) goes non-finite — localizing the blowup to a
layer in one command:

    python tools/replay_blackbox.py nan_blackbox/step_4300.pt \
        [-ae_config P] [-pc_config P] [--device cuda:0] [--bf16]

Prints the dump's recorded per-component losses and corruption map first,
so the pre-recorded evidence and the replay can be compared directly.
"""

from __future__ import annotations

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from dsin_amd import config as config_mod           # noqa: E402
from dsin_amd.models import DSIN                    # noqa: E402


def _nonfinite(t) -> int:
    if not isinstance(t, torch.Tensor) or not t.is_floating_point():
        return 0
    return int((~torch.isfinite(t.detach())).sum())


def main(argv=None):
    cur = os.getcwd()
    ap = argparse.ArgumentParser()
    ap.add_argument("box", help="nan_blackbox/step_<N>.pt file")
    ap.add_argument("-ae_config", "--ae_config_path", type=str,
                    default=os.path.join(cur, "run_configs", "ae_run_configs"))
    ap.add_argument("-pc_config", "--pc_config_path", type=str,
                    default=os.path.join(cur, "run_configs", "pc_run_configs"))
    ap.add_argument("--device", type=str, default=None)
    ap.add_argument("--bf16", action="store_true",
                    help="replay under bf16 autocast (match the failing run)")
    args = ap.parse_args(argv)

    box = torch.load(args.box, weights_only=False, map_location="cpu")
    print(f"== black box: global step {box['global_step']}")
    print(f"   recorded components: {box['components']}")
    corrupt = {k: v for k, v in {**box["param_stats"],
                                 **box["optimizer_stats"]}.items()
               if v.get("nonfinite")}
    print(f"   corrupt buffers at dump time: {corrupt or 'none'}")

    device = torch.device(args.device if args.device else
                          ("cuda:0" if torch.cuda.is_available() else "cpu"))
    ae_config, _ = config_mod.parse(args.ae_config_path)
    pc_config, _ = config_mod.parse(args.pc_config_path)
    model = DSIN(ae_config, pc_config).to(device)
    missing, unexpected = model.load_state_dict(box["model_state"],
                                                strict=False)
    if missing or unexpected:
        print(f"   state_dict: missing={missing} unexpected={unexpected}")

    x = box["x"].to(device)
    y = box["y"].to(device) if box["y"] is not None else None

    first_fwd: list = []
    first_bwd: list = []
    names = {m: n for n, m in model.named_modules() if n}

    def fwd_hook(mod, inp, out):
        if first_fwd:
            return
        outs = out if isinstance(out, (tuple, list)) else (out,)
        bad = sum(_nonfinite(o) for o in outs)
        if bad:
            first_fwd.append((names.get(mod, "?"), bad))

    def bwd_hook(mod, gin, gout):
        # full-backward hooks fire in reverse topological order; the LAST
        # recorded one is the earliest layer, so keep appending
        bad = sum(_nonfinite(g) for g in gout if g is not None)
        if bad:
            first_bwd.append((names.get(mod, "?"), bad))

    handles = []
    for m in names:
        handles.append(m.register_forward_hook(fwd_hook))
        handles.append(m.register_full_backward_hook(bwd_hook))

    ac = (torch.autocast(device_type="cuda", dtype=torch.bfloat16)
          if args.bf16 and device.type == "cuda" else torch.enable_grad())
    with ac:
        out = model.train_losses(x, y)
    print("== replayed components:")
    for k in ("loss", "bpp", "H_real", "pc_loss", "d_loss", "reg",
              "loss_sinet"):
        v = out.get(k)
        if isinstance(v, torch.Tensor) and v.numel() == 1:
            print(f"   {k:12s} {float(v.detach()):.6g}")
    print(f"== first non-finite FORWARD output: "
          f"{first_fwd[0] if first_fwd else 'none'}")

    import warnings
    with warnings.catch_warnings():
        # modules whose inputs don't require grad (first layer) fire the
        # full-backward hook on module OUTPUTS only — that is exactly what
        # we inspect, so the advisory warning is noise here
        warnings.filterwarnings(
            "ignore", message="Full backward hook is firing")
        out["loss"].backward()
    grad_bad = {n: _nonfinite(p.grad) for n, p in model.named_parameters()
                if p.grad is not None and _nonfinite(p.grad)}
    print(f"== earliest non-finite BACKWARD grad-output: "
          f"{first_bwd[-1] if first_bwd else 'none'} "
          f"({len(first_bwd)} modules total)")
    print(f"== non-finite parameter grads: {grad_bad or 'none'}")
    for h in handles:
        h.remove()


if __name__ == "__main__":
    main()
