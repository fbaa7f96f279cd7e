"""Checkpoint manager with the reference's staged-scope layout.

The reference saves one TF checkpoint under ``weights/<model_name>/model``
with model_name = 'target_bpp{t}_{AE_only|sinet}_{ddmmyyyy-HHMM}'
(src/main.py:141-165) and restores scope-filtered variable lists
(src/AE.py:158-175):

  * always: encoder (incl. centers), decoder, imgcomp (PC);
  * + training-step (optimizer state, global step) when load_train_step;
  * + siNetwork when continuing SI training, or when doing SI inference
    (test_model and not train_model and not AE_only).

We keep that contract with torch state_dict groups {encoder, decoder,
imgcomp, siNetwork, training-step} in a single ``model.pt`` plus the
``last_saved_<name>.txt`` / ``configs_<name>.txt`` sidecars.
"""

from __future__ import annotations

import datetime
import os
from typing import Optional

import torch


def model_name_for(ae_config, now: Optional[str] = None) -> str:
    target_bpp = ae_config.H_target / (64.0 / ae_config.num_chan_bn)
    mode = "_AE_only_" if ae_config.AE_only else "_sinet_"
    now = now or datetime.datetime.today().strftime("%d%m%Y-%H%M")
    return "target_bpp" + str(target_bpp) + mode + now


def save(model, optimizers, global_step: int, root_weights: str, model_name: str,
         iteration: int, total_iterations: int, best_val: float,
         ae_config=None, pc_config=None, save_config: bool = True) -> str:
    path = os.path.join(root_weights, model_name)
    os.makedirs(path, exist_ok=True)
    blob = {name: mod.state_dict() for name, mod in model.state_groups().items()}
    blob["training-step"] = {
        "global_step": global_step,
        "optimizers": [opt.state_dict() for opt in optimizers],
    }
    torch.save(blob, os.path.join(path, "model.pt"))

    with open(os.path.join(root_weights, f"last_saved_{model_name}.txt"), "w") as f:
        f.write(f"{path}\nlast saved iteration number: {iteration}/{total_iterations}"
                f"\nlast saved val loss: {best_val}")
    cfg_path = os.path.join(root_weights, f"configs_{model_name}.txt")
    if save_config and ae_config is not None and not os.path.exists(cfg_path):
        with open(cfg_path, "w") as f:
            f.write("#  ae configs:\n" + str(ae_config))
            f.write("\n\n#  pc configs:\n" + str(pc_config))
    return path


def load(model, optimizers, load_path: str, ae_config) -> int:
    """Staged restore per the reference's scope rules. Returns the restored
    global step (0 when training-step is not loaded)."""
    blob = torch.load(os.path.join(load_path, "model.pt"),
                      map_location="cpu", weights_only=False)
    groups = model.state_groups()
    want = ["encoder", "decoder", "imgcomp"]
    load_train_step = bool(ae_config.load_train_step)
    if load_train_step and not ae_config.AE_only:
        want.append("siNetwork")
    elif ae_config.test_model and not ae_config.train_model and not ae_config.AE_only:
        want.append("siNetwork")
    for name in want:
        if name in groups and name in blob:
            groups[name].load_state_dict(blob[name])
    step = 0
    if load_train_step and "training-step" in blob:
        ts = blob["training-step"]
        step = int(ts.get("global_step", 0))
        for opt, sd in zip(optimizers, ts.get("optimizers", [])):
            try:
                opt.load_state_dict(sd)
            except (KeyError, ValueError, RuntimeError) as e:
                import warnings
                warnings.warn(
                    "optimizer state not restored (checkpoint written by a "
                    f"different optimizer implementation): {e}")
    return step
