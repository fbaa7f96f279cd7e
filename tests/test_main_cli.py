"""End-to-end CLI smoke: train + validate + checkpoint + test on synthetic
data (the reference entry-point contract, src/main.py)."""

import os
import shutil
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_main_cli_end_to_end(tmp_path):
    cfg = open(os.path.join(REPO, "run_configs", "ae_run_configs")).read()
    cfg = cfg.replace("iterations = 300000", "iterations = 2")
    cfg = cfg.replace("crop_size = (320, 960)", "crop_size = (64, 96)")
    cfg = cfg.replace("y_patch_size = (20, 24)", "y_patch_size = (16, 16)")
    cfg = cfg.replace("root_data = ''", "root_data = 'synthetic'")
    cfg = cfg.replace("validate_every = 100000", "validate_every = 2")
    cfg = cfg.replace("show_every = 1000", "show_every = 1")
    cfg = cfg.replace("test_model = False", "test_model = True")
    d = str(tmp_path)
    os.makedirs(os.path.join(d, "run_configs"))
    with open(os.path.join(d, "run_configs", "ae_run_configs"), "w") as f:
        f.write(cfg)
    shutil.copy(os.path.join(REPO, "run_configs", "pc_run_configs"),
                os.path.join(d, "run_configs"))
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "main.py"),
         "-ae_config", os.path.join(d, "run_configs", "ae_run_configs"),
         "-pc_config", os.path.join(d, "run_configs", "pc_run_configs"),
         "--metrics", os.path.join(d, "m.jsonl"), "--plots"],
        cwd=d, env=env, capture_output=True, text=True, timeout=900)
    assert r.returncode == 0, r.stderr[-2000:]
    weights = os.listdir(os.path.join(d, "weights"))
    assert any(w.startswith("target_bpp0.02_sinet_") for w in weights)
    assert any(w.startswith("last_saved_") for w in weights)
    assert any(w.startswith("configs_") for w in weights)
    # test images + loss lists written
    img_dirs = os.listdir(os.path.join(d, "images", "images"))
    assert img_dirs
    pngs = os.listdir(os.path.join(d, "images", "images", img_dirs[0]))
    assert any(p.endswith("bpp.png") for p in pngs)
    lists = os.listdir(os.path.join(d, "images", "loss_lists"))
    assert {n.split("_")[0] for n in lists} >= {"bpp", "l1", "psnr", "mse",
                                               "msssim", "pearson"}
    assert os.path.exists(os.path.join(d, "m.jsonl"))
    # --plots wrote the loss figure and the 5-panel inference figure
    figs = [f for f in os.listdir(os.path.join(d, "images"))
            if f.endswith(".png")]
    assert any(f.startswith("loss_") for f in figs), figs
    assert any(f.startswith("inference_") for f in figs), figs
