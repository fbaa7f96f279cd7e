"""Loss/inference figure equivalents (reference src/utils.py:12-79)."""

import os

import numpy as np

from dsin_amd.utils.plots import plot_inference, plot_loss


def test_plot_loss_writes_png(tmp_path):
    out = plot_loss([10.0, 5.0, 3.0], [6.0, 4.0], [100, 200], [50, 100, 150],
                    300, best_val=4.0, best_iter=200, model_name="m",
                    out_path=str(tmp_path / "loss.png"))
    assert os.path.exists(out) and os.path.getsize(out) > 1000


def test_plot_inference_writes_png(tmp_path):
    rng = np.random.default_rng(0)
    imgs = [rng.uniform(0, 255, (3, 48, 64)) for _ in range(5)]
    out = plot_inference(*imgs, model_name="m", total_iterations=100,
                         cnt=10, bpp="0.02",
                         out_path=str(tmp_path / "inf.png"))
    assert os.path.exists(out) and os.path.getsize(out) > 1000
