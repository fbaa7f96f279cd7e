import os

import numpy as np
import torch
import pytest

from dsin_amd.data import SyntheticStereo, read_pair_list, read_png, write_png
from dsin_amd.data.provider import PairCropper


def test_png_roundtrip(tmp_path):
    rng = np.random.default_rng(0)
    img = rng.integers(0, 256, (17, 23, 3), dtype=np.uint8)
    p = str(tmp_path / "t.png")
    write_png(p, img)
    back = read_png(p)
    assert np.array_equal(back, img)


def test_png_gray_roundtrip(tmp_path):
    rng = np.random.default_rng(1)
    img = rng.integers(0, 256, (9, 11), dtype=np.uint8)
    p = str(tmp_path / "g.png")
    write_png(p, img)
    assert np.array_equal(read_png(p)[..., 0], img)


def test_pair_list(tmp_path):
    f = tmp_path / "pairs.txt"
    f.write_text("a/x1.png\na/y1.png\nb/x2.png\nb/y2.png\n")
    pairs = read_pair_list(str(f), root="/data/")
    assert pairs == [("/data/a/x1.png", "/data/a/y1.png"),
                     ("/data/b/x2.png", "/data/b/y2.png")]


def test_cropper_joint():
    rng = np.random.default_rng(0)
    cropper = PairCropper(8, 8, do_flips=True, rng=rng)
    x = np.arange(16 * 16 * 3, dtype=np.uint8).reshape(16, 16, 3)
    y = x.copy()
    for _ in range(5):
        xc, yc = cropper.random(x, y)
        assert xc.shape == (8, 8, 3)
        assert np.array_equal(xc, yc)  # joint crop+flip keeps x/y aligned


def test_cropper_center():
    rng = np.random.default_rng(0)
    cropper = PairCropper(4, 4, do_flips=False, rng=rng)
    x = np.zeros((8, 8, 3), np.uint8)
    x[2:6, 2:6] = 1
    xc, _ = cropper.center(x, x)
    assert xc.sum() == 4 * 4 * 3


def test_synthetic_shapes_and_determinism():
    g1 = SyntheticStereo(64, 96, batch_size=2, seed=7)
    g2 = SyntheticStereo(64, 96, batch_size=2, seed=7)
    x1, y1 = g1.next_batch()
    x2, y2 = g2.next_batch()
    assert x1.shape == (2, 3, 64, 96)
    assert torch.equal(x1, x2) and torch.equal(y1, y2)
    assert x1.min() >= 0 and x1.max() <= 255
    # x and y are correlated but not identical
    assert not torch.equal(x1, y1)
