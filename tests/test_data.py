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


def _write_kitti_like_tree(tmp_path, n_pairs=2, h=40, w=56):
    """A mini on-disk KITTI-shaped tree: real PNG pairs + a reference-format
    list file (x/y paths on alternating lines)."""
    rng = np.random.default_rng(3)
    img_dir = tmp_path / "imgs"
    img_dir.mkdir(exist_ok=True)
    lines = []
    for i in range(n_pairs):
        x = rng.integers(0, 256, (h, w, 3), dtype=np.uint8)
        y = np.roll(x, 4, axis=1)  # correlated side image
        write_png(str(img_dir / f"{i}_x.png"), x)
        write_png(str(img_dir / f"{i}_y.png"), y)
        lines += [f"imgs/{i}_x.png", f"imgs/{i}_y.png"]
    lists = tmp_path / "data_paths"
    lists.mkdir(exist_ok=True)
    for name in ("train", "val", "test"):
        (lists / f"pairs_{name}.txt").write_text("\n".join(lines) + "\n")
    return str(lists) + os.sep


def test_dataset_real_files_end_to_end(tmp_path):
    """Decode real PNG pairs through the file-backed Dataset: train batches
    (joint crop+flip) and center-cropped val batches, no synthetic fallback."""
    from dsin_amd import config as cm
    from dsin_amd.data import make_dataset
    from dsin_amd.data.provider import Dataset
    here = os.path.dirname(os.path.abspath(__file__))
    cfg, _ = cm.parse(os.path.join(here, "..", "run_configs", "ae_run_configs"))
    cfg.crop_size = (32, 48)
    cfg.root_data = str(tmp_path) + os.sep
    cfg.file_path_train = "pairs_train.txt"
    cfg.file_path_val = "pairs_val.txt"
    cfg.file_path_test = "pairs_test.txt"
    lists_dir = _write_kitti_like_tree(tmp_path)
    data = make_dataset(cfg, lists_dir)
    assert isinstance(data, Dataset)  # no silent synthetic fallback
    x, y = data.get_data_for_train()
    assert x.shape == (1, 3, 32, 48) and y.shape == (1, 3, 32, 48)
    assert x.dtype == torch.float32 and 0 <= x.min() and x.max() <= 255
    xv, yv = data.get_data_for_val()
    assert xv.shape == (1, 3, 32, 48)
    # center crop is deterministic: same pair -> same tensor across epochs
    for _ in range(len(data.val_pairs) - 1):
        data.get_data_for_val()
    xv2, _ = data.get_data_for_val()
    assert torch.equal(xv, xv2)


def test_make_dataset_missing_images_falls_back(tmp_path):
    """Lists present but images absent -> loud synthetic fallback."""
    from dsin_amd import config as cm
    from dsin_amd.data import make_dataset
    from dsin_amd.data.provider import SyntheticDataset
    here = os.path.dirname(os.path.abspath(__file__))
    cfg, _ = cm.parse(os.path.join(here, "..", "run_configs", "ae_run_configs"))
    cfg.root_data = str(tmp_path / "nowhere") + os.sep
    lists = tmp_path / "data_paths"
    lists.mkdir()
    (lists / cfg.file_path_train).write_text("a/x.png\na/y.png\n")
    with pytest.warns(UserWarning, match="images missing"):
        data = make_dataset(cfg, str(lists) + os.sep)
    assert isinstance(data, SyntheticDataset)


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
