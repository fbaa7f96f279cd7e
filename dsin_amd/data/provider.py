"""Data pipeline: KITTI stereo path-pair lists + synthetic generator.

Mirrors the reference Dataset semantics (/root/reference/src/DataProvider.py):
path files list x/y PNG pairs on ALTERNATING lines (:119-126); train samples
get a joint random (crop_h, crop_w) 6-channel crop of the concatenated pair
plus a joint random left-right flip, and x is then re-cropped to crop_size
(a no-op when the sizes match, :32-60); val/test get a joint center crop
(:62-94). Tensors come out NCHW float32 in 0..255 (:189-199).

The torch-native replacement for tf.data: a background-thread prefetcher over
a numpy loader (PNG decode is pure Python here — see png.py — so the
benchmark path uses SyntheticStereo, as BASELINE.json prescribes for this
offline environment: synthetic KITTI-shaped data, stated in bench output).
"""

from __future__ import annotations

import os
import queue
import threading
from typing import Iterator, List, Tuple

import numpy as np
import torch

from .png import read_png


def read_pair_list(path: str, root: str = "") -> List[Tuple[str, str]]:
    with open(path) as f:
        lines = [root + ln.strip() for ln in f if ln.strip()]
    return list(zip(lines[0::2], lines[1::2]))


class PairCropper:
    """Joint crop + flip logic (train) and center crop (eval)."""

    def __init__(self, crop_h: int, crop_w: int, do_flips: bool, rng: np.random.Generator):
        self.crop_h, self.crop_w, self.do_flips = crop_h, crop_w, do_flips
        self.rng = rng

    def random(self, x: np.ndarray, y: np.ndarray) -> Tuple[np.ndarray, np.ndarray]:
        h, w, _ = x.shape
        top = int(self.rng.integers(0, h - self.crop_h + 1))
        left = int(self.rng.integers(0, w - self.crop_w + 1))
        xs = x[top:top + self.crop_h, left:left + self.crop_w]
        ys = y[top:top + self.crop_h, left:left + self.crop_w]
        if self.do_flips and self.rng.random() < 0.5:
            xs, ys = xs[:, ::-1], ys[:, ::-1]
        return xs, ys

    def center(self, x: np.ndarray, y: np.ndarray) -> Tuple[np.ndarray, np.ndarray]:
        h, w, _ = x.shape
        top = (h - self.crop_h) // 2
        left = (w - self.crop_w) // 2
        return (x[top:top + self.crop_h, left:left + self.crop_w],
                y[top:top + self.crop_h, left:left + self.crop_w])


def _to_nchw(batch: List[np.ndarray]) -> torch.Tensor:
    arr = np.stack(batch).astype(np.float32)
    return torch.from_numpy(arr).permute(0, 3, 1, 2).contiguous()


class Dataset:
    """File-backed dataset with the reference's train/val/test split views."""

    def __init__(self, config, data_paths_dir: str, seed: int = 0,
                 prefetch: int = 2):
        self.config = config
        self.crop_h, self.crop_w = config.crop_size
        self.batch_size = config.batch_size if config.AE_only else 1
        root = config.root_data
        self.train_pairs = read_pair_list(
            os.path.join(data_paths_dir, config.file_path_train), root)
        self.val_pairs = read_pair_list(
            os.path.join(data_paths_dir, config.file_path_val), root)
        self.test_pairs = read_pair_list(
            os.path.join(data_paths_dir, config.file_path_test), root)
        self.rng = np.random.default_rng(seed)
        self.cropper = PairCropper(self.crop_h, self.crop_w, config.do_flips, self.rng)
        self._train_iter = None
        self._val_iter = None
        self._test_iter = None
        self._prefetch = prefetch

    def get_data_size(self):
        return self.val_pairs, self.test_pairs

    def _load_pair(self, pair):
        x = read_png(pair[0])[..., :3]
        y = read_png(pair[1])[..., :3]
        return x, y

    def _train_gen(self) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
        while True:
            order = self.rng.permutation(len(self.train_pairs))
            xb, yb = [], []
            for i in order:
                x, y = self._load_pair(self.train_pairs[i])
                xc, yc = self.cropper.random(x, y)
                xb.append(xc)
                yb.append(yc)
                if len(xb) == self.batch_size:
                    yield _to_nchw(xb), _to_nchw(yb)
                    xb, yb = [], []

    def _eval_gen(self, pairs) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
        while True:
            xb, yb = [], []
            for pair in pairs:
                x, y = self._load_pair(pair)
                xc, yc = self.cropper.center(x, y)
                xb.append(xc)
                yb.append(yc)
                if len(xb) == self.batch_size:
                    yield _to_nchw(xb), _to_nchw(yb)
                    xb, yb = [], []

    @staticmethod
    def _prefetched(gen, depth: int):
        q: "queue.Queue" = queue.Queue(maxsize=depth)

        def worker():
            for item in gen:
                q.put(item)

        t = threading.Thread(target=worker, daemon=True)
        t.start()
        while True:
            yield q.get()

    def get_data_for_train(self):
        if self._train_iter is None:
            self._train_iter = self._prefetched(self._train_gen(), self._prefetch)
        return next(self._train_iter)

    def get_data_for_val(self):
        if self._val_iter is None:
            self._val_iter = self._eval_gen(self.val_pairs)
        return next(self._val_iter)

    def get_data_for_test(self):
        if self._test_iter is None:
            self._test_iter = self._eval_gen(self.test_pairs)
        return next(self._test_iter)


class SyntheticStereo:
    """Synthetic KITTI-shaped correlated stereo pairs for benchmarks.

    y is x horizontally shifted by a per-image disparity plus noise — enough
    structure that the NCC search does real work (varied argmax positions)
    while requiring no dataset on disk. Values uniform-ish in 0..255 float32,
    matching decoded-PNG scale. Deterministic per (seed, index).
    """

    def __init__(self, crop_h: int, crop_w: int, batch_size: int = 1,
                 seed: int = 1234, device: str = "cpu"):
        self.h, self.w, self.n = crop_h, crop_w, batch_size
        self.seed = seed
        self.device = device
        self._step = 0

    def next_batch(self) -> Tuple[torch.Tensor, torch.Tensor]:
        g = torch.Generator().manual_seed(self.seed + self._step)
        self._step += 1
        n, h, w = self.n, self.h, self.w
        # smooth random field: upsampled low-res noise -> natural-ish stats
        base = torch.rand(n, 3, h // 8 + 2, w // 8 + 2, generator=g)
        img = torch.nn.functional.interpolate(
            base, size=(h, w + 64), mode="bilinear", align_corners=False)
        img = img * 220.0 + torch.rand(n, 3, h, w + 64, generator=g) * 35.0
        # standardize each image to fixed KITTI-like global statistics:
        # real KITTI has stable per-image stats (that is what makes the
        # reference's FIXED normalization work, src/AE.py:222-248); without
        # this the raw field's per-image mean/contrast swing makes batch-1
        # BN behave like instance norm at train but mismatch its running
        # stats at eval, inflating eval-mode bpp on synthetic data
        mu = img.mean(dim=(2, 3), keepdim=True)
        sd = img.std(dim=(2, 3), keepdim=True).clamp_min(1e-3)
        img = (img - mu) / sd * 70.0 + 96.0
        shift = int(torch.randint(4, 48, (1,), generator=g))
        x = img[..., 64:64 + w]
        y = img[..., 64 - shift:64 - shift + w].clone()
        y = y + torch.randn(n, 3, h, w, generator=g) * 2.0
        x = x.clamp(0, 255).to(self.device)
        y = y.clamp(0, 255).to(self.device)
        return x, y


class SyntheticDataset:
    """Dataset-compatible synthetic source (train/val/test views) for runs
    without KITTI on disk (`root_data = synthetic` in the ae config). Val and
    test iterate fixed seeds so metrics are reproducible."""

    def __init__(self, config, n_train: int = 1576, n_val: int = 16,
                 n_test: int = 16):
        crop_h, crop_w = config.crop_size
        self.batch_size = config.batch_size if config.AE_only else 1
        self.train_pairs = [("synthetic", str(i)) for i in range(n_train)]
        self.val_pairs = [("synthetic", str(i)) for i in range(n_val)]
        self.test_pairs = [("synthetic", str(i)) for i in range(n_test)]
        self._train = SyntheticStereo(crop_h, crop_w, self.batch_size, seed=1)
        self._val = SyntheticStereo(crop_h, crop_w, self.batch_size, seed=2)
        self._test = SyntheticStereo(crop_h, crop_w, self.batch_size, seed=3)
        self._n_val, self._n_test = n_val, n_test

    def get_data_size(self):
        return self.val_pairs, self.test_pairs

    def get_data_for_train(self):
        return self._train.next_batch()

    def get_data_for_val(self):
        if self._val._step >= self._n_val:
            self._val._step = 0
        return self._val.next_batch()

    def get_data_for_test(self):
        if self._test._step >= self._n_test:
            self._test._step = 0
        return self._test.next_batch()


def make_dataset(config, data_paths_dir: str, seed: int = 0):
    """Dataset factory: the file-backed Dataset when the lists AND the
    images they reference are present, synthetic otherwise (stated loudly).
    """
    import warnings
    if str(config.root_data) == "synthetic":
        return SyntheticDataset(config)
    train_list = os.path.join(data_paths_dir, config.file_path_train)
    if not os.path.exists(train_list):
        warnings.warn(f"{train_list} not found - using synthetic data")
        return SyntheticDataset(config)
    pairs = read_pair_list(train_list, config.root_data)
    if not pairs or not os.path.exists(pairs[0][0]):
        probe = pairs[0][0] if pairs else train_list
        warnings.warn(
            f"data list {train_list} present but images missing "
            f"(probed {probe}); set root_data to the KITTI root to train "
            "on real data - using synthetic data")
        return SyntheticDataset(config)
    return Dataset(config, data_paths_dir, seed=seed)
