"""Datasets: sharded-NPY image datasets + synthetic data.

Capability parity with the reference's TFRecord dataset runtime (ref
src/training/dataset.py [R]): streaming, shuffling,
optional labels. The on-disk format is ours (simple .npy shards written
by prepare_data.py — uint8 NCHW), since TFRecords are a TF1 artifact.
SyntheticDataset generates random images of the target shape on the fly
(the north-star benchmarks run on synthetic data, BASELINE.json:5).

Images are stored/produced as uint8 [C,H,W]; normalization to [-1,1]
float happens on the GPU in the training loop (uint8 over PCIe/xGMI, 4x
less host->device traffic than float32).
"""

from __future__ import annotations

import glob
import json
import os

import numpy as np
import torch
from torch.utils.data import Dataset


class SyntheticDataset(Dataset):
    """Random uint8 images + optional labels; deterministic per index."""

    def __init__(self, resolution=256, channels=3, size=50000, label_dim=0,
                 seed=0):
        self.resolution = resolution
        self.channels = channels
        self.size = size
        self.label_dim = label_dim
        self.seed = seed

    def __len__(self):
        return self.size

    @property
    def image_shape(self):
        return (self.channels, self.resolution, self.resolution)

    def __getitem__(self, idx):
        rng = np.random.RandomState((self.seed * 1000003 + idx) % (2 ** 31))
        img = rng.randint(0, 256, size=self.image_shape, dtype=np.uint8)
        label = np.zeros(self.label_dim, dtype=np.float32)
        if self.label_dim:
            label[rng.randint(self.label_dim)] = 1.0
        return torch.from_numpy(img), torch.from_numpy(label)


class ShapesDataset(Dataset):
    """CLEVR-like structured synthetic data: 2-5 anti-aliased colored
    shapes (circle / square / triangle) on a gray floor, deterministic
    per index. Unlike uniform-noise SyntheticDataset, this has real
    image statistics (edges, flat color regions, occlusion), so GAN
    losses and FID respond to training — used for the convergence
    evidence runs and the CLEVR-64 config."""

    PALETTE = np.array([
        [173, 35, 35], [42, 75, 215], [29, 105, 20], [129, 74, 25],
        [129, 38, 192], [160, 160, 160], [129, 197, 122], [157, 175, 255],
        [41, 208, 208], [255, 146, 51], [255, 238, 51], [233, 222, 187],
        [255, 205, 243], [255, 255, 255]], dtype=np.float32)

    def __init__(self, resolution=64, channels=3, size=50000, label_dim=0,
                 seed=0):
        self.resolution = resolution
        self.channels = channels
        self.size = size
        self.label_dim = label_dim
        self.seed = seed
        r = resolution
        ys, xs = np.mgrid[0:r, 0:r].astype(np.float32)
        self._ys, self._xs = (ys + 0.5) / r, (xs + 0.5) / r

    def __len__(self):
        return self.size

    @property
    def image_shape(self):
        return (self.channels, self.resolution, self.resolution)

    def __getitem__(self, idx):
        rng = np.random.RandomState((self.seed * 1000003 + idx) % (2 ** 31))
        r = self.resolution
        ys, xs = self._ys, self._xs
        # gray floor with a slight vertical gradient (CLEVR-ish)
        base = 90.0 + 40.0 * rng.rand()
        img = np.empty((r, r, 3), dtype=np.float32)
        img[:] = (base + 25.0 * ys)[..., None]
        aa = 1.5 / r  # anti-alias width
        n_shapes = rng.randint(2, 6)
        for _ in range(n_shapes):
            cx, cy = 0.15 + 0.7 * rng.rand(2)
            rad = 0.06 + 0.12 * rng.rand()
            color = self.PALETTE[rng.randint(len(self.PALETTE))] \
                * (0.7 + 0.3 * rng.rand())
            kind = rng.randint(3)
            dx, dy = xs - cx, ys - cy
            if kind == 0:       # circle: signed distance to boundary
                sd = np.sqrt(dx * dx + dy * dy) - rad
            elif kind == 1:     # square (axis-aligned, Chebyshev)
                sd = np.maximum(np.abs(dx), np.abs(dy)) - rad
            else:               # triangle pointing up
                sd = np.maximum(dy - rad,
                                np.abs(dx) * 1.732 - (rad - dy))
            mask = np.clip(0.5 - sd / (2 * aa), 0.0, 1.0)
            # cheap top-lit shading inside the shape
            shade = 1.0 - 0.35 * np.clip((dy + rad) / (2 * rad), 0, 1)
            img = img * (1 - mask[..., None]) \
                + (color * shade[..., None]) * mask[..., None]
        img = np.clip(img, 0, 255).astype(np.uint8)
        img = np.repeat(img[..., :1], 3, axis=-1) if self.channels == 1 \
            else img[..., :self.channels]
        img = np.ascontiguousarray(img.transpose(2, 0, 1))
        label = np.zeros(self.label_dim, dtype=np.float32)
        if self.label_dim:
            label[min(n_shapes - 2, self.label_dim - 1)] = 1.0
        return torch.from_numpy(img), torch.from_numpy(label)


class ShardedNpyDataset(Dataset):
    """Directory of `shard-*.npy` files (uint8 [N,C,H,W]) + meta.json,
    as written by prepare_data.py. Optional labels.npy [total, label_dim]."""

    def __init__(self, path, resolution=None):
        self.path = path
        with open(os.path.join(path, "meta.json")) as f:
            self.meta = json.load(f)
        self.shard_files = sorted(glob.glob(os.path.join(path, "shard-*.npy")))
        if not self.shard_files:
            raise FileNotFoundError(f"no shard-*.npy under {path}")
        self.shard_sizes = self.meta["shard_sizes"]
        self.offsets = np.cumsum([0] + self.shard_sizes)
        self.resolution = resolution or self.meta["resolution"]
        assert self.resolution == self.meta["resolution"], \
            "on-disk resolution mismatch; re-run prepare_data.py"
        self._cache = {}
        lbl = os.path.join(path, "labels.npy")
        self.labels = np.load(lbl) if os.path.exists(lbl) else None
        self.label_dim = 0 if self.labels is None else self.labels.shape[1]

    def __len__(self):
        return int(self.offsets[-1])

    @property
    def image_shape(self):
        return (self.meta["channels"], self.resolution, self.resolution)

    def _shard(self, i):
        arr = self._cache.get(i)
        if arr is None:
            arr = np.load(self.shard_files[i], mmap_mode="r")
            self._cache[i] = arr
        return arr

    def __getitem__(self, idx):
        s = int(np.searchsorted(self.offsets, idx, side="right")) - 1
        img = np.array(self._shard(s)[idx - self.offsets[s]])
        label = (self.labels[idx].astype(np.float32) if self.labels is not None
                 else np.zeros(0, dtype=np.float32))
        return torch.from_numpy(img), torch.from_numpy(label)


class MirroredDataset(Dataset):
    """x-flip augmentation, reference style [R]: the dataset is doubled
    and indices past the base length return the horizontally flipped
    image (deterministic per index, so resume stays reproducible)."""

    def __init__(self, base):
        self.base = base

    def __len__(self):
        return 2 * len(self.base)

    @property
    def resolution(self):
        return self.base.resolution

    @property
    def image_shape(self):
        return self.base.image_shape

    @property
    def label_dim(self):
        return getattr(self.base, "label_dim", 0)

    def __getitem__(self, idx):
        n = len(self.base)
        img, label = self.base[idx % n]
        if idx >= n:
            img = torch.flip(img, dims=[-1])
        return img, label


def load_dataset(dataset=None, data_dir=None, resolution=256, channels=3,
                 synthetic_size=50000, seed=0, mirror_augment=False,
                 label_dim=0):
    """Resolve a dataset spec: a prepared dir if it exists, else
    synthetic (`dataset="shapes"` selects the structured shapes data).
    label_dim > 0 makes the synthetic datasets emit one-hot labels
    (prepared datasets carry their own labels.npy)."""
    ds = None
    if data_dir and dataset:
        path = os.path.join(data_dir, dataset)
        if os.path.isdir(path):
            ds = ShardedNpyDataset(path, resolution=resolution)
    if ds is None and dataset in ("shapes", "clevr-synth"):
        ds = ShapesDataset(resolution=resolution, channels=channels,
                           size=synthetic_size, seed=seed,
                           label_dim=label_dim)
    if ds is None:
        ds = SyntheticDataset(resolution=resolution, channels=channels,
                              size=synthetic_size, seed=seed,
                              label_dim=label_dim)
    if mirror_augment:
        ds = MirroredDataset(ds)
    return ds


def make_loader(ds, batch_size, rank=0, world_size=1, num_workers=2, seed=0):
    """Infinite shuffled loader, sharded across ranks."""
    sampler = torch.utils.data.distributed.DistributedSampler(
        ds, num_replicas=world_size, rank=rank, shuffle=True, seed=seed,
        drop_last=True) if world_size > 1 else None
    loader = torch.utils.data.DataLoader(
        ds, batch_size=batch_size, shuffle=(sampler is None),
        sampler=sampler, num_workers=num_workers, pin_memory=True,
        drop_last=True, persistent_workers=num_workers > 0)

    def forever():
        epoch = 0
        while True:
            if sampler is not None:
                sampler.set_epoch(epoch)
            yield from loader
            epoch += 1

    return forever()


def normalize_images(imgs_uint8, device, dtype=torch.float32):
    """uint8 [B,C,H,W] -> float in [-1, 1] on device."""
    x = imgs_uint8.to(device, non_blocking=True)
    return x.to(dtype).div(127.5).sub(1.0)
