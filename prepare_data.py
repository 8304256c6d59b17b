#!/usr/bin/env python3
"""Prepare datasets as sharded .npy (parity with ref prepare_data.py,
SURVEY.md §2.1 #4 / §3.3 [R] — which downloaded FFHQ/CLEVR/LSUN/Cityscapes
and wrote TFRecords; this environment has no network, so input must be a
local directory of images (.npy / .npz arrays, raw uint8) or `--synthetic`
for a generated placeholder set).

Output layout under <data-dir>/<name>/:
    meta.json                {"resolution": R, "channels": C, "shard_sizes": [...]}
    shard-00000.npy ...      uint8 [N, C, R, R]
    labels.npy               optional float32 [total, label_dim]
"""

import argparse
import glob
import json
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def center_crop_resize(img: np.ndarray, res: int) -> np.ndarray:
    """img: uint8 HWC -> uint8 [C, res, res] (crop + nearest/area resize)."""
    h, w = img.shape[:2]
    s = min(h, w)
    y0, x0 = (h - s) // 2, (w - s) // 2
    img = img[y0:y0 + s, x0:x0 + s]
    # integer-factor area downsample, then nearest for the remainder
    if s >= res and s % res == 0:
        f = s // res
        img = img.reshape(res, f, res, f, -1).mean(axis=(1, 3)).astype(np.uint8)
    else:
        idx = (np.linspace(0, s - 1, res)).astype(np.int64)
        img = img[idx][:, idx]
    if img.ndim == 2:
        img = img[:, :, None]
    return img.transpose(2, 0, 1)


def iter_images(src):
    for path in sorted(glob.glob(os.path.join(src, "**", "*"), recursive=True)):
        if path.endswith(".npy"):
            arr = np.load(path)
            if arr.ndim == 3:
                yield arr
            elif arr.ndim == 4:
                yield from arr
        elif path.endswith(".npz"):
            data = np.load(path)
            for k in data.files:
                arr = data[k]
                if arr.ndim == 3:
                    yield arr
                elif arr.ndim == 4:
                    yield from arr


def main(argv=None):
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--name", required=True, help="dataset name (output subdir)")
    p.add_argument("--source", default=None,
                   help="directory of .npy/.npz images (HWC or NHWC uint8)")
    p.add_argument("--synthetic", type=int, default=0,
                   help="generate N synthetic images instead of reading --source")
    p.add_argument("--data-dir", default="datasets")
    p.add_argument("--res", type=int, default=256)
    p.add_argument("--channels", type=int, default=3)
    p.add_argument("--shard-size", type=int, default=4096)
    p.add_argument("--seed", type=int, default=0)
    args = p.parse_args(argv)

    out = os.path.join(args.data_dir, args.name)
    os.makedirs(out, exist_ok=True)
    shard_sizes = []
    buf = []
    shard_idx = 0

    def flush():
        nonlocal shard_idx
        if not buf:
            return
        arr = np.stack(buf)
        np.save(os.path.join(out, f"shard-{shard_idx:05d}.npy"), arr)
        shard_sizes.append(len(buf))
        shard_idx += 1
        buf.clear()

    if args.synthetic:
        rng = np.random.RandomState(args.seed)
        for _ in range(args.synthetic):
            buf.append(rng.randint(
                0, 256, size=(args.channels, args.res, args.res), dtype=np.uint8))
            if len(buf) >= args.shard_size:
                flush()
    else:
        if not args.source:
            p.error("--source or --synthetic required")
        for img in iter_images(args.source):
            buf.append(center_crop_resize(np.asarray(img, dtype=np.uint8),
                                          args.res)[: args.channels])
            if len(buf) >= args.shard_size:
                flush()
    flush()

    meta = dict(resolution=args.res, channels=args.channels,
                shard_sizes=shard_sizes)
    with open(os.path.join(out, "meta.json"), "w") as f:
        json.dump(meta, f)
    print(f"wrote {sum(shard_sizes)} images in {shard_idx} shards to {out}")


if __name__ == "__main__":
    main()
