"""Snapshots: image grids + network pkls + full resume state.

Parity with ref src/training/misc.py [R] (save image grids, pkl helpers,
`reals.png`). Interchange checkpoints are `.pkl` (pkl_compat layout);
full training state (optimizers, EMA schedule, pl_mean, RNG) goes in a
sibling `.pt` so `--resume-pkl` restores bit-compatible training.
"""

from __future__ import annotations

import os

import numpy as np
import torch

from .. import pkl_compat


def images_to_grid(imgs: torch.Tensor, grid_w=None, grid_h=None):
    """[-1,1] float [B,C,H,W] -> uint8 HWC grid array."""
    B, C, H, W = imgs.shape
    if grid_w is None:
        grid_w = max(1, int(np.sqrt(B)))
    if grid_h is None:
        grid_h = (B + grid_w - 1) // grid_w
    x = (imgs.detach().float().cpu().clamp(-1, 1) + 1) * 127.5
    x = x.to(torch.uint8).numpy()
    grid = np.zeros((grid_h * H, grid_w * W, C), dtype=np.uint8)
    for i in range(min(B, grid_w * grid_h)):
        r, c = divmod(i, grid_w)
        grid[r * H:(r + 1) * H, c * W:(c + 1) * W] = x[i].transpose(1, 2, 0)
    return grid


def _write_png(path, arr):
    """Minimal PNG writer (no Pillow in this environment). arr: HWC uint8."""
    import struct
    import zlib

    h, w = arr.shape[:2]
    c = arr.shape[2] if arr.ndim == 3 else 1
    color_type = {1: 0, 3: 2, 4: 6}[c]
    raw = b"".join(
        b"\x00" + arr[i].tobytes() for i in range(h))

    def chunk(tag, data):
        out = struct.pack(">I", len(data)) + tag + data
        return out + struct.pack(">I", zlib.crc32(tag + data) & 0xFFFFFFFF)

    header = struct.pack(">IIBBBBB", w, h, 8, color_type, 0, 0, 0)
    with open(path, "wb") as f:
        f.write(b"\x89PNG\r\n\x1a\n")
        f.write(chunk(b"IHDR", header))
        f.write(chunk(b"IDAT", zlib.compress(raw, 6)))
        f.write(chunk(b"IEND", b""))


def save_image_grid(imgs: torch.Tensor, path, grid_w=None, grid_h=None):
    _write_png(path, images_to_grid(imgs, grid_w, grid_h))


def save_snapshot(run_dir, cur_nimg, G, D, Gs, extra_state=None):
    kimg = cur_nimg // 1000
    pkl_path = os.path.join(run_dir, f"network-snapshot-{kimg:06d}.pkl")
    pkl_compat.save_network_pkl(pkl_path, G, D, Gs)
    if extra_state is not None:
        torch.save(extra_state, pkl_path.replace(".pkl", ".pt"))
    return pkl_path


def load_resume(pkl_path, G, D, Gs, map_location="cpu"):
    """Load network weights from pkl into existing modules; return extra
    state dict from the sibling .pt if present."""
    states = pkl_compat.load_network_states(pkl_path)
    assert len(states) == 3
    for net, st in zip((G, D, Gs), states):
        pkl_compat.load_variables(net, st["variables"])
    pt = pkl_path.replace(".pkl", ".pt")
    if os.path.exists(pt):
        return torch.load(pt, map_location=map_location, weights_only=False)
    return None
