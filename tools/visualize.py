"""Visualization: sample grids, latent interpolation, attention heatmaps.

Parity with ref src/training/visualize.py [R] — the
attention-map dumps per latent component are the GANsformer's signature
output.
"""

from __future__ import annotations

import os

import numpy as np
import torch

from gansformer_amd.models.networks import BipartiteLayer
from gansformer_amd.training.snapshot import _write_png, save_image_grid


def _colorize(a: np.ndarray) -> np.ndarray:
    """[H,W] in [0,1] -> uint8 heat map [H,W,3]."""
    a = np.clip(a, 0, 1)
    r = np.clip(1.5 - np.abs(4 * a - 3), 0, 1)
    g = np.clip(1.5 - np.abs(4 * a - 2), 0, 1)
    b = np.clip(1.5 - np.abs(4 * a - 1), 0, 1)
    return (np.stack([r, g, b], -1) * 255).astype(np.uint8)


@torch.no_grad()
def save_attention_maps(Gs, z, outdir, truncation_psi=0.7):
    """Dump, for each attention layer and latent component, the simplex
    attention heatmap over image positions."""
    os.makedirs(outdir, exist_ok=True)
    layers = [m for m in Gs.modules() if isinstance(m, BipartiteLayer)]
    for m in layers:
        m.capture_attn = True
    try:
        img = Gs(z, truncation_psi=truncation_psi, noise_mode="const")
    finally:
        for m in layers:
            m.capture_attn = False
    save_image_grid(img, os.path.join(outdir, "attn_source.png"))
    for li, m in enumerate(layers):
        a = getattr(m, "last_attn", None)
        if a is None:
            continue
        a = a[0].float().cpu().numpy()  # [H, W, k]
        H, W, k = a.shape
        lo, hi = a.min(), a.max()
        for c in range(k):
            heat = _colorize((a[:, :, c] - lo) / max(hi - lo, 1e-8))
            _write_png(os.path.join(
                outdir, f"attn-layer{li}-res{H}-comp{c:02d}.png"), heat)


@torch.no_grad()
def save_interpolation(Gs, outdir, steps=8, seed=0, truncation_psi=0.7):
    """Linear interpolation between two latent draws."""
    os.makedirs(outdir, exist_ok=True)
    device = next(Gs.parameters()).device
    g = torch.Generator(device="cpu").manual_seed(seed)
    z0 = Gs.sample_z(1, generator=g).to(device)
    z1 = Gs.sample_z(1, generator=g).to(device)
    zs = torch.cat([torch.lerp(z0, z1, t)
                    for t in torch.linspace(0, 1, steps)])
    img = Gs(zs, truncation_psi=truncation_psi, noise_mode="const")
    save_image_grid(img, os.path.join(outdir, "interp.png"),
                    grid_w=steps, grid_h=1)
