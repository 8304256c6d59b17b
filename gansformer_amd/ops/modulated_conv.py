"""Modulated conv2d (StyleGAN2 weight modulation + demodulation).

Capability parity with the reference's modulated conv (built in the TF
graph around cuDNN grouped conv, ref src/training/networks.py [R],
SURVEY.md K3). Here the per-sample modulated weight tensor [B,O,I,kh,kw]
is fed directly to our per-sample implicit-GEMM conv kernel
(conv2d_gradfix) — no grouped-conv reshaping trick.

Modulation/demodulation math runs in fp32 (bf16-safe), the convolution
runs in the activation dtype on the MFMA path.
"""

from __future__ import annotations

import torch

from .conv2d_grad import conv2d_gradfix
from .upfirdn2d import upsample2d, upfirdn2d


def modulated_conv2d(
    x,                  # [B, I, H, W]
    weight,             # [O, I, kh, kw]
    styles,             # [B, I] per-sample per-input-channel scales
    demodulate=True,
    up=1,
    down=1,
    resample_filter=None,  # 2D FIR tensor (setup_filter) when up/down > 1
    padding=None,          # default: 'same' for the kernel size
):
    B, I, H, W = x.shape
    O, I2, kh, kw = weight.shape
    assert I2 == I and styles.shape == (B, I)
    if padding is None:
        padding = (kh - 1) // 2

    w = weight.to(torch.float32)
    s = styles.to(torch.float32)
    if demodulate:
        # pre-normalize for low-precision safety: bound |w|*|s| ~ 1
        w = w * w.square().mean([1, 2, 3], keepdim=True).rsqrt()
        s = s / s.abs().amax(dim=1, keepdim=True).clamp(min=1e-8)
    w = w.unsqueeze(0) * s.reshape(B, 1, I, 1, 1)  # [B,O,I,kh,kw]
    if demodulate:
        d = (w.square().sum(dim=[2, 3, 4]) + 1e-8).rsqrt()  # [B,O]
        w = w * d.reshape(B, O, 1, 1, 1)
    w = w.to(x.dtype)

    if up > 1:
        x = upsample2d(x, resample_filter, up=up)
    if down > 1:
        # blur (same-size, with the downsample pad baked in), then strided conv
        fh = resample_filter.shape[0]
        p0 = (fh - down + 1) // 2 + padding
        p1 = (fh - down) // 2 + padding
        x = upfirdn2d(x, resample_filter, padding=(p0, p1, p0, p1))
        return conv2d_gradfix(x, w, stride=down, padding=0)
    return conv2d_gradfix(x, w, stride=1, padding=padding)
