"""Modulated conv2d (StyleGAN2 weight modulation + demodulation).

Capability parity with the reference's modulated conv (built in the TF
graph around cuDNN grouped conv, ref src/training/networks.py [R],
SURVEY.md K3), re-derived for MI355X memory behavior.

Because convolution is linear in its input,

    conv(x, w * s[b,i] * d[b,o])  ==  d[b,o] * conv(x * s[b,i], w)

so instead of materializing a per-sample weight tensor [B,O,I,kh,kw]
(151 MB/layer at 512ch/batch-32 — it made weights the dominant HBM
traffic of the whole step), we scale the *input* by the styles, run a
single SHARED-weight implicit-GEMM conv on MFMA (the base weight is
4.7 MB and stays L2-resident across the whole grid), and apply the
demodulation factor to the output. The demod factor itself is computed
without materializing anything: sum_{i,kh,kw} (w*s)^2 over taps equals
(s^2) @ W2^T with W2[o,i] = sum_{kh,kw} w^2 — one tiny [B,I]x[I,O] GEMM.

All pieces are plain differentiable torch ops around conv2d_gradfix, so
R1 / path-length double-backward is exact by construction.
Modulation/demodulation math runs in fp32 (bf16-safe), the convolution
runs in the activation dtype on the MFMA path.
"""

from __future__ import annotations

import torch

from .conv2d_grad import conv2d_gradfix, conv2d_up2
from .upfirdn2d import upfirdn2d


def modulated_conv2d(
    x,                  # [B, I, H, W]
    weight,             # [O, I, kh, kw]
    styles,             # [B, I] per-sample per-input-channel scales
    demodulate=True,
    up=1,
    down=1,
    resample_filter=None,  # 2D FIR tensor (setup_filter) when up/down > 1
    padding=None,          # default: 'same' for the kernel size
    return_demod=False,    # skip applying d; return (y, d) for a fused
                           # epilogue (mod_bias_act)
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
        # demod factor via the W2 trick: no [B,O,I,kh,kw] intermediate
        w2 = w.square().sum(dim=[2, 3])            # [O, I]
        d = (s.square() @ w2.t() + 1e-8).rsqrt()   # [B, O]

    x = x * s.reshape(B, I, 1, 1).to(x.dtype)
    w = w.to(x.dtype)

    if up > 1:
        # transposed conv at input resolution (1/4 the MACs of
        # upsample-then-conv), then the FIR blur — convs commute, so in
        # the interior this equals blurring first like the reference did
        assert up == 2 and kh == 3 and padding == 1
        t = conv2d_up2(x, w)
        fh = resample_filter.shape[0]
        y = upfirdn2d(t, resample_filter,
                      padding=(fh // 2, fh // 2 - 1, fh // 2, fh // 2 - 1),
                      gain=up * up)
    elif down > 1:
        # same-size blur (pads sum to fh-1, output stays even), strided
        # conv keeps its own padding -> stride-2 slab kernel eligible;
        # interior-identical to baking the pad into the blur
        fh = resample_filter.shape[0]
        p0 = (fh - down + 1) // 2
        p1 = fh - 1 - p0
        x = upfirdn2d(x, resample_filter, padding=(p0, p1, p0, p1))
        y = conv2d_gradfix(x, w, stride=down, padding=padding)
    else:
        y = conv2d_gradfix(x, w, stride=1, padding=padding)

    if demodulate:
        if return_demod:
            return y, d
        y = y * d.reshape(B, O, 1, 1).to(y.dtype)
    elif return_demod:
        return y, torch.ones(B, O, device=y.device, dtype=torch.float32)
    return y
