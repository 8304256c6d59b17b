"""Minibatch standard deviation layer (discriminator).

Computes, per batch-group and channel-feature-group, the mean over
(channels, H, W) of the per-element stddev across the group, and appends
it as constant feature maps.

Training path uses differentiable torch ops (required: R1 double-backward
flows through D including this layer). The native HIP kernel (mbstd.hip)
computes the group stats on the no-grad inference path.
"""

from __future__ import annotations

import torch

from . import native


def minibatch_stddev(x, group_size=4, num_channels=1, eps=1e-8):
    B, C, H, W = x.shape
    G = min(group_size, B)
    if B % G != 0:
        G = 1
    F = num_channels
    assert C % F == 0
    use_kernel = (
        not torch.is_grad_enabled()
        and x.is_cuda
        and native.have_ext()
        and x.dtype in (torch.float32, torch.bfloat16)
    )
    if use_kernel:
        stats = native.require_ext().mbstd(x.contiguous(), G, F, eps)  # [B//G, F]
    else:
        c = C // F
        y = x.reshape(G, -1, F, c, H, W).to(torch.float32)
        y = y - y.mean(dim=0)
        y = y.square().mean(dim=0)
        y = (y + eps).sqrt()
        stats = y.mean(dim=[2, 3, 4])  # [B//G, F]
    # batch index b = g*(B//G) + m gets the stats of chunk m (matches the
    # reshape(G, -1, ...) grouping above).
    maps = stats.reshape(1, -1, F, 1, 1).expand(G, B // G, F, H, W)
    maps = maps.reshape(B, F, H, W).to(x.dtype)
    return torch.cat([x, maps], dim=1)
