"""Minibatch standard deviation layer (discriminator).

Computes, per batch-group and channel-feature-group, the mean over
(channels, H, W) of the per-element stddev across the group, and appends
it as constant feature maps.

The native HIP kernel (mbstd.hip) computes the group stats on BOTH the
training and inference paths; its backward is the analytic gradient
  d stats[m,f] / d x[g*M+m, c, hw] = (x - mean) / (G * n * s),
expressed in differentiable torch ops so R1 double-backward through D
 replays exactly.
"""

from __future__ import annotations

import torch

from . import native


class _MbStdStats(torch.autograd.Function):
    """stats[m, f] from the HIP kernel; analytic differentiable bwd."""

    @staticmethod
    def forward(ctx, x, G, F, eps):
        ctx.save_for_backward(x)
        ctx.params = (G, F, eps)
        return native.require_ext().mbstd(x.contiguous(), G, F, eps)

    @staticmethod
    def backward(ctx, dstats):
        (x,) = ctx.saved_tensors
        G, F, eps = ctx.params
        B, C, H, W = x.shape
        c = C // F
        n = c * H * W
        xf = x.reshape(G, B // G, F, c, H, W).to(torch.float32)
        mean = xf.mean(dim=0, keepdim=True)
        var = (xf - mean).square().mean(dim=0)
        s = (var + eps).sqrt()  # [B//G, F, c, H, W]
        dx = (xf - mean) * (dstats.to(torch.float32)
                            .reshape(1, B // G, F, 1, 1, 1)
                            / (G * n * s.unsqueeze(0)))
        return dx.reshape(B, C, H, W).to(x.dtype), None, None, None


def _eager_stats(x, G, F, eps):
    B, C, H, W = x.shape
    c = C // F
    y = x.reshape(G, -1, F, c, H, W).to(torch.float32)
    y = y - y.mean(dim=0)
    y = y.square().mean(dim=0)
    y = (y + eps).sqrt()
    return y.mean(dim=[2, 3, 4])  # [B//G, F]


def minibatch_stddev(x, group_size=4, num_channels=1, eps=1e-8):
    B, C, H, W = x.shape
    G = min(group_size, B)
    if B % G != 0:
        G = 1
    F = num_channels
    assert C % F == 0
    use_kernel = (
        x.is_cuda
        and native.have_ext()
        and x.dtype in (torch.float32, torch.bfloat16)
        and G <= 32
    )
    if use_kernel:
        stats = _MbStdStats.apply(x, G, F, eps)  # [B//G, F]
    else:
        stats = _eager_stats(x, G, F, eps)
    # batch index b = g*(B//G) + m gets the stats of chunk m (matches the
    # reshape(G, -1, ...) grouping above).
    maps = stats.reshape(1, -1, F, 1, 1).expand(G, B // G, F, H, W)
    maps = maps.reshape(B, F, H, W).to(x.dtype)
    return torch.cat([x, maps], dim=1)
