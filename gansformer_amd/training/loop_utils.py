"""Shared helpers for the training loop / trainer."""

from __future__ import annotations

import torch


def lazy_adam(params, lr, betas, eps, reg_interval):
    """Adam with lazy-regularization-adjusted hyperparams (interval I:
    effective lr and betas scaled by c = I/(I+1), mirroring the lazy-reg
    scheme the reference's loss used [R])."""
    if reg_interval and reg_interval > 1:
        c = reg_interval / (reg_interval + 1)
        lr = lr * c
        betas = tuple(b ** c for b in betas)
    # foreach batches the per-parameter update math into a few wide
    # kernels instead of ~8 tiny launches per tensor
    return torch.optim.Adam(params, lr=lr, betas=betas, eps=eps,
                            foreach=True)


@torch.no_grad()
def ema_update(Gs, G, beta):
    sp = dict(G.named_parameters())
    ema_params, src_params = [], []
    for name, p_ema in Gs.named_parameters():
        ema_params.append(p_ema)
        src_params.append(sp[name].detach())
    torch._foreach_lerp_(ema_params, src_params, 1.0 - beta)
    sb = dict(G.named_buffers())
    for name, b_ema in Gs.named_buffers():
        b_ema.copy_(sb[name])
