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
    return torch.optim.Adam(params, lr=lr, betas=betas, eps=eps)


@torch.no_grad()
def ema_update(Gs, G, beta):
    sp = dict(G.named_parameters())
    for name, p_ema in Gs.named_parameters():
        p_ema.lerp_(sp[name].detach().to(p_ema.dtype), 1.0 - beta)
    sb = dict(G.named_buffers())
    for name, b_ema in Gs.named_buffers():
        b_ema.copy_(sb[name])
