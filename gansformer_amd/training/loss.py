"""GAN losses + lazy regularizers (ref src/training/loss.py [R]).

Non-saturating logistic loss for G and D, lazy R1 gradient penalty on D
(every d_reg_interval steps, scaled by the interval), lazy path-length
regularization on G (every g_reg_interval steps). The regularizers
differentiate through the generator/discriminator graphs built from our
custom ops — their double-backward entry points carry the
second-order replays.
"""

from __future__ import annotations

import math

import torch
import torch.nn.functional as F


def d_logistic_loss(real_logits, fake_logits):
    # E[softplus(-D(real))] + E[softplus(D(fake))]
    return F.softplus(-real_logits).mean() + F.softplus(fake_logits).mean()


def g_nonsaturating_loss(fake_logits):
    return F.softplus(-fake_logits).mean()


def r1_penalty(real_logits, real_images):
    """0.5 * E[ ||grad_x D(x)||^2 ]  (multiply by gamma at the call site)."""
    (grads,) = torch.autograd.grad(
        outputs=real_logits.sum(), inputs=real_images, create_graph=True)
    return grads.to(torch.float32).square().sum(dim=[1, 2, 3]).mean() * 0.5


class PathLengthRegularizer:
    """StyleGAN2 path-length regularization with running mean of path norms."""

    def __init__(self, decay=0.01, weight=2.0):
        self.decay = decay
        self.weight = weight
        self.pl_mean = None  # lazily created on the right device

    def __call__(self, fake_images, ws):
        B = fake_images.shape[0]
        hw = fake_images.shape[2] * fake_images.shape[3]
        noise = torch.randn_like(fake_images) / math.sqrt(hw)
        (grads,) = torch.autograd.grad(
            outputs=(fake_images.to(torch.float32) * noise).sum(), inputs=ws,
            create_graph=True)
        # ws: [B, num_ws, L, w_dim] -> norm over (L, w_dim), mean over num_ws
        lengths = grads.to(torch.float32).square().sum(dim=[2, 3]).mean(dim=1).sqrt()
        if self.pl_mean is None:
            self.pl_mean = torch.zeros((), device=lengths.device)
        self.pl_mean = self.pl_mean.to(lengths.device)
        pl_mean = self.pl_mean.lerp(lengths.mean(), self.decay)
        self.pl_mean = pl_mean.detach()
        return (lengths - pl_mean).square().mean() * self.weight

    def state_dict(self):
        return {"pl_mean": None if self.pl_mean is None else self.pl_mean.item()}

    def load_state_dict(self, state):
        v = state.get("pl_mean")
        self.pl_mean = None if v is None else torch.tensor(float(v))
