"""Fused bias + activation (+gain +clamp) with full double-backward.

Capability parity with the reference's `fused_bias_act` CUDA op (ref
src/dnnlib/tflib/ops/fused_bias_act.{py,cu} [R]): one fused
elementwise pass computing y = clamp(act(x + b) * gain), and gradient
variants selected by a `grad` flag that recompute the activation slope
from the saved OUTPUT y (not the input), so the same kernel serves
forward, backward, and the second-order replays needed by R1 and
path-length regularization.

Autograd structure:
    _FusedBiasAct.forward       -> kernel grad=0, saves y
    _FusedBiasAct.backward      -> _FusedActGrad.apply(dy, y)  (+ bias sum)
    _FusedActGrad.forward       -> kernel grad=1: dx = dy * slope(y)
    _FusedActGrad.backward      -> d(dy) = _FusedActGrad.apply(ddx, y)
                                   d(y)  = 0 for piecewise-linear acts
The chain is exact to all orders for {linear, relu, lrelu} (the acts used
by the networks), because dx is linear in dy with an a.e.-constant
coefficient in y.
"""

from __future__ import annotations

import math

import torch

from . import native

# act name -> (code, default_alpha, default_gain, has_curvature)
activation_defs = {
    "linear": (0, 0.0, 1.0, False),
    "relu": (1, 0.0, math.sqrt(2.0), False),
    "lrelu": (2, 0.2, math.sqrt(2.0), False),
    "tanh": (3, 0.0, 1.0, True),
    "sigmoid": (4, 0.0, 1.0, True),
}


def _eager_act(x: torch.Tensor, act: str, alpha: float) -> torch.Tensor:
    if act == "linear":
        return x
    if act == "relu":
        return torch.relu(x)
    if act == "lrelu":
        return torch.nn.functional.leaky_relu(x, alpha)
    if act == "tanh":
        return torch.tanh(x)
    if act == "sigmoid":
        return torch.sigmoid(x)
    raise ValueError(f"unknown activation {act!r}")


def _eager_fwd(x, b, act, alpha, gain, clamp):
    if b is not None:
        shape = [1] * x.ndim
        shape[1] = -1
        x = x + b.reshape(shape)
    y = _eager_act(x, act, alpha) * gain
    if clamp is not None:
        y = y.clamp(-clamp, clamp)
    return y


def _eager_grad1(dy, y, act, alpha, gain, clamp):
    """dx = dy * d(act)/dx reconstructed from the saved output y."""
    if clamp is not None:
        live = (y.abs() < clamp).to(dy.dtype)
    else:
        live = None
    if act == "linear":
        dx = dy * gain
    elif act == "relu":
        dx = dy * gain * (y > 0).to(dy.dtype)
    elif act == "lrelu":
        slope = torch.where(y > 0, torch.full_like(dy, gain), torch.full_like(dy, gain * alpha))
        dx = dy * slope
    elif act == "tanh":
        t = y / gain
        dx = dy * gain * (1 - t * t)
    elif act == "sigmoid":
        s = y / gain
        dx = dy * gain * s * (1 - s)
    else:
        raise NotImplementedError(f"grad of activation {act!r}")
    if live is not None:
        dx = dx * live
    return dx


class _FusedActGrad(torch.autograd.Function):
    @staticmethod
    def forward(ctx, dy, y, act, alpha, gain, clamp):
        ctx.save_for_backward(y)
        ctx.params = (act, alpha, gain, clamp)
        if native.use_native(dy, y):
            code = activation_defs[act][0]
            return native.require_ext().fba(
                dy.contiguous(), torch.empty(0, device=dy.device, dtype=dy.dtype),
                y.contiguous(), code, 1, alpha, gain,
                float("inf") if clamp is None else clamp)
        return _eager_grad1(dy, y, act, alpha, gain, clamp)

    @staticmethod
    def backward(ctx, ddx):
        (y,) = ctx.saved_tensors
        act, alpha, gain, clamp = ctx.params
        if activation_defs[act][3]:
            raise NotImplementedError(
                f"double-backward through curved activation {act!r}")
        d_dy = _FusedActGrad.apply(ddx, y, act, alpha, gain, clamp)
        # d/dy of dy*slope(y) is zero a.e. for piecewise-linear acts.
        return d_dy, None, None, None, None, None


class _FusedBiasAct(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, b, act, alpha, gain, clamp):
        if native.use_native(x):
            code = activation_defs[act][0]
            bb = b.contiguous() if b is not None else torch.empty(0, device=x.device, dtype=x.dtype)
            y = native.require_ext().fba(
                x.contiguous(), bb, torch.empty(0, device=x.device, dtype=x.dtype),
                code, 0, alpha, gain, float("inf") if clamp is None else clamp)
        else:
            y = _eager_fwd(x, b, act, alpha, gain, clamp)
        ctx.save_for_backward(y)
        ctx.params = (act, alpha, gain, clamp, b is not None, x.ndim)
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        act, alpha, gain, clamp, has_bias, ndim = ctx.params
        dx = _FusedActGrad.apply(dy, y, act, alpha, gain, clamp)
        db = None
        if has_bias:
            db = dx.sum([d for d in range(ndim) if d != 1])
        return dx, db, None, None, None, None


def fused_bias_act(x, b=None, act="lrelu", alpha=None, gain=None, clamp=None):
    """y = clamp(act(x + b[c]) * gain). b broadcast over channel dim 1."""
    if act not in activation_defs:
        raise ValueError(f"unknown activation {act!r}")
    _, def_alpha, def_gain, _ = activation_defs[act]
    alpha = float(def_alpha if alpha is None else alpha)
    gain = float(def_gain if gain is None else gain)
    clamp = None if clamp is None or clamp < 0 else float(clamp)
    return _FusedBiasAct.apply(x, b, act, alpha, gain, clamp)


# Alias matching common naming.
bias_act = fused_bias_act


def _eager_mod_bias_act(x, d, noise, sigma, b, act, alpha, gain, clamp):
    v = x * d.reshape(d.shape[0], d.shape[1], 1, 1).to(x.dtype)
    if noise is not None:
        v = v + noise.to(v.dtype) * sigma.to(v.dtype)
    return _eager_fwd(v, b, act, alpha, gain, clamp)


class _ModBiasAct(torch.autograd.Function):
    """SynthesisLayer epilogue: y = clamp(act(x*d + noise*sigma + b)*gain)
    in ONE pass (was demod-scale, noise addcmul, and bias_act — three
    full-tensor passes after every modulated conv)."""

    @staticmethod
    def forward(ctx, x, d, noise, sigma, b, act, alpha, gain, clamp):
        d32 = d if d.dtype == torch.float64 else d.to(torch.float32)
        if (native.use_native(x) and x.dtype == torch.bfloat16
                and x.shape[2] * x.shape[3] % 8 == 0
                and act in ("linear", "lrelu")):
            code = activation_defs[act][0]
            empty = torch.empty(0, device=x.device, dtype=x.dtype)
            y = native.require_ext().fba_mod(
                x.contiguous(), d32.contiguous(),
                noise.contiguous().to(x.dtype) if noise is not None else empty,
                b if b is not None else empty,
                sigma, code, alpha, gain,
                float("inf") if clamp is None else clamp)
        else:
            y = _eager_mod_bias_act(x, d, noise, sigma, b, act, alpha, gain,
                                    clamp)
        ctx.save_for_backward(x, d32, y,
                              noise if noise is not None else x.new_empty(0),
                              sigma)
        ctx.params = (act, alpha, gain, clamp, b is not None,
                      noise is not None)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, d32, y, noise, sigma = ctx.saved_tensors
        act, alpha, gain, clamp, has_bias, has_noise = ctx.params
        B, C = x.shape[0], x.shape[1]
        g = _FusedActGrad.apply(dy, y, act, alpha, gain, clamp)
        dx = dd = dsigma = db = None
        if ctx.needs_input_grad[0]:
            dx = g * d32.reshape(B, C, 1, 1).to(g.dtype)
        rdt = torch.float64 if x.dtype == torch.float64 else torch.float32
        if ctx.needs_input_grad[1]:
            dd = (g * x).sum(dim=[2, 3], dtype=rdt)
        if has_noise and ctx.needs_input_grad[3]:
            gc = g.sum(dim=1, keepdim=True, dtype=rdt)
            dsigma = (gc * noise.to(rdt)).sum().reshape(sigma.shape)
        if has_bias and ctx.needs_input_grad[4]:
            db = g.sum(dim=[0, 2, 3], dtype=rdt)
        return dx, dd, None, dsigma, db, None, None, None, None


def mod_bias_act(x, d, noise=None, sigma=None, b=None, act="lrelu",
                 alpha=None, gain=None, clamp=None):
    """Fused demodulation-scale + noise + bias + activation.

    x: [B,C,H,W] conv output BEFORE demodulation; d: [B,C] demod factors;
    noise: [B,1,H,W] or None; sigma: noise strength (0-dim tensor)."""
    spec = activation_defs[act]
    if alpha is None:
        alpha = spec[1]
    if gain is None:
        gain = spec[2]
    if sigma is None:
        sigma = torch.zeros((), device=x.device)
    if noise is not None and noise.requires_grad:
        # _ModBiasAct.backward returns no grad for the noise input; a
        # differentiable noise tensor would get its gradient silently
        # dropped, so refuse it loudly.
        raise ValueError("mod_bias_act: noise must not require grad "
                         "(its gradient is not computed)")
    return _ModBiasAct.apply(x, d, noise, sigma, b, act, float(alpha),
                             float(gain), clamp)
