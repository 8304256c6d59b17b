"""conv2d with custom, infinitely differentiable autograd.

Replaces the reference's cuDNN convs (TF 1.14 library kernels, SURVEY.md
K3/L1) with our own MFMA implicit-GEMM HIP kernels on gfx950:

  * 3x3 stride-1/2 shared-weight convs -> tap-major LDS-slab kernels
    (csrc/conv2d_slab.hip, conv2d_s2.hip, conv2d_wgrad_slab.hip)
  * 2x-upsampling 3x3 conv -> parity decomposition at input resolution
    (csrc/conv2d_up2.hip, 1/4 the MACs of upsample-then-conv)
  * 1x1 convs -> plain hipBLASLt GEMMs
  * everything else (f32, odd shapes, per-sample weights) -> the generic
    per-tap implicit-GEMM kernels in csrc/conv2d.hip

Gradient structure (all pieces are themselves autograd Functions or plain
differentiable torch ops, so R1 / path-length double-backward replays are
exact):
    dX = conv_fwd(zero_stuff(dY, stride), transpose_flip(W), stride=1)
    dW = wgrad(X, dY)            (bilinear; its own Function)
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from . import native
from .upfirdn2d import upfirdn2d


def _eager_conv2d(x, w, stride, pad):
    orig_dtype = x.dtype
    if x.device.type == "cpu" and x.dtype not in (torch.float32, torch.float64):
        x, w = x.float(), w.float()
    if w.ndim == 4:
        y = F.conv2d(x, w, stride=stride, padding=pad)
    else:
        B, O, I, kh, kw = w.shape
        y = F.conv2d(
            x.reshape(1, B * x.shape[1], *x.shape[2:]),
            w.reshape(B * O, I, kh, kw),
            stride=stride, padding=pad, groups=B)
        y = y.reshape(B, O, *y.shape[2:])
    return y.to(orig_dtype)


def _eager_wgrad(x, dy, stride, pad, kh, kw, per_sample):
    orig_dtype = x.dtype
    if x.device.type == "cpu" and x.dtype not in (torch.float32, torch.float64):
        x, dy = x.float(), dy.float()
    B, I = x.shape[0], x.shape[1]
    O = dy.shape[1]
    unf = F.unfold(x, (kh, kw), padding=pad, stride=stride)  # [B, I*kh*kw, P]
    dyf = dy.reshape(B, O, -1)  # [B, O, P]
    dw = torch.einsum("bop,bip->boi", dyf, unf)  # [B, O, I*kh*kw]
    dw = dw.reshape(B, O, I, kh, kw)
    if not per_sample:
        dw = dw.sum(0)
    return dw.to(orig_dtype)


def _transpose_flip(w):
    """[.., O, I, kh, kw] -> [.., I, O, kh, kw] flipped spatially."""
    return w.transpose(-4, -3).flip([-2, -1])


def _id_filter(device):
    return torch.ones(1, 1, dtype=torch.float32, device=device)


def _conv_input_grad(dy, w, stride, pad, in_hw):
    """dX for a forward conv with (w, stride, pad) and input size in_hw."""
    kh, kw = w.shape[-2], w.shape[-1]
    H, W = in_hw
    Hout, Wout = dy.shape[-2], dy.shape[-1]
    wt = _transpose_flip(w)
    q0y, q0x = kh - 1 - pad, kw - 1 - pad
    q1y = H + kh - 1 - Hout * stride - q0y
    q1x = W + kw - 1 - Wout * stride - q0x
    if stride == 1 and q0y == q1y and q0x == q1x and q0y >= 0 and q0y == q0x:
        return conv2d_gradfix(dy, wt, stride=1, padding=q0y)
    # fold one unit of each pad into the conv so the stride-1 conv keeps
    # pad=1 and stays eligible for the 3x3 slab kernel
    fold = 1 if (kh == 3 and kw == 3 and min(q0y, q1y, q0x, q1x) >= 1) else 0
    z = upfirdn2d(dy, _id_filter(dy.device), up=stride,
                  padding=(q0y - fold, q1y - fold, q0x - fold, q1x - fold))
    return conv2d_gradfix(z, wt, stride=1, padding=fold)


def _is_1x1(kh, kw, stride, pad):
    return kh == 1 and kw == 1 and stride == 1 and pad == 0


class _Conv2dWgrad(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, dy, stride, pad, kh, kw, per_sample):
        ctx.save_for_backward(x, dy)
        ctx.params = (stride, pad, kh, kw, per_sample)
        # NOTE on fp32 operands below: every faulting variant of the
        # batch-64 D-backward abort (KNOWN_ISSUES.md) contained one of
        # these double-batched bf16 bmm calls, and every variant that
        # replaced exactly them with fp32 passed — the bf16 strided
        # batched hipBLASLt path at batch 64 is the prime suspect.
        # fp32 here also matches the slab wgrad's accumulation
        # precision, and these branches only serve small maps, so the
        # cost is negligible.
        if native.use_native(x, dy):
            if _is_1x1(kh, kw, stride, pad) and not per_sample:
                # 1x1 wgrad is a plain GEMM: dw = sum_b dY_b X_b^T
                # (hipBLASLt; the MFMA tile kernels waste M rows on the
                # skinny tRGB/fromRGB shapes)
                B, I = x.shape[0], x.shape[1]
                O = dy.shape[1]
                dw = torch.matmul(dy.reshape(B, O, -1).float(),
                                  x.reshape(B, I, -1).transpose(1, 2)
                                  .float())
                return dw.sum(0).reshape(O, I, 1, 1).to(x.dtype)
            if not per_sample and dy.shape[2] * dy.shape[3] <= 256:
                # small feature maps: im2col + batched GEMM (see fwd)
                B, I = x.shape[0], x.shape[1]
                O = dy.shape[1]
                unf = _unfold_batched(x, kh, kw, pad, stride)
                dw = torch.matmul(dy.reshape(B, O, -1).float(),
                                  unf.transpose(1, 2).float())
                return dw.sum(0).reshape(O, I, kh, kw).to(x.dtype)
            return native.require_ext().conv2d_wgrad(
                x.contiguous(), dy.contiguous(), stride, pad, kh, kw, per_sample)
        return _eager_wgrad(x, dy, stride, pad, kh, kw, per_sample)

    @staticmethod
    def backward(ctx, ddw):
        x, dy = ctx.saved_tensors
        stride, pad, kh, kw, per_sample = ctx.params
        d_x = d_dy = None
        if ctx.needs_input_grad[0]:
            d_x = _conv_input_grad(dy, ddw, stride, pad, x.shape[-2:])
        if ctx.needs_input_grad[1]:
            d_dy = conv2d_gradfix(x, ddw, stride=stride, padding=pad)
        return d_x, d_dy, None, None, None, None, None


def _out_hw(H, W, kh, kw, stride, pad):
    return (H + 2 * pad - kh) // stride + 1, (W + 2 * pad - kw) // stride + 1


def _unfold_batched(x, kh, kw, pad, stride):
    """F.unfold with the batch folded into channels: torch's im2col
    launches one kernel PER SAMPLE, which turns the small-conv GEMM
    routing into a launch storm; per-channel independence makes
    unfold([1, B*I, H, W]) identical in one launch.

    The fold is capped at B*I <= 16384 channels: beyond that the
    ROCm im2col path memory-faults (bisected 2026-09-14: batch 64 x
    512 ch = 32768 folded channels aborts with a write to a read-only
    page inside the D backward; 56 x 512 = 28672 is clean — see
    KNOWN_ISSUES.md for the full bisect). Above the cap we pay torch's
    per-sample launches; this branch only serves the small-feature-map
    tail, so the cost is bounded."""
    B, I, H, W = x.shape
    if B * I > 16384:
        return F.unfold(x, (kh, kw), padding=pad, stride=stride)
    u = F.unfold(x.reshape(1, B * I, H, W), (kh, kw), padding=pad,
                 stride=stride)
    return u.reshape(B, I * kh * kw, -1)


class _Conv2dFwd(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, stride, pad):
        ctx.save_for_backward(x, w)
        ctx.params = (stride, pad)
        if native.use_native(x, w):
            if w.ndim == 4 and _is_1x1(w.shape[-2], w.shape[-1], stride, pad):
                # 1x1 conv is a plain GEMM (hipBLASLt)
                B, I, H, W = x.shape
                O = w.shape[0]
                y = torch.matmul(w.reshape(O, I), x.reshape(B, I, H * W))
                return y.reshape(B, O, H, W)
            if w.ndim == 4:
                kh, kw = w.shape[-2], w.shape[-1]
                OH, OW = _out_hw(x.shape[2], x.shape[3], kh, kw, stride, pad)
                if OH * OW <= 256:
                    # small feature maps (res <= 8 layers, mbstd tail):
                    # the tiled kernels serialize K here; im2col +
                    # batched hipBLASLt GEMM parallelizes it instead
                    B, O = x.shape[0], w.shape[0]
                    unf = _unfold_batched(x, kh, kw, pad, stride)
                    y = torch.matmul(w.reshape(O, -1), unf)
                    return y.reshape(B, O, OH, OW)
            return native.require_ext().conv2d_fwd(
                x.contiguous(), w.contiguous(), stride, pad)
        return _eager_conv2d(x, w, stride, pad)

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        stride, pad = ctx.params
        dx = dw = None
        if ctx.needs_input_grad[0]:
            dx = _conv_input_grad(dy, w, stride, pad, x.shape[-2:])
        if ctx.needs_input_grad[1]:
            kh, kw = w.shape[-2], w.shape[-1]
            dw = _Conv2dWgrad.apply(x, dy, stride, pad, kh, kw, w.ndim == 5)
        return dx, dw, None, None


def conv2d_gradfix(x, w, stride=1, padding=0):
    """Conv2d (cross-correlation). x: [B,I,H,W]; w: [O,I,kh,kw] shared or
    [B,O,I,kh,kw] per-sample. Returns [B,O,Ho,Wo]."""
    assert x.ndim == 4 and w.ndim in (4, 5)
    return _Conv2dFwd.apply(x, w, int(stride), int(padding))


def _zero_stuff2(x):
    if x.is_cuda:
        # single-pass scatter kernel (upfirdn 1x1 filter, up 2) instead
        # of zero-fill + strided copy_ over a 4x tensor
        return upfirdn2d(x, _id_filter(x.device), up=2, padding=0)
    B, I, H, W = x.shape
    z = x.new_zeros(B, I, 2 * H, 2 * W)
    z[:, :, ::2, ::2] = x
    return z


class _Conv2dUp2(torch.autograd.Function):
    """y = conv2d(zero_stuff2(x), w, pad=1) -> [B,O,2H,2W].

    The native kernel computes the 4 output parity classes at input
    resolution (1/4 the MACs the reference's upsample-then-conv graph
    spent). Backward pieces are themselves autograd
    ops, so R1/path-length double-backward replays are exact:
        dx = conv2d(dy, transpose_flip(w), stride=2, pad=1)
        dw = wgrad(zero_stuff2(x), dy, pad=1)
    """

    @staticmethod
    def forward(ctx, x, w):
        ctx.save_for_backward(x, w)
        if native.use_native(x, w):
            if (w.ndim == 4 and x.dtype == torch.bfloat16
                    and w.shape[-1] == 3 and x.shape[1] % 32 == 0
                    and x.shape[2] % 8 == 0 and x.shape[3] % 16 == 0
                    and x.shape[2] >= 8 and x.shape[3] >= 16):
                return native.require_ext().conv2d_up2(
                    x.contiguous(), w.contiguous())
            # small/odd shapes: zero-stuff + our generic conv kernel
            # (never MIOpen: its naive bf16 conv is ~10x slower)
            return native.require_ext().conv2d_fwd(
                _zero_stuff2(x).contiguous(), w.contiguous(), 1, 1)
        return _eager_conv2d(_zero_stuff2(x), w, 1, 1)

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dx = dw = None
        if ctx.needs_input_grad[0]:
            dx = conv2d_gradfix(dy, _transpose_flip(w), stride=2, padding=1)
        if ctx.needs_input_grad[1]:
            dw = _Conv2dWgrad.apply(_zero_stuff2(x), dy, 1, 1, 3, 3, False)
        return dx, dw


def conv2d_up2(x, w):
    """2x-upsampling 3x3 conv (transposed-conv equivalent)."""
    assert x.ndim == 4 and w.ndim == 4 and w.shape[-2:] == (3, 3)
    return _Conv2dUp2.apply(x, w)
