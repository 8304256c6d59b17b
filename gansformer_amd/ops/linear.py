"""Linear layers with split-K weight gradients.

hipBLASLt runs the tall-K reduction GEMMs of token-side FC backwards
(dw = dY^T X with K = B*HW up to 2M) at 24-30 TF/s on MI355X; manually
splitting K into batched chunks and summing runs the same contraction at
200-770 TF/s (tools/gemmbench2.py). These Functions keep forward on the
library fast path and express backward with the split-K formulation.
All pieces are plain torch ops, so double-backward (path-length reg
through G's attention projections) composes exactly.
"""

from __future__ import annotations

import torch

from . import native

_SPLITK_MIN_M = 65536   # below this the direct GEMM is fine
_CHUNK = 16384          # target K-chunk per split


def _use_skinny(x2, n, k):
    """MEASURED OFF (r02): hipBLASLt runs these tall-skinny projection
    shapes at 258-765 TF/s on plain contiguous matmuls
    (gpurun_out/r02_gemmsk.log) — the r01 ~70 TF/s reading came from a
    different call path. Our gemm_skinny kernel reaches only 75-90 TF/s,
    so routing through it regressed the whole step (145 -> 131 imgs/s).
    The kernel stays in csrc/ with its numerics tests as the MFMA GEMM
    reference implementation, but nothing routes to it."""
    return False


def _splitk_tn(a, b):
    """a: [M, K1], b: [M, K2] -> a^T @ b [K1, K2] via K-split bmm+sum."""
    M = a.shape[0]
    S = max(1, min(64, M // _CHUNK))
    while M % S:
        S -= 1
    if S <= 1:
        return a.t() @ b
    a3 = a.reshape(S, M // S, a.shape[1])
    b3 = b.reshape(S, M // S, b.shape[1])
    return torch.bmm(a3.transpose(1, 2), b3).sum(0)


class _LinearSplitK(torch.autograd.Function):
    """y = x @ w^T for x [..., K], w [N, K]. Forward and dx run the
    tall-skinny MFMA kernel when the shape qualifies; dw stays on the
    split-K torch composition (already 200-770 TF/s)."""

    @staticmethod
    def forward(ctx, x, w):
        ctx.save_for_backward(x, w)
        x2 = x.reshape(-1, x.shape[-1])
        if _use_skinny(x2, w.shape[0], w.shape[1]) and w.is_contiguous():
            y = native.require_ext().gemm_skinny(x2.contiguous(),
                                                 w, True)
            return y.reshape(*x.shape[:-1], w.shape[0])
        return x.matmul(w.t())

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dx = dw = None
        if ctx.needs_input_grad[0]:
            dy2 = dy.reshape(-1, dy.shape[-1])
            if (not torch.is_grad_enabled()
                    and _use_skinny(dy2, w.shape[1], w.shape[0])
                    and w.is_contiguous()):
                # dx = dy @ W : B operand is [K=N_out, N=K_in] row-major
                dx = native.require_ext().gemm_skinny(
                    dy2.contiguous(), w, False).reshape(x.shape)
            else:
                dx = dy.matmul(w)
        if ctx.needs_input_grad[1]:
            dyf = dy.reshape(-1, dy.shape[-1])
            xf = x.reshape(-1, x.shape[-1])
            dw = _splitk_tn(dyf, xf)
        return dx, dw


def linear_nobias(x, w):
    """x [..., K] @ w[N, K]^T with split-K weight gradient on GPU."""
    if x.is_cuda and x.numel() // x.shape[-1] >= _SPLITK_MIN_M:
        return _LinearSplitK.apply(x, w)
    return x.matmul(w.t())


class _LinearTransposedOut(torch.autograd.Function):
    """y[B, N, M] = w[N, K] @ u[B, M, K]^T (strided GEMM, no permute
    copy); backward's dw is the per-batch bmm sum (split-K by batch)."""

    @staticmethod
    def forward(ctx, u, w):
        ctx.save_for_backward(u, w)
        return torch.matmul(w, u.transpose(1, 2))

    @staticmethod
    def backward(ctx, dy):
        u, w = ctx.saved_tensors
        du = dw = None
        if ctx.needs_input_grad[0]:
            du = torch.matmul(dy.transpose(1, 2), w)
        if ctx.needs_input_grad[1]:
            dw = torch.bmm(dy, u).sum(0)
        return du, dw


def linear_transposed(u, w):
    """w[N,K] @ u[B,M,K]^T -> [B, N, M]."""
    if u.is_cuda:
        return _LinearTransposedOut.apply(u, w)
    return torch.matmul(w, u.transpose(1, 2))
