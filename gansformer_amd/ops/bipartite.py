"""Bipartite attention core op: softmax(Q K^T / sqrt(d)) V.

The GANsformer's signature computation: image
tokens X (HW of them) attend to k latent components (Simplex direction,
softmax over k <= 32), and in Duplex mode the latents also attend back to
the image (softmax over HW). Both directions reduce to this op with the
roles of Q and K/V swapped.

Forward runs the fused HIP kernel on gfx950 (bipartite_attn.hip):
  * softmax-over-small-N path (N_kv <= 64): K/V staged in LDS, logits
    kept per-row, one pass.
  * softmax-over-long-N path: two-pass online-softmax partials.
Backward is composed from saved Q,K,V with differentiable torch ops
(GEMMs via hipBLASLt = library GEMMs), so path-length double-backward
through G's attention layers is exact. Hand-written backward kernels are
a later optimization.
"""

from __future__ import annotations

import math
import os

import torch

from . import native

# debug escape hatch: force the eager recompute backward everywhere
_FORCE_EAGER_BWD = os.environ.get("GFA_ATTN_BWD_EAGER") == "1"


def _up(t):
    """upcast low precision to fp32; leave fp32/fp64 alone"""
    return t.float() if t.dtype in (torch.bfloat16, torch.float16) else t


def _eager_attention(q, k, v, scale):
    a = torch.einsum("bqd,bkd->bqk", _up(q), _up(k)) * scale
    a = torch.softmax(a, dim=-1)
    out = torch.einsum("bqk,bke->bqe", a, _up(v))
    return out.to(v.dtype), a


class _BipartiteAttn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale):
        ctx.scale = scale
        ctx.native = native.use_native(q, k, v)
        if ctx.native:
            q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
            out, ml = native.require_ext().bipartite_attn_fwd(q, k, v, scale)
            if k.shape[1] > 64:
                # long-N: the fused backward needs out (for drow) + the
                # softmax stats. Small-N uses the eager recompute and
                # must not pin the [B,HW,E] output through backward.
                ctx.save_for_backward(q, k, v, out, ml)
            else:
                e = torch.empty(0, device=q.device)
                ctx.save_for_backward(q, k, v, e, e)
            return out
        e = torch.empty(0, device=q.device)
        out, _ = _eager_attention(q, k, v, scale)
        ctx.save_for_backward(q, k, v, e, e)
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, ml = ctx.saved_tensors
        scale = ctx.scale
        # Fused backward is routed for the LONG-N direction only
        # (duplex reverse, softmax over HW): measured 1.5-1.6x over the
        # eager recompute (gpurun_out/r02_attnbwd2.log). The small-N
        # fused kernel is correct but 0.7x (its 5-phase LDS dance can't
        # beat hipBLASLt's 400+ TF/s on the skinny recompute GEMMs), so
        # small-N keeps the eager composition.
        if (ctx.native and k.shape[1] > 64 and not _FORCE_EAGER_BWD
                and not torch.is_grad_enabled()):
            # rebuild A from the saved softmax stats; drow uses the
            # flash identity rowsum(dA*A) == rowsum(dO*O).
            dout_c = dout.contiguous()
            drow = (dout_c.float() * out.float()).sum(-1).contiguous()
            dq, dk, dv = native.require_ext().bipartite_attn_bwd(
                q, k, v, dout_c, drow, ml, scale)
            return dq, dk, dv, None
        # create_graph replay (path-length reg) or eager/CPU path:
        # recompute A with differentiable ops (double-backward-capable).
        # Logits/softmax stay fp32 (the numerics policy the forward
        # kernel implements); the surrounding GEMMs run in the input
        # dtype — hipBLASLt accumulates fp32 internally, and this avoids
        # materializing fp32 copies of the big [B,HW,*] operands.
        qf, kf = _up(q), _up(k)
        s = torch.einsum("bqd,bkd->bqk", qf, kf) * scale
        a = torch.softmax(s, dim=-1)
        a_lp = a.to(v.dtype)
        dv = torch.einsum("bqk,bqe->bke", a_lp, dout)
        daf = _up(torch.einsum("bqe,bke->bqk", dout, v))
        ds = a * (daf - (daf * a).sum(dim=-1, keepdim=True))
        ds_lp = ds.to(q.dtype)
        dq = torch.einsum("bqk,bkd->bqd", ds_lp, k) * scale
        dk = torch.einsum("bqk,bqd->bkd", ds_lp, q) * scale
        return dq.to(q.dtype), dk.to(k.dtype), dv.to(v.dtype), None


def bipartite_attention(q, k, v, scale=None, need_weights=False):
    """q: [B,Nq,D], k: [B,Nk,D], v: [B,Nk,E] -> out [B,Nq,E].

    With need_weights=True also returns the attention map [B,Nq,Nk]
    (eager path — used by tools/visualize for attention heatmaps).
    """
    assert q.ndim == k.ndim == v.ndim == 3
    assert q.shape[2] == k.shape[2] and k.shape[1] == v.shape[1]
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[2])
    if need_weights:
        out, a = _eager_attention(q, k, v, scale)
        return out, a
    if q.requires_grad or k.requires_grad or v.requires_grad:
        return _BipartiteAttn.apply(q, k, v, scale)
    # no-grad fast path
    if native.use_native(q, k, v):
        return native.require_ext().bipartite_attn(
            q.contiguous(), k.contiguous(), v.contiguous(), scale)
    return _eager_attention(q, k, v, scale)[0]
