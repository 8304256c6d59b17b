#!/usr/bin/env python3
"""Microbenchmark for the hot HIP kernels (run on a GPU box).

    python tools/kbench.py [conv|wgrad|attn|upfirdn|all]

Prints per-shape wall time and achieved TFLOP/s (or GB/s for memory-bound
ops) so kernel tuning can proceed without full-model noise.
"""

import sys
import time

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])

import gansformer_amd._C as C  # noqa: E402


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters


def bench_conv(dtype=torch.bfloat16):
    dev = "cuda:0"
    shapes = [
        # (B, I, O, H, k, stride, note) — flagship layer shapes
        (8, 128, 128, 256, 3, 1, "res256 conv1"),
        (8, 256, 128, 256, 3, 1, "res256 conv0(post-up)"),
        (8, 256, 256, 128, 3, 1, "res128 conv1"),
        (8, 512, 512, 64, 3, 1, "res64 conv"),
        (8, 512, 512, 32, 3, 1, "res32 conv"),
        (8, 512, 512, 16, 3, 1, "res16 conv"),
        (8, 128, 3, 256, 1, 1, "tRGB"),
        (8, 128, 256, 256, 3, 2, "D down 256->128"),
        (32, 512, 512, 64, 3, 1, "res64 conv B32"),
        (32, 256, 256, 128, 3, 1, "res128 conv1 B32"),
        (32, 128, 128, 256, 3, 1, "res256 conv1 B32"),
    ]
    for B, I, O, H, k, s, note in shapes:
        x = torch.randn(B, I, H, H, device=dev, dtype=dtype)
        w = torch.randn(O, I, k, k, device=dev, dtype=dtype) * 0.05
        OH = (H + 2 * (k // 2) - k) // s + 1
        t = timeit(lambda: C.conv2d_fwd(x, w, s, k // 2))
        fl = 2.0 * B * O * I * k * k * OH * OH
        print(f"conv fwd {note:22s} B{B} {I:4d}->{O:4d} @{H:4d} k{k} s{s}: "
              f"{t * 1e3:8.3f} ms  {fl / t / 1e12:7.1f} TF")


def bench_wgrad(dtype=torch.bfloat16):
    dev = "cuda:0"
    shapes = [
        (8, 128, 128, 256, 3, 1, "res256 conv1"),
        (8, 512, 512, 64, 3, 1, "res64 conv"),
        (8, 128, 3, 256, 1, 1, "tRGB"),
        (8, 3, 128, 256, 1, 1, "fromRGB"),
        (32, 512, 512, 64, 3, 1, "res64 conv B32"),
        (32, 256, 256, 128, 3, 1, "res128 conv1 B32"),
        (32, 128, 128, 256, 3, 1, "res256 conv1 B32"),
    ]
    for B, I, O, H, k, s, note in shapes:
        x = torch.randn(B, I, H, H, device=dev, dtype=dtype)
        OH = (H + 2 * (k // 2) - k) // s + 1
        dy = torch.randn(B, O, OH, OH, device=dev, dtype=dtype)
        t = timeit(lambda: C.conv2d_wgrad(x, dy, s, k // 2, k, k, False))
        fl = 2.0 * B * O * I * k * k * OH * OH
        print(f"wgrad    {note:22s} B{B} {I:4d}->{O:4d} @{H:4d}: "
              f"{t * 1e3:8.3f} ms  {fl / t / 1e12:7.1f} TF")


def bench_attn(dtype=torch.bfloat16):
    dev = "cuda:0"
    shapes = [
        (8, 65536, 16, 128, 128, "simplex res256"),
        (8, 16384, 16, 256, 256, "simplex res128"),
        (8, 16, 16384, 256, 256, "duplex-rev res128"),
        (8, 16, 65536, 128, 128, "duplex-rev res256"),
    ]
    for B, Nq, Nk, D, E, note in shapes:
        q = torch.randn(B, Nq, D, device=dev, dtype=dtype)
        k = torch.randn(B, Nk, D, device=dev, dtype=dtype)
        v = torch.randn(B, Nk, E, device=dev, dtype=dtype)
        t = timeit(lambda: C.bipartite_attn(q, k, v, D ** -0.5))
        fl = 2.0 * B * Nq * Nk * (D + E)
        print(f"attn     {note:22s}: {t * 1e3:8.3f} ms  "
              f"{fl / t / 1e12:7.1f} TF")


def bench_attn_bwd(dtype=torch.bfloat16):
    """Fused attention backward vs the eager recompute composition."""
    dev = "cuda:0"
    import math as _m
    from gansformer_amd.ops import bipartite as bp
    shapes = [
        (32, 4096, 17, 512, 512, "simplex res64 b32"),
        (32, 16384, 17, 256, 256, "simplex res128 b32"),
        (32, 17, 16384, 256, 256, "duplex-rev res128 b32"),
        (32, 17, 4096, 512, 512, "duplex-rev res64 b32"),
    ]
    for B, Nq, Nk, D, E, note in shapes:
        scale = 1.0 / _m.sqrt(D)
        q = torch.randn(B, Nq, D, device=dev, dtype=dtype)
        k = torch.randn(B, Nk, D, device=dev, dtype=dtype)
        v = torch.randn(B, Nk, E, device=dev, dtype=dtype)
        dout = torch.randn(B, Nq, E, device=dev, dtype=dtype)
        out, ml = C.bipartite_attn_fwd(q, k, v, scale)

        def fused():
            drow = (dout.float() * out.float()).sum(-1).contiguous()
            return C.bipartite_attn_bwd(q, k, v, dout, drow, ml, scale)

        def eager():
            qf, kf = q.float(), k.float()
            s = torch.einsum("bqd,bkd->bqk", qf, kf) * scale
            a = torch.softmax(s, dim=-1)
            a_lp = a.to(v.dtype)
            dv = torch.einsum("bqk,bqe->bke", a_lp, dout)
            daf = torch.einsum("bqe,bke->bqk", dout, v).float()
            ds = a * (daf - (daf * a).sum(dim=-1, keepdim=True))
            ds_lp = ds.to(q.dtype)
            dq = torch.einsum("bqk,bkd->bqd", ds_lp, k) * scale
            dk = torch.einsum("bqk,bqd->bkd", ds_lp, q) * scale
            return dq, dk, dv

        tf = timeit(fused)
        te = timeit(eager)
        fl = 5.0 * B * Nq * Nk * (D + E)  # 5 GEMMs vs fwd's 2
        print(f"attnbwd  {note:22s}: fused {tf * 1e3:8.3f} ms "
              f"({fl / tf / 1e12:6.1f} TF)  eager {te * 1e3:8.3f} ms "
              f"({te / tf:4.1f}x)")


def bench_attn_bwd_one(dtype=torch.bfloat16):
    """Single hot shape, many iters — for rocprofv3 runs (small-N bwd)."""
    import math as _m
    dev = "cuda:0"
    B, Nq, Nk, D, E = 32, 4096, 17, 512, 512
    scale = 1.0 / _m.sqrt(D)
    q = torch.randn(B, Nq, D, device=dev, dtype=dtype)
    k = torch.randn(B, Nk, D, device=dev, dtype=dtype)
    v = torch.randn(B, Nk, E, device=dev, dtype=dtype)
    dout = torch.randn(B, Nq, E, device=dev, dtype=dtype)
    out, ml = C.bipartite_attn_fwd(q, k, v, scale)
    drow = (dout.float() * out.float()).sum(-1).contiguous()
    t = timeit(lambda: C.bipartite_attn_bwd(q, k, v, dout, drow, ml, scale),
               iters=30, warmup=5)
    fl = 5.0 * B * Nq * Nk * (D + E)
    print(f"attnbwd1 smalln res64 d512: {t * 1e3:8.3f} ms "
          f"({fl / t / 1e12:6.1f} TF)")


def bench_gemm_skinny(dtype=torch.bfloat16):
    """Custom tall-skinny MFMA GEMM vs hipBLASLt (x @ w^T)."""
    dev = "cuda:0"
    shapes = [
        (32 * 4096, 512, 512, "proj res64 d512"),
        (32 * 16384, 256, 256, "proj res128 d256"),
        (32 * 65536, 128, 128, "proj res256 d128"),
        (32 * 4096, 128, 512, "gamma res64"),
        (32 * 65536, 512, 128, "gamma res256"),
    ]
    for M, N, K, note in shapes:
        x = torch.randn(M, K, device=dev, dtype=dtype)
        w = torch.randn(N, K, device=dev, dtype=dtype)
        tk = timeit(lambda: C.gemm_skinny(x, w, True))
        tl = timeit(lambda: x.matmul(w.t()))
        fl = 2.0 * M * N * K
        print(f"gemmsk   {note:22s}: ours {tk * 1e3:8.3f} ms "
              f"({fl / tk / 1e12:6.1f} TF)  blaslt {tl * 1e3:8.3f} ms "
              f"({fl / tl / 1e12:6.1f} TF)")
        # dgrad layout (B = [K,N] row-major)
        wt = w.t().contiguous()
        tk2 = timeit(lambda: C.gemm_skinny(x, wt, False))
        print(f"gemmskT  {note:22s}: ours {tk2 * 1e3:8.3f} ms "
              f"({fl / tk2 / 1e12:6.1f} TF)")


def bench_upfirdn(dtype=torch.bfloat16):
    dev = "cuda:0"
    from gansformer_amd.ops.upfirdn2d import setup_filter
    f = setup_filter([1, 3, 3, 1], device=torch.device(dev))
    shapes = [
        (8, 128, 256, 1, 1, "blur res256"),
        (8, 256, 128, 2, 1, "up2 res128->256"),
        (8, 128, 256, 1, 2, "down2 res256->128"),
    ]
    for B, Cn, H, up, down, note in shapes:
        x = torch.randn(B, Cn, H, H, device=dev, dtype=dtype)
        from gansformer_amd.ops.upfirdn2d import upfirdn2d as ufd
        t = timeit(lambda: ufd(x, f, up=up, down=down, padding=(2, 1, 2, 1)))
        nbytes = x.numel() * x.element_size() * (1 + up * up / (down * down))
        print(f"upfirdn  {note:22s}: {t * 1e3:8.3f} ms  "
              f"{nbytes / t / 1e9:7.1f} GB/s")


def bench_conv_one(dtype=torch.bfloat16):
    # single hot shape, many iters — for rocprofv3 --pmc runs
    dev = "cuda:0"
    B, I, O, H, k, s = 32, 512, 512, 64, 3, 1
    x = torch.randn(B, I, H, H, device=dev, dtype=dtype)
    w = torch.randn(O, I, k, k, device=dev, dtype=dtype) * 0.05
    t = timeit(lambda: C.conv2d_fwd(x, w, s, k // 2), iters=30, warmup=5)
    fl = 2.0 * B * O * I * k * k * H * H
    print(f"conv fwd res64 B32: {t * 1e3:8.3f} ms  {fl / t / 1e12:7.1f} TF")


def bench_wgrad_one(dtype=torch.bfloat16):
    dev = "cuda:0"
    B, I, O, H, k, s = 32, 512, 512, 64, 3, 1
    x = torch.randn(B, I, H, H, device=dev, dtype=dtype)
    dy = torch.randn(B, O, H, H, device=dev, dtype=dtype)
    t = timeit(lambda: C.conv2d_wgrad(x, dy, s, k // 2, k, k, False),
               iters=30, warmup=5)
    fl = 2.0 * B * O * I * k * k * H * H
    print(f"wgrad res64 B32: {t * 1e3:8.3f} ms  {fl / t / 1e12:7.1f} TF")


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    torch.manual_seed(0)
    if which == "conv1":
        bench_conv_one()
    if which == "wgrad1":
        bench_wgrad_one()
    if which in ("conv", "all"):
        bench_conv()
    if which in ("wgrad", "all"):
        bench_wgrad()
    if which in ("attn", "all"):
        bench_attn()
    if which in ("attnbwd", "all"):
        bench_attn_bwd()
    if which == "attnbwd1":
        bench_attn_bwd_one()
    if which in ("gemmsk", "all"):
        bench_gemm_skinny()
    if which in ("upfirdn", "all"):
        bench_upfirdn()
