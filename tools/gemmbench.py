import sys, time, torch
def t(fn, it=30, w=5):
    for _ in range(w): fn()
    torch.cuda.synchronize(); t0=time.time()
    for _ in range(it): fn()
    torch.cuda.synchronize(); return (time.time()-t0)/it
dev='cuda:0'
shapes = [
    # attention projections (B=32): tokens [B,HW,C] @ [C,d]
    (32*16384, 256, 128, "attn res128 qkv"),
    (32*4096, 512, 512, "attn res64 qkv"),
    (32*65536, 128, 128, "attn res256-like"),
    (32*16384, 128, 256, "gamma/beta bwd-ish"),
    (544, 512, 512, "mapping"),
]
for M,K,N,note in shapes:
    a = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    b = torch.randn(K, N, device=dev, dtype=torch.bfloat16)
    dt = t(lambda: a @ b)
    fl = 2.0*M*K*N
    print(f"{note:22s} M{M:>8} K{K:>4} N{N:>4}: {dt*1e3:7.3f} ms {fl/dt/1e12:7.1f} TF")
# batched strided (the _fc_transposed pattern)
u = torch.randn(32, 16384, 256, device=dev, dtype=torch.bfloat16)
w = torch.randn(128, 256, device=dev, dtype=torch.bfloat16)
dt = t(lambda: torch.matmul(w, u.transpose(1,2)))
print(f"fc_transposed bmm     : {dt*1e3:7.3f} ms {2.0*32*16384*256*128/dt/1e12:7.1f} TF")
