import time, torch
def t(fn, it=30, w=5):
    for _ in range(w): fn()
    torch.cuda.synchronize(); t0=time.time()
    for _ in range(it): fn()
    torch.cuda.synchronize(); return (time.time()-t0)/it
dev='cuda:0'
for M,K,N,note in [(32*16384, 128, 256, "fc dw res128"), (32*4096, 512, 512, "fc dw res64"), (32*65536, 128, 128, "fc dw res256")]:
    dy = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    x = torch.randn(M, N, device=dev, dtype=torch.bfloat16)
    fl = 2.0*M*K*N
    dt = t(lambda: dy.t() @ x)
    print(f"{note:14s} direct  dy.T@x [{K}x{M}]x[{M}x{N}]: {dt*1e3:7.3f} ms {fl/dt/1e12:6.1f} TF")
    S = 32
    dys = dy.reshape(S, M//S, K); xs = x.reshape(S, M//S, N)
    dt = t(lambda: torch.bmm(dys.transpose(1,2), xs).sum(0))
    print(f"{note:14s} splitK  bmm+sum S={S}          : {dt*1e3:7.3f} ms {fl/dt/1e12:6.1f} TF")
