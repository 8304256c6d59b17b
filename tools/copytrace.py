#!/usr/bin/env python3
"""Attribute copy_ volume by tensor shape via TorchDispatchMode."""
import sys, os, collections
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torch.utils._python_dispatch import TorchDispatchMode

agg = collections.Counter()
cnt = collections.Counter()

class Trace(TorchDispatchMode):
    def __torch_dispatch__(self, func, types, args=(), kwargs=None):
        name = str(func)
        if "copy_" in name or "clone" in name or "_to_copy" in name:
            t = args[0]
            key = (name.split('.')[0], tuple(t.shape), str(t.dtype).replace("torch.",""))
            agg[key] += t.numel() * t.element_size()
            cnt[key] += 1
        return func(*args, **(kwargs or {}))

def main():
    import copy
    from gansformer_amd.models.networks import Discriminator, Generator
    from gansformer_amd.training.trainer import GANTrainer
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    G = Generator(img_resolution=256, num_components=16,
                  transformer="duplex").to(dev).train()
    D = Discriminator(img_resolution=256).to(dev).train()
    Gs = copy.deepcopy(G).eval()
    for p in Gs.parameters():
        p.requires_grad_(False)
    tr = GANTrainer(G, D, Gs, dev, 16, 16)
    gen = torch.Generator(device=dev).manual_seed(1)
    def real():
        return (torch.randint(0, 256, (16, 3, 256, 256), device=dev,
                              dtype=torch.uint8, generator=gen)
                .float().div(127.5).sub(1.0))
    for i in range(2):
        tr.step(real, i, i * 16)
    torch.cuda.synchronize()
    with Trace():
        tr.step(real, 2, 64)
        torch.cuda.synchronize()
    rows = sorted(agg.items(), key=lambda kv: -kv[1])[:22]
    for (op, shape, dt), byt in rows:
        print(f"{byt/1e6:9.1f} MB x{cnt[(op,shape,dt)]:>4} {op:10s} {dt:9s} {shape}")

if __name__ == "__main__":
    main()
