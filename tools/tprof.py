#!/usr/bin/env python3
"""torch.profiler attribution of one training step (GPU box).
Prints top ops by CUDA time so elementwise costs map back to Python."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torch.profiler import profile, ProfilerActivity

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
    tr = GANTrainer(G, D, Gs, dev, 32, 32)
    gen = torch.Generator(device=dev).manual_seed(1)
    def real():
        return (torch.randint(0, 256, (32, 3, 256, 256), device=dev,
                              dtype=torch.uint8, generator=gen)
                .float().div(127.5).sub(1.0))
    for i in range(3):
        tr.step(real, i, i * 32)
    torch.cuda.synchronize()
    with profile(activities=[ProfilerActivity.CUDA, ProfilerActivity.CPU],
                 with_stack=True) as pr:
        for i in range(3, 5):
            tr.step(real, i, i * 32)
        torch.cuda.synchronize()
    print(pr.key_averages().table(sort_by="cuda_time_total", row_limit=25,
                                  max_name_column_width=55))
    ka = pr.key_averages(group_by_stack_n=6)
    rows = [e for e in ka if e.key in ("aten::copy_", "aten::clone",
                                       "aten::contiguous")]
    rows.sort(key=lambda e: -e.self_device_time_total)
    for e in rows[:10]:
        print(f"== {e.key} {e.self_device_time_total / 1e3:.1f} ms "
              f"x{e.count}")
        for fr in (e.stack or [])[:6]:
            print("   ", fr)

if __name__ == "__main__":
    main()
