#!/usr/bin/env python3
"""GPU shape fuzz: random eligible shapes through every kernel branch vs
the CPU eager reference. Run on a GPU box; exits nonzero on mismatch."""
import sys, os, random
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F

def rel(a, b):
    a, b = a.float().cpu(), b.float().cpu()
    return ((a - b).abs().max() / (b.abs().max() + 1e-6)).item()

def main():
    from gansformer_amd.ops.conv2d_grad import (conv2d_gradfix, conv2d_up2,
                                                _zero_stuff2, _eager_conv2d,
                                                _eager_wgrad, _Conv2dWgrad)
    dev = torch.device("cuda:0")
    rng = random.Random(0)
    bad = 0
    for trial in range(40):
        B = rng.choice([1, 2, 3, 5])
        I = rng.choice([32, 64, 96, 160, 256])
        O = rng.choice([32, 48, 128, 160, 224])
        H = rng.choice([8, 16, 24, 32, 48, 64])
        W = rng.choice([16, 32, 48, 64])
        mode = rng.choice(["s1", "s2", "up2", "wg1", "wg2"])
        torch.manual_seed(trial)
        x = torch.randn(B, I, H, W, device=dev, dtype=torch.bfloat16)
        w = torch.randn(O, I, 3, 3, device=dev, dtype=torch.bfloat16) * 0.1
        try:
            if mode == "s1":
                y = conv2d_gradfix(x, w, stride=1, padding=1)
                r = _eager_conv2d(x.float().cpu(), w.float().cpu(), 1, 1)
            elif mode == "s2":
                if H % 2 or W % 2:
                    continue
                y = conv2d_gradfix(x, w, stride=2, padding=1)
                r = _eager_conv2d(x.float().cpu(), w.float().cpu(), 2, 1)
            elif mode == "up2":
                y = conv2d_up2(x, w)
                r = F.conv2d(_zero_stuff2(x.float().cpu()), w.float().cpu(),
                             padding=1)
            elif mode == "wg1":
                dy = torch.randn(B, O, H, W, device=dev, dtype=torch.bfloat16)
                y = _Conv2dWgrad.apply(x, dy, 1, 1, 3, 3, False)
                r = _eager_wgrad(x.float().cpu(), dy.float().cpu(), 1, 1, 3,
                                 3, False)
            else:
                if H % 2 or W % 2:
                    continue
                dy = torch.randn(B, O, H // 2, W // 2, device=dev,
                                 dtype=torch.bfloat16)
                y = _Conv2dWgrad.apply(x, dy, 2, 1, 3, 3, False)
                r = _eager_wgrad(x.float().cpu(), dy.float().cpu(), 2, 1, 3,
                                 3, False)
            e = rel(y, r)
            status = "OK " if e < 5e-2 else "BAD"
            if e >= 5e-2:
                bad += 1
            print(f"{status} {mode} B{B} I{I} O{O} {H}x{W} rel={e:.2e}")
        except Exception as ex:
            bad += 1
            print(f"ERR {mode} B{B} I{I} O{O} {H}x{W}: {ex}")
    print("bad:", bad)
    sys.exit(1 if bad else 0)

if __name__ == "__main__":
    main()
