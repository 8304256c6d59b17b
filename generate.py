#!/usr/bin/env python3
"""Generate images from a trained GANsformer .pkl — flag-compatible entry
point (parity with ref generate.py / run_network.py, SURVEY.md §3.2 [R]):
load the (G, D, Gs) pickle, sample z, run Gs with truncation, save a PNG
grid (and optionally per-sample PNGs and attention maps).
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch  # noqa: E402


def main(argv=None):
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--network", "--network-pkl", dest="network", required=True,
                   help="path to network-snapshot-*.pkl")
    p.add_argument("--num-images", type=int, default=16)
    p.add_argument("--truncation-psi", type=float, default=0.7)
    p.add_argument("--seed", type=int, default=42)
    p.add_argument("--batch-size", type=int, default=8)
    p.add_argument("--output-dir", "--outdir", dest="output_dir",
                   default="generated")
    p.add_argument("--grid", action="store_true", default=True)
    p.add_argument("--no-grid", dest="grid", action="store_false")
    p.add_argument("--individual", action="store_true",
                   help="also save each image separately")
    p.add_argument("--attention-maps", action="store_true",
                   help="dump per-component attention heatmaps")
    p.add_argument("--class", dest="class_idx", type=int, default=None,
                   help="class index for conditional models (one-hot)")
    p.add_argument("--noise-mode", choices=["random", "const", "none"],
                   default="const")
    args = p.parse_args(argv)

    from gansformer_amd import pkl_compat
    from gansformer_amd.training.snapshot import save_image_grid, _write_png

    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    G, D, Gs = pkl_compat.load_network_pkl(args.network)
    Gs = Gs.to(device).eval()
    os.makedirs(args.output_dir, exist_ok=True)

    label_dim = getattr(Gs, "label_dim", 0)
    if label_dim > 0 and args.class_idx is None:
        print(f"# conditional model (label_dim={label_dim}); sampling "
              "random classes — pass --class N to fix one")

    def make_label(n):
        if label_dim == 0:
            return None
        lab = torch.zeros(n, label_dim, device=device)
        if args.class_idx is not None:
            lab[:, args.class_idx % label_dim] = 1.0
        else:
            idx = torch.randint(label_dim, (n,), device=device)
            lab[torch.arange(n, device=device), idx] = 1.0
        return lab

    torch.manual_seed(args.seed)
    imgs = []
    with torch.no_grad():
        remaining = args.num_images
        while remaining > 0:
            n = min(args.batch_size, remaining)
            z = Gs.sample_z(n, device=device)
            img = Gs(z, label=make_label(n),
                     truncation_psi=args.truncation_psi,
                     noise_mode=args.noise_mode)
            imgs.append(img.cpu())
            remaining -= n
    imgs = torch.cat(imgs)

    if args.grid:
        path = os.path.join(args.output_dir, "grid.png")
        save_image_grid(imgs, path)
        print("wrote", path)
    if args.individual:
        for i in range(imgs.shape[0]):
            path = os.path.join(args.output_dir, f"img{i:04d}.png")
            save_image_grid(imgs[i:i + 1], path, grid_w=1, grid_h=1)
        print(f"wrote {imgs.shape[0]} images to {args.output_dir}")

    if args.attention_maps:
        from tools.visualize import save_attention_maps
        torch.manual_seed(args.seed)
        z = Gs.sample_z(1, device=device)
        save_attention_maps(Gs, z, args.output_dir,
                            truncation_psi=args.truncation_psi)
        print("wrote attention maps")


if __name__ == "__main__":
    main()
