#!/usr/bin/env python3
"""Plot training curves from a run dir's metrics.jsonl / metric-*.txt.

Usage: python tools/plot_metrics.py RUN_DIR [-o out.png] [--title T]
Produces a two-panel figure: D/G losses (+R1/pl) over kimg, and the
FID curve (with the extractor tag in the legend).
"""
import argparse
import glob
import json
import os
import sys

import matplotlib
matplotlib.use("Agg")
import matplotlib.pyplot as plt  # noqa: E402


def load_jsonl(path):
    rows = []
    with open(path) as f:
        for line in f:
            try:
                rows.append(json.loads(line))
            except json.JSONDecodeError:
                pass
    return rows


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("run_dir")
    p.add_argument("-o", "--out", default=None)
    p.add_argument("--title", default=None)
    args = p.parse_args(argv)

    rows = load_jsonl(os.path.join(args.run_dir, "metrics.jsonl"))
    rows = [r for r in rows if "Loss/D" in r]
    if not rows:
        print("no loss rows found", file=sys.stderr)
        return 1
    kimg = [r.get("kimg", r["step"] / 1000) for r in rows]

    fids = {}
    for mf in sorted(glob.glob(os.path.join(args.run_dir, "metric-*.txt"))):
        name = os.path.basename(mf)[7:-4]
        pts, tag = [], ""
        for line in open(mf):
            parts = line.split()
            if len(parts) >= 2 and parts[0].endswith("kimg"):
                pts.append((float(parts[0][:-4]), float(parts[1])))
                for tok in parts[2:]:
                    if tok.startswith("extractor="):
                        tag = tok.split("=", 1)[1]
        if pts:
            fids[name] = (pts, tag)

    ncols = 2 if fids else 1
    fig, axes = plt.subplots(1, ncols, figsize=(6 * ncols, 4.2))
    if ncols == 1:
        axes = [axes]
    ax = axes[0]
    for key, style in (("Loss/D", "-"), ("Loss/G", "-"),
                       ("Loss/r1", "--"), ("Loss/pl", "--")):
        ys = [(k, r[key]) for k, r in zip(kimg, rows) if key in r]
        if ys:
            ax.plot([a for a, _ in ys], [b for _, b in ys], style,
                    label=key, linewidth=1.1)
    ax.set_xlabel("kimg")
    ax.set_ylabel("loss")
    ax.legend(fontsize=8)
    ax.grid(alpha=0.3)
    ax.set_title(args.title or os.path.basename(args.run_dir.rstrip("/")))

    if fids:
        ax2 = axes[1]
        for name, (pts, tag) in fids.items():
            ax2.plot([a for a, _ in pts], [b for _, b in pts], "o-",
                     label=f"{name} ({tag})" if tag else name)
        ax2.set_xlabel("kimg")
        ax2.set_ylabel("FID")
        ax2.set_yscale("log")
        ax2.legend(fontsize=8)
        ax2.grid(alpha=0.3)
        ax2.set_title("FID (same-extractor A/B)")

    out = args.out or os.path.join(args.run_dir, "curves.png")
    fig.tight_layout()
    fig.savefig(out, dpi=110)
    print("wrote", out)
    return 0


if __name__ == "__main__":
    sys.exit(main())
