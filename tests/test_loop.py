"""End-to-end training loop on CPU — driver config 1: CLEVR 64x64
simplex-attention, k=8 latents, batch=2 (BASELINE.json:7). Exercises the
full loop: D/G phases, lazy R1 + path-length, EMA, snapshot, resume,
metrics cadence."""

import glob
import json
import os

import pytest
import torch

from gansformer_amd.training.loop import training_loop


def cfg(tmp_path, **over):
    base = dict(
        run_dir=str(tmp_path),
        dataset_args=dict(dataset="clevr-synth", resolution=64,
                          synthetic_size=64),
        G_args=dict(z_dim=32, w_dim=32, num_components=8,
                    transformer="simplex", channel_base=1024,
                    channel_max=64, bf16_res_count=0, mapping_layers=2),
        D_args=dict(channel_base=1024, channel_max=64, mbstd_group_size=2,
                    bf16_res_count=0),
        loss_args=dict(gamma=10.0, d_reg_interval=4, g_reg_interval=2),
        total_kimg=0.008,   # 8 imgs => 4 steps at batch 2
        batch_gpu=2,
        snapshot_kimg=0.008,
        image_snapshot_kimg=0.008,
        log_interval_kimg=0.002,
        ema_kimg=0.01,
        num_workers=0,
        seed=0,
    )
    base.update(over)
    return base


def test_training_loop_cpu_config1(tmp_path):
    out = training_loop(**cfg(tmp_path))
    assert out["cur_nimg"] >= 8
    # artifacts
    assert os.path.exists(tmp_path / "reals.png")
    assert glob.glob(str(tmp_path / "fakes*.png"))
    pkls = glob.glob(str(tmp_path / "network-snapshot-*.pkl"))
    assert pkls
    assert os.path.exists(tmp_path / "metrics.jsonl")
    with open(tmp_path / "metrics.jsonl") as f:
        rows = [json.loads(l) for l in f if l.strip()]
    assert rows and "imgs_sec" in rows[-1]
    assert all(torch.isfinite(torch.tensor(r.get("Loss/D", 0.0)))
               for r in rows)


def test_training_loop_resume(tmp_path):
    d1 = tmp_path / "a"
    d1.mkdir()
    training_loop(**cfg(d1))
    pkl = sorted(glob.glob(str(d1 / "network-snapshot-*.pkl")))[-1]
    d2 = tmp_path / "b"
    d2.mkdir()
    out = training_loop(**cfg(d2, resume_pkl=pkl, total_kimg=0.012))
    assert out["cur_nimg"] >= 12


def test_training_loop_metrics(tmp_path):
    out = training_loop(**cfg(
        tmp_path, metrics=["fid1k"], metric_images=16, metric_kimg=0.008))
    assert os.path.exists(tmp_path / "metric-fid1k.txt")
    with open(tmp_path / "metric-fid1k.txt") as f:
        line = f.read().strip()
    assert "kimg" in line and float(line.split()[1]) >= 0
    assert "extractor=" in line  # same-extractor A/B protocol tag


def test_training_loop_grad_accumulation(tmp_path):
    """rounds > 1: global batch = 2x the per-device batch, accumulated
    over two rounds per phase."""
    out = training_loop(**cfg(tmp_path, batch_gpu=2, batch_size=4,
                              total_kimg=0.016))
    assert out["cur_nimg"] >= 16
    with open(tmp_path / "metrics.jsonl") as f:
        rows = [json.loads(l) for l in f if l.strip()]
    assert rows and all(
        torch.isfinite(torch.tensor(r.get("Loss/D", 0.0))) for r in rows)


def test_training_loop_conditional(tmp_path):
    """Dataset with labels -> conditional G/D end to end (class embed in
    mapping, projection head in D; SURVEY M1 '+class embed')."""
    out = training_loop(**cfg(
        tmp_path,
        dataset_args=dict(dataset="synthetic", resolution=32,
                          synthetic_size=64, label_dim=5)))
    assert out["steps"] >= 1
