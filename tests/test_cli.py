"""CLI surface: train.py / generate.py / prepare_data.py / bench.py."""

import glob
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run(args, cwd=None, timeout=600):
    return subprocess.run([sys.executable] + args, cwd=cwd or REPO,
                          capture_output=True, text=True, timeout=timeout)


@pytest.mark.timeout(600)
def test_train_cli_smoke(tmp_path):
    r = run(["train.py",
             "--result-dir", str(tmp_path),
             "--dataset", "synthetic", "--res", "32",
             "--transformer", "simplex", "--components-num", "4",
             "--latent-size", "32", "--dlatent-size", "32",
             "--fmap-base", "512", "--fmap-max", "32",
             "--mapping-layers", "1", "--bf16-res", "0",
             "--batch-gpu", "2", "--total-kimg", "0.004",
             "--snapshot-kimg", "0.004", "--synthetic-size", "16",
             "--num-workers", "0", "--mbstd-group", "2",
             "--d-reg-interval", "2", "--g-reg-interval", "2"])
    assert r.returncode == 0, r.stderr[-2000:]
    runs = glob.glob(str(tmp_path / "*-synthetic*"))
    assert runs, r.stdout
    assert os.path.exists(os.path.join(runs[0], "log.txt"))
    pkls = glob.glob(os.path.join(runs[0], "network-snapshot-*.pkl"))
    assert pkls


@pytest.mark.timeout(600)
def test_generate_cli(tmp_path):
    # train a micro model, then generate from the snapshot
    r = run(["train.py", "--result-dir", str(tmp_path / "runs"),
             "--dataset", "synthetic", "--res", "16",
             "--transformer", "simplex", "--components-num", "2",
             "--latent-size", "16", "--dlatent-size", "16",
             "--fmap-base", "256", "--fmap-max", "16",
             "--mapping-layers", "1", "--bf16-res", "0",
             "--batch-gpu", "2", "--total-kimg", "0.002",
             "--snapshot-kimg", "0.002", "--synthetic-size", "8",
             "--num-workers", "0", "--mbstd-group", "2"])
    assert r.returncode == 0, r.stderr[-2000:]
    pkl = glob.glob(str(tmp_path / "runs" / "*" / "network-snapshot-*.pkl"))[0]
    r = run(["generate.py", "--network", pkl, "--num-images", "4",
             "--output-dir", str(tmp_path / "gen"), "--attention-maps"])
    assert r.returncode == 0, r.stderr[-2000:]
    assert os.path.exists(tmp_path / "gen" / "grid.png")
    assert glob.glob(str(tmp_path / "gen" / "attn-layer*.png"))


@pytest.mark.timeout(300)
def test_prepare_data_and_dataset(tmp_path):
    r = run(["prepare_data.py", "--name", "toy", "--synthetic", "12",
             "--data-dir", str(tmp_path), "--res", "16", "--shard-size", "5"])
    assert r.returncode == 0, r.stderr[-2000:]
    from gansformer_amd.training.dataset import ShardedNpyDataset
    ds = ShardedNpyDataset(str(tmp_path / "toy"))
    assert len(ds) == 12
    img, label = ds[7]
    assert img.shape == (3, 16, 16) and img.dtype.is_floating_point is False


@pytest.mark.timeout(900)
def test_bench_cli_cpu(tmp_path):
    r = run(["bench.py", "--gpus", "1", "--steps", "2", "--warmup", "1",
             "--res", "32", "--batch-gpu", "2", "--device", "cpu"])
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    out = json.loads(line)
    assert out["unit"] == "imgs/sec"
    assert out["value"] > 0
    assert out["n_gpus"] == 1
    assert out["steps"] == 2
    assert out["scaling"] == "weak"
    assert out["data"] == "synthetic"


def test_mirror_augment_dataset():
    from gansformer_amd.training.dataset import (MirroredDataset,
                                                 SyntheticDataset)
    import torch
    base = SyntheticDataset(resolution=16, size=8, seed=1)
    ds = MirroredDataset(base)
    assert len(ds) == 16 and ds.resolution == 16
    img, _ = ds[3]
    img_f, _ = ds[3 + 8]
    assert torch.equal(img_f, torch.flip(img, dims=[-1]))
    assert not torch.equal(img_f, img)


@pytest.mark.timeout(300)
def test_conditional_train_and_generate_cli(tmp_path):
    """--label-dim trains a conditional model end to end; generate.py
    detects label_dim from the pkl and honors --class."""
    r = run(["train.py", "--result-dir", str(tmp_path / "runs"),
             "--dataset", "synthetic", "--res", "16", "--label-dim", "3",
             "--transformer", "simplex", "--components-num", "2",
             "--latent-size", "16", "--dlatent-size", "16",
             "--fmap-base", "256", "--fmap-max", "16",
             "--mapping-layers", "1", "--bf16-res", "0",
             "--batch-gpu", "2", "--total-kimg", "0.002",
             "--snapshot-kimg", "0.002", "--synthetic-size", "8",
             "--num-workers", "0", "--mbstd-group", "2"])
    assert r.returncode == 0, r.stderr[-2000:]
    pkl = glob.glob(str(tmp_path / "runs" / "*" / "network-snapshot-*.pkl"))[0]
    r = run(["generate.py", "--network", pkl, "--num-images", "2",
             "--class", "1", "--output-dir", str(tmp_path / "gen")])
    assert r.returncode == 0, r.stderr[-2000:]
    assert os.path.exists(tmp_path / "gen" / "grid.png")


@pytest.mark.timeout(600)
def test_bench_torchrun_world2(tmp_path):
    """The driver's SCALE entry path: torchrun --nproc-per-node 2
    bench.py ... must emit exactly one JSON line from rank 0 with the
    whole-job aggregate (gloo on CPU here; RCCL on GPU boxes)."""
    import json as _json
    import subprocess, sys
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29882", os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1", "--res", "32",
         "--batch-gpu", "2", "--components-num", "2"],
        capture_output=True, text=True, timeout=560, cwd=str(tmp_path))
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout
    row = _json.loads(lines[0])
    assert row["n_gpus"] == 2
    assert row["config"]["global_batch"] == 4
    assert row["config"]["parallelism"] == "dp2"
    assert row["value"] > 0 and row["ms_per_step"] > 0
