"""Training orchestration (ref src/training/training_loop.py [R]).

One process per GPU (RCCL data parallel); builds G / D / Gs, runs the
kimg-scheduled loop alternating D and G phases with lazy R1 (every
d_reg_interval steps) and path-length regularization (every
g_reg_interval steps) via GANTrainer, maintains the EMA copy Gs, writes
snapshots (.pkl, pkl_compat layout) and image grids, and runs metrics on
cadence.
"""

from __future__ import annotations

import copy
import json
import os
import signal
import time

import numpy as np
import torch

from ..config import EasyDict
from ..metrics.fid import METRIC_SPECS, compute_fid, load_extractor
from ..metrics.scalars import ScalarLogger
from ..models.networks import Discriminator, Generator
from ..parallel import broadcast_params
from ..parallel.dist import any_rank, barrier, cleanup, setup_distributed
from ..rundir import RunContext
from . import snapshot as snap
from .dataset import load_dataset, make_loader, normalize_images
from .trainer import GANTrainer


def training_loop(
    run_dir=None,
    dataset_args: dict | None = None,
    G_args: dict | None = None,
    D_args: dict | None = None,
    loss_args: dict | None = None,
    sched_args: dict | None = None,
    total_kimg=25000,
    batch_gpu=8,
    batch_size=None,            # global; default world_size * batch_gpu
    ema_kimg=10.0,
    ema_rampup=0.05,
    snapshot_kimg=200,
    image_snapshot_kimg=50,
    metrics=(),
    metric_kimg=1000,
    metric_images=None,         # override per-metric image count (tests)
    resume_pkl=None,
    resume_kimg=0,
    seed=0,
    num_workers=2,
    profile_steps=0,
    log_interval_kimg=1,
    device=None,
    progress_fn=None,
):
    dataset_args = EasyDict(dataset_args or {})
    G_args = EasyDict(G_args or {})
    D_args = EasyDict(D_args or {})
    loss_args = EasyDict(loss_args or {})
    sched_args = EasyDict(sched_args or {})

    rank, world_size, dev = setup_distributed()
    if device is not None:
        dev = torch.device(device)
    is_main = rank == 0
    torch.manual_seed(seed * 100 + rank)
    np.random.seed(seed * 100 + rank)

    if batch_size is None:
        batch_size = batch_gpu * world_size
    assert batch_size % (batch_gpu * world_size) == 0
    rounds = batch_size // (batch_gpu * world_size)

    # dataset
    ds = load_dataset(**{k: v for k, v in dataset_args.items()
                         if k != "inception_path"})
    loader = make_loader(ds, batch_gpu, rank=rank, world_size=world_size,
                         num_workers=num_workers, seed=seed)
    res = ds.resolution
    chans = ds.image_shape[0]

    # networks (conditional iff the dataset carries labels)
    label_dim = getattr(ds, "label_dim", 0)
    G_kwargs = dict(img_resolution=res, img_channels=chans,
                    label_dim=label_dim)
    G_kwargs.update(G_args)
    D_kwargs = dict(img_resolution=res, img_channels=chans,
                    label_dim=label_dim)
    D_kwargs.update(D_args)
    G = Generator(**G_kwargs).to(dev).train()
    D = Discriminator(**D_kwargs).to(dev).train()
    Gs = copy.deepcopy(G).eval()
    for p in Gs.parameters():
        p.requires_grad_(False)

    start_nimg = int(resume_kimg * 1000)
    extra = None
    if resume_pkl:
        extra = snap.load_resume(resume_pkl, G, D, Gs)
        if is_main:
            print(f"Resumed from {resume_pkl}")

    broadcast_params(G)
    broadcast_params(D)
    broadcast_params(Gs)

    conditional = G_kwargs.get("label_dim", 0) > 0

    def next_real():
        imgs_u8, labels = next(loader)
        imgs = normalize_images(imgs_u8, dev)
        return (imgs, labels.to(dev)) if conditional else imgs

    def sample_labels(n):
        idx = np.random.randint(0, len(ds), size=n)
        return torch.stack([ds[int(i)][1] for i in idx]).to(dev)

    trainer = GANTrainer(
        G, D, Gs, dev, batch_gpu, batch_size,
        g_lr=sched_args.get("g_lr", 0.002),
        d_lr=sched_args.get("d_lr", 0.002),
        beta1=sched_args.get("beta1", 0.0),
        beta2=sched_args.get("beta2", 0.99),
        eps=sched_args.get("eps", 1e-8),
        gamma=loss_args.get("gamma", 10.0),
        d_reg_interval=loss_args.get("d_reg_interval", 16),
        g_reg_interval=loss_args.get("g_reg_interval", 8),
        pl_weight=loss_args.get("pl_weight", 2.0),
        pl_decay=loss_args.get("pl_decay", 0.01),
        ema_kimg=ema_kimg, ema_rampup=ema_rampup, rounds=rounds,
        label_fn=sample_labels if conditional else None)
    if extra is not None:
        trainer.load_state_dict(extra)
        if "cur_nimg" in extra and not resume_kimg:
            start_nimg = int(extra["cur_nimg"])

    logger = ScalarLogger(run_dir if is_main else None)
    ctx = RunContext(run_dir if is_main else None, total_kimg=total_kimg)
    stop_flag = {"stop": False}

    def _sigterm(signum, frame):
        stop_flag["stop"] = True
    try:
        signal.signal(signal.SIGTERM, _sigterm)
    except ValueError:
        pass  # not in main thread

    # reals preview grid
    if is_main and run_dir is not None:
        imgs = torch.stack([ds[i][0] for i in range(min(16, len(ds)))])
        snap.save_image_grid(imgs.float() / 127.5 - 1,
                             os.path.join(run_dir, "reals.png"))

    metric_extractor = None
    real_stats_cache = {}

    def grab_real_batch(n):
        idx = np.random.randint(0, len(ds), size=n)
        xs = torch.stack([ds[int(i)][0] for i in idx])
        return normalize_images(xs, dev)

    def run_metrics(cur_nimg):
        nonlocal metric_extractor
        results = {}
        for name in metrics:
            if name not in METRIC_SPECS:
                if is_main:
                    print(f"unknown metric {name!r}; skipping")
                continue
            n_img = metric_images or METRIC_SPECS[name]
            if metric_extractor is None:
                metric_extractor = load_extractor(
                    dataset_args.get("inception_path"), dev)

            def gen_batch(n):
                z = Gs.sample_z(n, device=dev)
                lab = sample_labels(n) if conditional else None
                return Gs(z, label=lab, truncation_psi=1.0,
                          noise_mode="random")

            fid, stats = compute_fid(
                gen_batch, grab_real_batch, n_img, min(batch_gpu * 2, 64),
                dev, extractor=metric_extractor,
                real_stats_cache=real_stats_cache.get(name),
                rank=rank, world_size=world_size)
            real_stats_cache[name] = stats
            results[name] = fid
            if is_main and run_dir is not None:
                from ..metrics.fid import extractor_tag
                tag = extractor_tag(dataset_args.get("inception_path"))
                with open(os.path.join(run_dir, f"metric-{name}.txt"), "a") as f:
                    f.write(f"{cur_nimg // 1000:d}kimg {fid:.4f} "
                            f"extractor={tag}\n")
        return results

    cur_nimg = start_nimg
    step = start_nimg // batch_size
    # SIGTERM/abort reaction latency: every ~quarter log interval, at
    # least every 16 steps' worth of images
    stopchk_nimg = max(batch_size, min(log_interval_kimg * 250,
                                       batch_size * 16))
    next_stopchk = cur_nimg + stopchk_nimg
    next_log = cur_nimg + log_interval_kimg * 1000
    next_snap = cur_nimg + snapshot_kimg * 1000
    next_img = 0  # write fakes at start too
    next_metric = cur_nimg + metric_kimg * 1000
    t_start = time.time()
    t_tick = t_start
    nimg_tick = cur_nimg
    prof = None
    if profile_steps and is_main and run_dir is not None:
        prof = torch.profiler.profile(
            schedule=torch.profiler.schedule(wait=8, warmup=2,
                                             active=profile_steps, repeat=1),
            on_trace_ready=torch.profiler.tensorboard_trace_handler(run_dir),
            with_stack=False)
        prof.start()

    grid_z = G.sample_z(16, device=dev)
    grid_label = sample_labels(16) if conditional else None

    while cur_nimg < total_kimg * 1000:
        trainer.step(next_real, step, cur_nimg, start_nimg=start_nimg,
                     logger=logger)
        cur_nimg += batch_size
        step += 1
        if prof is not None:
            prof.step()

        # ---------------- bookkeeping ----------------
        if cur_nimg >= next_log or cur_nimg >= total_kimg * 1000:
            now = time.time()
            imgs_sec = (cur_nimg - nimg_tick) / max(now - t_tick, 1e-9)
            t_tick, nimg_tick = now, cur_nimg
            row = None
            if is_main:
                row = logger.flush(step, kimg=cur_nimg / 1000,
                                   imgs_sec=imgs_sec,
                                   sec_total=now - t_start)
                msg = " ".join(f"{k}={v:.4g}" for k, v in row.items()
                               if isinstance(v, float))
                print(f"tick step={step} {msg}", flush=True)
            ctx.update(cur_kimg=cur_nimg / 1000)
            next_log = cur_nimg + log_interval_kimg * 1000
            if progress_fn is not None:
                progress_fn(cur_nimg, row)

        if is_main and run_dir is not None and cur_nimg >= next_img:
            with torch.no_grad():
                fakes = Gs(grid_z, label=grid_label, truncation_psi=0.7,
                           noise_mode="const")
            snap.save_image_grid(
                fakes, os.path.join(run_dir, f"fakes{cur_nimg // 1000:06d}.png"))
            next_img = cur_nimg + image_snapshot_kimg * 1000

        # Stop decisions must be COLLECTIVE: the abort file exists only
        # on rank 0 and SIGTERM may reach a subset of ranks; a rank
        # leaving alone would hang the others on the next collective.
        # The collective runs on the lockstep stop-check cadence (not
        # every step: the 4-byte all-reduce + .item() is a cross-rank
        # hard sync that would cap backward/comm overlap), so stop
        # latency is bounded by stopchk_interval.
        want_stop = False
        if cur_nimg >= next_stopchk or cur_nimg >= total_kimg * 1000:
            want_stop = any_rank(stop_flag["stop"] or ctx.should_stop())
            next_stopchk = cur_nimg + stopchk_nimg
        if cur_nimg >= next_snap or cur_nimg >= total_kimg * 1000 or want_stop:
            trainer.sync_ranks()
            if is_main and run_dir is not None:
                extra_state = dict(trainer.state_dict(),
                                   cur_nimg=cur_nimg, step=step)
                path = snap.save_snapshot(run_dir, cur_nimg, G, D, Gs,
                                          extra_state)
                print(f"snapshot: {path}", flush=True)
            next_snap = cur_nimg + snapshot_kimg * 1000

        if metrics and (cur_nimg >= next_metric
                        or cur_nimg >= total_kimg * 1000):
            trainer.sync_ranks()  # all ranks run Gs for FID: keep it identical
            results = run_metrics(cur_nimg)
            if is_main and results:
                print("metrics: " + json.dumps(results), flush=True)
            next_metric = cur_nimg + metric_kimg * 1000

        if want_stop:
            if is_main:
                print("stopping (SIGTERM/abort)", flush=True)
            break

    if prof is not None:
        prof.stop()
    barrier()
    logger.close()
    cleanup()
    return dict(cur_nimg=cur_nimg, steps=step)
