#!/usr/bin/env python3
"""Train a GANsformer on MI355X — flag-compatible entry point.

Parity with the reference's train.py (SURVEY.md §2.1 #1, §3.1 [R]): the
argparse surface (--dataset, --res, --transformer, --components-num,
--latent-size, --gamma, --batch-size, --total-kimg, --resume-pkl,
--metrics, --num-gpus, ...) maps onto nested EasyDict configs driving
gansformer_amd.training.loop.training_loop.

Multi-GPU: either launch under `torch.distributed.run` yourself, or pass
--num-gpus N and this script spawns one rank per GPU over RCCL.
"""

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from gansformer_amd.config import EasyDict, parse_comma_list  # noqa: E402


def build_configs(args):
    dataset_args = EasyDict(
        dataset=args.dataset, data_dir=args.data_dir,
        resolution=args.res, synthetic_size=args.synthetic_size,
        seed=args.seed, mirror_augment=args.mirror_augment,
        label_dim=args.label_dim)
    if args.inception_path:
        dataset_args.inception_path = args.inception_path
    G_args = EasyDict(
        z_dim=args.latent_size, w_dim=args.dlatent_size,
        num_components=args.components_num, transformer=args.transformer,
        integration=args.integration, use_pos=not args.no_pos,
        channel_base=args.fmap_base, channel_max=args.fmap_max,
        bf16_res_count=args.bf16_res, mapping_layers=args.mapping_layers,
        style_mixing_prob=args.style_mixing)
    if args.attn_resolutions:
        G_args.attn_resolutions = [int(r) for r in parse_comma_list(args.attn_resolutions)]
    D_args = EasyDict(
        channel_base=args.fmap_base, channel_max=args.fmap_max,
        mbstd_group_size=args.mbstd_group, bf16_res_count=args.bf16_res)
    loss_args = EasyDict(
        gamma=args.gamma, d_reg_interval=args.d_reg_interval,
        g_reg_interval=args.g_reg_interval, pl_weight=args.pl_weight)
    sched_args = EasyDict(
        g_lr=args.g_lr, d_lr=args.d_lr, beta1=args.beta1, beta2=args.beta2)
    loop_args = EasyDict(
        dataset_args=dataset_args, G_args=G_args, D_args=D_args,
        loss_args=loss_args, sched_args=sched_args,
        total_kimg=args.total_kimg, batch_gpu=args.batch_gpu,
        batch_size=args.batch_size, ema_kimg=args.ema_kimg,
        snapshot_kimg=args.snapshot_kimg,
        image_snapshot_kimg=args.image_snapshot_kimg,
        metrics=parse_comma_list(args.metrics),
        metric_kimg=args.metric_kimg,
        resume_pkl=args.resume_pkl, resume_kimg=args.resume_kimg,
        seed=args.seed, num_workers=args.num_workers,
        profile_steps=args.profile)
    return loop_args


def main(argv=None):
    p = argparse.ArgumentParser(description=__doc__,
                                formatter_class=argparse.ArgumentDefaultsHelpFormatter)
    # run
    p.add_argument("--result-dir", default="results")
    p.add_argument("--desc", default=None, help="run dir description suffix")
    p.add_argument("--num-gpus", type=int, default=1)
    p.add_argument("--seed", type=int, default=0)
    # dataset
    p.add_argument("--dataset", default="synthetic")
    p.add_argument("--data-dir", default="datasets")
    p.add_argument("--res", "--resolution", dest="res", type=int, default=256)
    p.add_argument("--synthetic-size", type=int, default=50000)
    p.add_argument("--label-dim", type=int, default=0,
                   help="one-hot label dim for synthetic data (prepared "
                        "datasets carry their own labels; >0 trains "
                        "class-conditional G/D)")
    p.add_argument("--mirror-augment", action="store_true")
    # model
    p.add_argument("--transformer", choices=["none", "simplex", "duplex"],
                   default="duplex", help="bipartite attention variant")
    p.add_argument("--components-num", "--k", dest="components_num", type=int,
                   default=16, help="number of latent components k")
    p.add_argument("--latent-size", type=int, default=512)
    p.add_argument("--dlatent-size", type=int, default=512)
    p.add_argument("--integration", choices=["add", "mul", "both"], default="mul")
    p.add_argument("--no-pos", action="store_true",
                   help="disable positional encodings in attention")
    p.add_argument("--attn-resolutions", default=None,
                   help="comma list of resolutions with attention (default 8..128)")
    p.add_argument("--fmap-base", type=int, default=32768)
    p.add_argument("--fmap-max", type=int, default=512)
    p.add_argument("--mapping-layers", type=int, default=8)
    p.add_argument("--mbstd-group", type=int, default=4)
    p.add_argument("--bf16-res", type=int, default=16,
                   help="number of top resolutions computed in bf16")
    # loss / sched
    p.add_argument("--gamma", type=float, default=10.0, help="R1 weight")
    p.add_argument("--d-reg-interval", type=int, default=16)
    p.add_argument("--g-reg-interval", type=int, default=8)
    p.add_argument("--pl-weight", type=float, default=2.0)
    p.add_argument("--style-mixing", type=float, default=0.9)
    p.add_argument("--g-lr", type=float, default=0.002)
    p.add_argument("--d-lr", type=float, default=0.002)
    p.add_argument("--beta1", type=float, default=0.0)
    p.add_argument("--beta2", type=float, default=0.99)
    # loop
    p.add_argument("--total-kimg", type=float, default=25000)
    p.add_argument("--batch-size", type=int, default=None, help="global batch")
    p.add_argument("--batch-gpu", type=int, default=8)
    p.add_argument("--ema-kimg", type=float, default=10.0)
    p.add_argument("--snapshot-kimg", type=float, default=200)
    p.add_argument("--image-snapshot-kimg", type=float, default=50)
    p.add_argument("--metrics", default="",
                   help="comma list, e.g. fid50k or fid1k")
    p.add_argument("--metric-kimg", type=float, default=1000)
    p.add_argument("--inception-path", default=None)
    p.add_argument("--resume-pkl", default=None)
    p.add_argument("--resume-kimg", type=float, default=0)
    p.add_argument("--num-workers", type=int, default=2)
    p.add_argument("--profile", type=int, default=0,
                   help="profile N steps with torch.profiler")
    args = p.parse_args(argv)

    loop_args = build_configs(args)

    rank_env = int(os.environ.get("RANK", "0"))
    world_env = int(os.environ.get("WORLD_SIZE", "1"))
    if args.num_gpus > 1 and world_env == 1:
        # self-spawn one rank per GPU
        import torch.multiprocessing as mp
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29571")
        desc = args.desc or _default_desc(args)
        from gansformer_amd import rundir
        run_dir = rundir.create_run_dir(args.result_dir, desc)
        rundir.save_submit_config(run_dir, dict(vars(args)))
        mp.spawn(_spawn_main, args=(args.num_gpus, loop_args, run_dir),
                 nprocs=args.num_gpus, join=True)
        return

    run_dir = None
    if rank_env == 0:
        from gansformer_amd import rundir
        desc = args.desc or _default_desc(args)
        run_dir = rundir.create_run_dir(args.result_dir, desc)
        rundir.tee_stdout(run_dir)
        rundir.save_submit_config(run_dir, dict(vars(args)))
        print("run dir:", run_dir)
        print("config:", json.dumps(loop_args, indent=2, default=repr))
    from gansformer_amd.training.loop import training_loop
    training_loop(run_dir=run_dir, **loop_args)


def _default_desc(args):
    return (f"{args.dataset}{args.res}-{args.transformer}"
            f"-k{args.components_num}-gpus{args.num_gpus}")


def _spawn_main(local_rank, world_size, loop_args, run_dir):
    os.environ["RANK"] = str(local_rank)
    os.environ["LOCAL_RANK"] = str(local_rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    if local_rank == 0:
        from gansformer_amd import rundir
        rundir.tee_stdout(run_dir)
    from gansformer_amd.training.loop import training_loop
    training_loop(run_dir=run_dir if local_rank == 0 else None, **loop_args)


if __name__ == "__main__":
    main()
