#!/usr/bin/env python3
"""Flagship benchmark: FFHQ-256 duplex-attention GANsformer training step
(BASELINE.json config 2), synthetic data, random-init weights, bf16
compute blocks. One full production training step per "step": D
main+backward+Adam, lazy R1 (every 16), G main+backward+Adam, lazy
path-length (every 8), EMA update, bucketed RCCL grad all-reduce — via
the same GANTrainer the training loop uses.

Contract (driver): `python bench.py --gpus N --steps K --warmup W`; for
N>1 launched under torch.distributed.run, one rank per GPU over RCCL.
W untimed warmup steps, then exactly K timed steps bracketed by
barrier+synchronize; MAX step-time over ranks; rank 0 prints ONE JSON
line with the whole-job imgs/sec.
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=16)
    p.add_argument("--warmup", type=int, default=4)
    p.add_argument("--batch-gpu", type=int, default=32)
    p.add_argument("--res", type=int, default=256)
    p.add_argument("--components-num", type=int, default=16)
    p.add_argument("--transformer", default="duplex")
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--device", default=None)
    args = p.parse_args(argv)

    from gansformer_amd.models.networks import Discriminator, Generator
    from gansformer_amd.parallel import broadcast_params
    from gansformer_amd.parallel.dist import cleanup, setup_distributed
    from gansformer_amd.training.trainer import GANTrainer

    rank, world_size, dev = setup_distributed()
    if args.device:
        dev = torch.device(args.device)
    if world_size != args.gpus and rank == 0:
        print(f"# note: WORLD_SIZE={world_size} != --gpus {args.gpus}; "
              f"using {world_size}", file=sys.stderr)
    n_gpus = world_size
    torch.manual_seed(args.seed * 100 + rank)

    batch_gpu = args.batch_gpu
    global_batch = batch_gpu * n_gpus

    G = Generator(img_resolution=args.res, num_components=args.components_num,
                  transformer=args.transformer).to(dev).train()
    D = Discriminator(img_resolution=args.res).to(dev).train()
    import copy
    Gs = copy.deepcopy(G).eval()
    for prm in Gs.parameters():
        prm.requires_grad_(False)
    broadcast_params(G)
    broadcast_params(D)

    trainer = GANTrainer(G, D, Gs, dev, batch_gpu, global_batch)

    gen = torch.Generator(device=dev).manual_seed(1234 + rank)

    def next_real():
        # synthetic data of the benchmark config's shape, made on-device
        u8 = torch.randint(0, 256, (batch_gpu, 3, args.res, args.res),
                           device=dev, dtype=torch.uint8, generator=gen)
        return u8.float().div(127.5).sub(1.0)

    def sync():
        if dist.is_initialized():
            dist.barrier()
        if dev.type == "cuda":
            torch.cuda.synchronize(dev)

    step_idx = 0
    nimg = 0
    for _ in range(args.warmup):
        trainer.step(next_real, step_idx, nimg)
        step_idx += 1
        nimg += global_batch

    sync()
    t0 = time.time()
    for _ in range(args.steps):
        trainer.step(next_real, step_idx, nimg)
        step_idx += 1
        nimg += global_batch
    sync()
    elapsed = time.time() - t0

    # MAX over ranks
    t = torch.tensor([elapsed], dtype=torch.float64,
                     device=dev if dev.type == "cuda" else "cpu")
    if dist.is_initialized():
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t.item())

    ms_per_step = elapsed / args.steps * 1000.0
    imgs_sec = global_batch * args.steps / elapsed

    if rank == 0:
        print(json.dumps({
            "metric": "imgs/sec (FFHQ-256 duplex-attn GANsformer training)",
            "value": imgs_sec,
            "unit": "imgs/sec",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": f"gansformer-{args.transformer}-k{args.components_num}",
                "global_batch": global_batch,
                "resolution": args.res,
                "seq_len": args.res * args.res,
                "parallelism": f"dp{n_gpus}",
            },
        }), flush=True)
    cleanup()


if __name__ == "__main__":
    main()
