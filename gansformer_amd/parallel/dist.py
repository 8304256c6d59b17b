"""Distributed setup: one process per GPU over RCCL.

The reference used single-process in-graph replication with TF's bundled
NCCL (`nccl_ops.all_sum`, ref src/dnnlib/tflib/optimizer.py [R]). The
MI355X-native design is one rank per GPU with torch.distributed — backend
"nccl" IS RCCL on ROCm, riding the 7x ~153 GB/s xGMI links; "gloo" is
used for CPU-only tests of the same code path.
"""

from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def setup_distributed(backend=None, timeout_sec=1800):
    """Initialise from torchrun-style env vars; no-op if WORLD_SIZE<=1.

    Returns (rank, world_size, device).
    """
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    use_gpu = torch.cuda.is_available()
    if use_gpu:
        ndev = torch.cuda.device_count()
        if local_rank >= ndev:
            # RCCL does NOT allow two ranks on one device (observed:
            # NCCL "invalid usage" for 2-ranks-1-GPU). Oversubscription
            # is opt-in for gloo-on-GPU tests only; a misconfigured
            # production launch must fail loudly here.
            if os.environ.get("GANSFORMER_OVERSUBSCRIBE_GPU") == "1":
                device = torch.device("cuda", local_rank % ndev)
            else:
                raise RuntimeError(
                    f"LOCAL_RANK={local_rank} but only {ndev} visible GPU(s);"
                    " one rank per GPU is required (set"
                    " GANSFORMER_OVERSUBSCRIBE_GPU=1 only for gloo tests)")
        else:
            device = torch.device("cuda", local_rank)
    else:
        device = torch.device("cpu")
    if use_gpu:
        torch.cuda.set_device(device)
    if world_size > 1 and not dist.is_initialized():
        if backend is None:
            backend = "nccl" if use_gpu else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29571")
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world_size,
            timeout=datetime.timedelta(seconds=timeout_sec))
    return rank, world_size, device


def get_rank():
    return dist.get_rank() if dist.is_initialized() else 0


def get_world_size():
    return dist.get_world_size() if dist.is_initialized() else 1


def is_main():
    return get_rank() == 0


def barrier():
    if dist.is_initialized():
        dist.barrier()


def any_rank(flag: bool) -> bool:
    """Collective OR of a per-rank boolean. Used for stop decisions:
    abort files / SIGTERM may reach only one rank, and a rank leaving
    the training loop alone would hang the others on the next
    collective."""
    if not dist.is_initialized() or dist.get_world_size() == 1:
        return flag
    backend = str(dist.get_backend()).lower()
    dev = torch.device("cuda", torch.cuda.current_device()) \
        if backend in ("nccl", "rccl") and torch.cuda.is_available() \
        else torch.device("cpu")
    t = torch.tensor([1 if flag else 0], dtype=torch.int32, device=dev)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return bool(t.item())


def cleanup():
    if dist.is_initialized():
        dist.destroy_process_group()
