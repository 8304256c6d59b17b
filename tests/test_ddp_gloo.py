"""Distributed data-parallel correctness over gloo (CPU, world_size=2):
the bucketed overlapped all-reduce must produce the average of per-rank
gradients, including lazy-reg steps where only a subset of grads exist."""

import multiprocessing as mp
import os

import pytest
import torch


def _worker(rank, world, port, q):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    import torch.distributed as dist
    from gansformer_amd.parallel.ddp import GradReducer, broadcast_params
    from gansformer_amd.parallel.dist import setup_distributed

    setup_distributed(backend="gloo")
    torch.manual_seed(0)  # same init everywhere
    net = torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.LeakyReLU(),
        torch.nn.Linear(32, 1))
    broadcast_params(net)
    red = GradReducer(net, bucket_mb=0.0001)  # force multiple buckets
    assert len(red.buckets) >= 2

    torch.manual_seed(100 + rank)  # different data per rank
    x = torch.randn(8, 16)
    red.prepare()
    net(x).sum().backward()
    red.finalize()
    g0 = net[0].weight.grad.clone()

    # lazy-reg-style step: only the last layer gets grads
    for p in net.parameters():
        p.grad = None
    red.prepare()
    h = net[0](x).detach()
    h = net[2](torch.nn.functional.leaky_relu(h))
    h.sum().backward()
    red.finalize()
    g_last = net[2].weight.grad.clone()
    assert net[0].weight.grad is None

    q.put((rank, g0.numpy(), g_last.numpy()))
    dist.destroy_process_group()


def _expected(world):
    grads0, grads_last = [], []
    torch.manual_seed(0)
    net = torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.LeakyReLU(),
        torch.nn.Linear(32, 1))
    sd = {k: v.clone() for k, v in net.state_dict().items()}
    for rank in range(world):
        net.load_state_dict(sd)
        for p in net.parameters():
            p.grad = None
        torch.manual_seed(100 + rank)
        x = torch.randn(8, 16)
        net(x).sum().backward()
        grads0.append(net[0].weight.grad.clone())
        for p in net.parameters():
            p.grad = None
        h = net[0](x).detach()
        net[2](torch.nn.functional.leaky_relu(h)).sum().backward()
        grads_last.append(net[2].weight.grad.clone())
    return (torch.stack(grads0).mean(0), torch.stack(grads_last).mean(0))


@pytest.mark.timeout(120)
def test_grad_reducer_two_ranks():
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29765
    procs = [ctx.Process(target=_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, g0, gl = q.get(timeout=110)
        results[rank] = (torch.tensor(g0), torch.tensor(gl))
    for p in procs:
        p.join(timeout=30)
    exp0, expl = _expected(world)
    for rank in range(world):
        g0, gl = results[rank]
        assert torch.allclose(g0, exp0, atol=1e-6), f"rank {rank} main grads"
        assert torch.allclose(gl, expl, atol=1e-6), f"rank {rank} lazy grads"
