"""RCCL-on-device proof (single rank — gpurun boxes expose one GPU and
RCCL rejects two ranks on one device, cf. gpurun_out/ddp2.log from r01).

What this CAN prove on one GPU: the nccl(=RCCL) backend initialises, the
bucketed GradReducer launches real RCCL all-reduces on device tensors
and writes back correct averages, broadcast_params and the FID feature
path run under the nccl backend without the CPU-tensor crash. The
multi-rank averaging math itself is covered by the world-2/8 gloo tests
(tests/test_ddp_gloo.py) over the identical code path.
"""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def nccl_group():
    assert torch.cuda.is_available()
    import torch.distributed as dist
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29581")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    if not dist.is_initialized():
        dist.init_process_group(backend="nccl", rank=0, world_size=1)
    yield dist
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_rccl_allreduce_and_broadcast(nccl_group):
    dist = nccl_group
    dev = torch.device("cuda:0")
    x = torch.arange(1024, device=dev, dtype=torch.float32)
    dist.all_reduce(x, op=dist.ReduceOp.SUM)
    torch.cuda.synchronize()
    assert torch.equal(x, torch.arange(1024, device=dev, dtype=torch.float32))
    net = torch.nn.Linear(8, 8).to(dev)
    from gansformer_amd.parallel.ddp import broadcast_params
    broadcast_params(net)
    torch.cuda.synchronize()


@pytest.mark.timeout(300)
def test_rccl_grad_reducer_path(nccl_group):
    """Full bucket machinery over real RCCL collectives, incl. a
    rounds=2 accumulation step: reduced grads must equal the (single
    rank's) accumulated grads exactly."""
    from gansformer_amd.parallel.ddp import GradReducer
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    net = torch.nn.Sequential(
        torch.nn.Linear(64, 128), torch.nn.LeakyReLU(),
        torch.nn.Linear(128, 1)).to(dev)
    red = GradReducer(net, bucket_mb=0.0002, force_enabled=True)
    assert red.enabled and len(red.buckets) >= 2

    rounds = 2
    red.prepare(rounds)
    xs = [torch.randn(16, 64, device=dev) for _ in range(rounds)]
    for x in xs:
        (net(x).sum() / rounds).backward()
    red.finalize()
    torch.cuda.synchronize()
    got = {n: p.grad.clone() for n, p in net.named_parameters()}

    # plain autograd reference (no reducer)
    for p in net.parameters():
        p.grad = None
    for x in xs:
        (net(x).sum() / rounds).backward()
    for n, p in net.named_parameters():
        assert torch.allclose(got[n], p.grad, atol=1e-6), n


@pytest.mark.timeout(300)
def test_rccl_fid_features(nccl_group):
    """collect_features under the nccl backend (world 1): must not touch
    the CPU-tensor gather path that RCCL rejects."""
    from gansformer_amd.metrics.fid import RandomConvFeatures, collect_features
    dev = torch.device("cuda:0")
    ext = RandomConvFeatures(feature_dim=64).to(dev).eval()

    def batch_fn(n):
        return torch.randn(n, 3, 32, 32, device=dev)

    feats = collect_features(batch_fn, ext, num_images=8, batch_size=4,
                             device=dev, rank=0, world_size=1)
    assert feats.shape == (8, 64)
