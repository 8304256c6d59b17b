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


# ---------------------------------------------------------------------------
# Gradient accumulation (rounds >= 2): the reducer must ship the FULLY
# accumulated per-rank gradient, not round 1's partial one (round-1 bug
# flagged in VERDICT r01).
# ---------------------------------------------------------------------------

def _worker_accum(rank, world, port, q):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    import torch.distributed as dist
    from gansformer_amd.parallel.ddp import GradReducer, broadcast_params
    from gansformer_amd.parallel.dist import setup_distributed

    setup_distributed(backend="gloo")
    torch.manual_seed(0)
    net = torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.LeakyReLU(),
        torch.nn.Linear(32, 1))
    broadcast_params(net)
    red = GradReducer(net, bucket_mb=0.0001)
    assert len(red.buckets) >= 2

    rounds = 3
    red.prepare(rounds)
    for rnd in range(rounds):
        torch.manual_seed(1000 + rank * 10 + rnd)
        x = torch.randn(8, 16)
        (net(x).sum() / rounds).backward()
    red.finalize()
    q.put((rank, net[0].weight.grad.numpy().copy(),
           net[2].weight.grad.numpy().copy()))
    dist.destroy_process_group()


def _expected_accum(world, rounds=3):
    torch.manual_seed(0)
    net = torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.LeakyReLU(),
        torch.nn.Linear(32, 1))
    sd = {k: v.clone() for k, v in net.state_dict().items()}
    g0, gl = [], []
    for rank in range(world):
        net.load_state_dict(sd)
        for p in net.parameters():
            p.grad = None
        for rnd in range(rounds):
            torch.manual_seed(1000 + rank * 10 + rnd)
            x = torch.randn(8, 16)
            (net(x).sum() / rounds).backward()
        g0.append(net[0].weight.grad.clone())
        gl.append(net[2].weight.grad.clone())
    return torch.stack(g0).mean(0), torch.stack(gl).mean(0)


@pytest.mark.timeout(120)
def test_grad_reducer_accumulation_rounds():
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_accum, args=(r, world, 29766, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, g0, gl = q.get(timeout=110)
        results[rank] = (torch.tensor(g0), torch.tensor(gl))
    for p in procs:
        p.join(timeout=30)
    exp0, expl = _expected_accum(world)
    for rank in range(world):
        g0, gl = results[rank]
        assert torch.allclose(g0, exp0, atol=1e-6), f"rank {rank} accum grads"
        assert torch.allclose(gl, expl, atol=1e-6), f"rank {rank} accum grads"


# ---------------------------------------------------------------------------
# World-8 dress rehearsal of the FULL trainer step (tiny model): after N
# steps with per-rank data and gradient accumulation + lazy reg, all ranks
# must hold bitwise-identical parameters, and sync_ranks() must make
# w_avg / pl_mean / Gs identical too.
# ---------------------------------------------------------------------------

def _worker_trainer(rank, world, port, q):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    torch.set_num_threads(1)
    import copy
    import torch.distributed as dist
    from gansformer_amd.models.networks import Discriminator, Generator
    from gansformer_amd.parallel.ddp import broadcast_params
    from gansformer_amd.parallel.dist import setup_distributed
    from gansformer_amd.training.trainer import GANTrainer

    setup_distributed(backend="gloo")
    torch.manual_seed(0)
    G = Generator(z_dim=32, w_dim=32, img_resolution=16, num_components=4,
                  transformer="simplex", channel_base=512, channel_max=64,
                  bf16_res_count=0, mapping_layers=2)
    D = Discriminator(img_resolution=16, channel_base=512, channel_max=64,
                      mbstd_group_size=2, bf16_res_count=0)
    Gs = copy.deepcopy(G).eval()
    for p in Gs.parameters():
        p.requires_grad_(False)
    broadcast_params(G)
    broadcast_params(D)
    broadcast_params(Gs)

    dev = torch.device("cpu")
    batch_gpu, rounds = 2, 2
    batch_size = batch_gpu * world * rounds
    tr = GANTrainer(G, D, Gs, dev, batch_gpu, batch_size,
                    d_reg_interval=2, g_reg_interval=3, rounds=rounds,
                    bucket_mb=0.25)
    torch.manual_seed(5000 + rank)  # per-rank data + z draws

    def next_real():
        return torch.randn(batch_gpu, 3, 16, 16)

    nimg = 0
    for step in range(4):  # hits d_reg at 0,2 and g_reg at 0,3
        tr.step(next_real, step, nimg)
        nimg += batch_size
    tr.sync_ranks()

    vec = torch.cat([p.detach().reshape(-1) for p in G.parameters()]
                    + [p.detach().reshape(-1) for p in D.parameters()])
    svec = torch.cat([p.detach().reshape(-1) for p in Gs.parameters()]
                     + [b.detach().reshape(-1) for b in Gs.buffers()])
    aux = torch.cat([G.mapping.w_avg.reshape(-1),
                     tr.pl_reg.pl_mean.reshape(-1)])
    q.put((rank, vec.numpy().copy(), svec.numpy().copy(),
           aux.numpy().copy()))
    dist.destroy_process_group()


@pytest.mark.timeout(420)
def test_trainer_world8_rank_identical():
    world = 8
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_trainer, args=(r, world, 29767, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, vec, svec, aux = q.get(timeout=400)
        results[rank] = (torch.tensor(vec), torch.tensor(svec),
                         torch.tensor(aux))
    for p in procs:
        p.join(timeout=30)
    v0, s0, a0 = results[0]
    assert torch.isfinite(v0).all()
    for rank in range(1, world):
        v, s, a = results[rank]
        assert torch.equal(v, v0), f"rank {rank} params diverged"
        assert torch.equal(s, s0), f"rank {rank} Gs diverged after sync"
        assert torch.equal(a, a0), f"rank {rank} w_avg/pl_mean diverged"


# ---------------------------------------------------------------------------
# Multi-rank FID feature gather (gloo): the gathered feature matrix must
# contain every rank's features (the old path crashed
# under RCCL; this exercises the same code over gloo).
# ---------------------------------------------------------------------------

def _worker_fid(rank, world, port, q):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    import torch.distributed as dist
    from gansformer_amd.metrics.fid import collect_features
    from gansformer_amd.parallel.dist import setup_distributed

    setup_distributed(backend="gloo")
    dev = torch.device("cpu")

    class ConstFeat(torch.nn.Module):
        def forward(self, x):
            return torch.full((x.shape[0], 4), float(rank) + x.mean() * 0)

    def batch_fn(n):
        return torch.zeros(n, 3, 8, 8)

    feats = collect_features(batch_fn, ConstFeat(), num_images=12,
                             batch_size=4, device=dev, rank=rank,
                             world_size=world)
    q.put((rank, feats.copy()))
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_fid_feature_gather_two_ranks():
    import numpy as np
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_fid, args=(r, world, 29768, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, feats = q.get(timeout=110)
        results[rank] = feats
    for p in procs:
        p.join(timeout=30)
    for rank in range(world):
        f = results[rank]
        assert f.shape == (12, 4)
        vals = set(np.unique(f).tolist())
        assert vals == {0.0, 1.0}, f"gather missing a rank's features: {vals}"


# ---------------------------------------------------------------------------
# Collective launch-order stress: 20 full trainer steps with per-rank
# style-mixing randomness (different graph branches per rank) and lazy
# reg. The immediate-launch bucket design requires identical hook order
# on every rank; divergent params or a hang here would catch an
# order/participation mismatch before it deadlocks real RCCL.
# ---------------------------------------------------------------------------

def _worker_stress(rank, world, port, q):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    torch.set_num_threads(1)
    import copy
    import torch.distributed as dist
    from gansformer_amd.models.networks import Discriminator, Generator
    from gansformer_amd.parallel.ddp import broadcast_params
    from gansformer_amd.parallel.dist import setup_distributed
    from gansformer_amd.training.trainer import GANTrainer

    setup_distributed(backend="gloo")
    torch.manual_seed(0)
    G = Generator(z_dim=32, w_dim=32, img_resolution=16, num_components=4,
                  transformer="duplex", channel_base=512, channel_max=64,
                  bf16_res_count=0, mapping_layers=2, style_mixing_prob=0.9)
    D = Discriminator(img_resolution=16, channel_base=512, channel_max=64,
                      mbstd_group_size=2, bf16_res_count=0)
    Gs = copy.deepcopy(G).eval()
    for p in Gs.parameters():
        p.requires_grad_(False)
    broadcast_params(G)
    broadcast_params(D)
    broadcast_params(Gs)
    tr = GANTrainer(G, D, Gs, torch.device("cpu"), 2, 2 * world,
                    d_reg_interval=3, g_reg_interval=2, bucket_mb=0.1)
    torch.manual_seed(7000 + rank)  # rank-divergent mixing draws + data
    for step in range(20):
        tr.step(lambda: torch.randn(2, 3, 16, 16), step, step * 2 * world)
    vec = torch.cat([p.detach().reshape(-1) for p in G.parameters()]
                    + [p.detach().reshape(-1) for p in D.parameters()])
    q.put((rank, vec.numpy().copy()))
    dist.destroy_process_group()


@pytest.mark.timeout(420)
def test_trainer_style_mixing_order_stress():
    world = 4
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_stress, args=(r, world, 29769, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, vec = q.get(timeout=400)
        results[rank] = torch.tensor(vec)
    for p in procs:
        p.join(timeout=30)
    for rank in range(1, world):
        assert torch.equal(results[rank], results[0]), f"rank {rank}"
    assert torch.isfinite(results[0]).all()


# ---------------------------------------------------------------------------
# Collective stop: an abort signal seen by ONE rank must stop ALL ranks
# cleanly (rank-local want_stop would hang the others on the next
# collective).
# ---------------------------------------------------------------------------

def _worker_stop(rank, world, port, q):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    torch.set_num_threads(1)
    import torch.distributed as dist
    from gansformer_amd.training.loop import training_loop
    import tempfile
    run_dir = os.path.join(tempfile.gettempdir(), f"stoptest-{port}")
    if rank == 0:
        os.makedirs(run_dir, exist_ok=True)
        with open(os.path.join(run_dir, "abort.txt"), "w") as f:
            f.write("stop")
    out = training_loop(
        run_dir=run_dir if rank == 0 else None,
        dataset_args=dict(dataset="synthetic", resolution=16,
                          synthetic_size=16),
        G_args=dict(z_dim=16, w_dim=16, num_components=2,
                    transformer="none", channel_base=256, channel_max=16,
                    bf16_res_count=0, mapping_layers=1),
        D_args=dict(channel_base=256, channel_max=16, mbstd_group_size=2,
                    bf16_res_count=0),
        total_kimg=10.0,  # would be ~2500 steps without the abort
        batch_gpu=2, snapshot_kimg=5.0, image_snapshot_kimg=5.0,
        log_interval_kimg=0.004, num_workers=0, seed=0)
    q.put((rank, out["steps"]))


@pytest.mark.timeout(300)
def test_abort_stops_all_ranks():
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_stop, args=(r, world, 29770, q))
             for r in range(world)]
    for p in procs:
        p.start()
    steps = {}
    for _ in range(world):
        rank, s = q.get(timeout=280)
        steps[rank] = s
    for p in procs:
        p.join(timeout=30)
    # both ranks stopped early (well before 2500 steps) at the same step
    assert steps[0] == steps[1]
    assert steps[0] < 50, steps


# ---------------------------------------------------------------------------
# Full training_loop with FID metrics on 2 ranks: the metric path's
# feature gather and the snapshot-time sync_ranks must line up as
# collectives on every rank (this is the path the 8-GPU run exercises).
# ---------------------------------------------------------------------------

def _worker_loop_metrics(rank, world, port, q):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    torch.set_num_threads(1)
    import tempfile
    from gansformer_amd.training.loop import training_loop
    run_dir = None
    if rank == 0:
        run_dir = os.path.join(tempfile.gettempdir(), f"loopm-{port}")
        os.makedirs(run_dir, exist_ok=True)
    out = training_loop(
        run_dir=run_dir,
        dataset_args=dict(dataset="clevr-synth", resolution=16,
                          synthetic_size=32),
        G_args=dict(z_dim=16, w_dim=16, num_components=2,
                    transformer="simplex", channel_base=256, channel_max=16,
                    bf16_res_count=0, mapping_layers=1),
        D_args=dict(channel_base=256, channel_max=16, mbstd_group_size=2,
                    bf16_res_count=0),
        total_kimg=0.008, batch_gpu=2,  # 4 global/step -> 2 steps
        metrics=["fid256"], metric_images=8, metric_kimg=0.008,
        snapshot_kimg=0.008, image_snapshot_kimg=0.016,
        log_interval_kimg=0.004, num_workers=0, seed=0)
    fid_file = (os.path.join(run_dir, "metric-fid256.txt")
                if rank == 0 else None)
    n_fid = 0
    if fid_file and os.path.exists(fid_file):
        n_fid = len(open(fid_file).readlines())
    q.put((rank, out["steps"], n_fid))


@pytest.mark.timeout(420)
def test_training_loop_metrics_world2():
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_loop_metrics,
                         args=(r, world, 29771, q)) for r in range(world)]
    for p in procs:
        p.start()
    res = {}
    for _ in range(world):
        rank, steps, n_fid = q.get(timeout=400)
        res[rank] = (steps, n_fid)
    for p in procs:
        p.join(timeout=30)
    assert res[0][0] == res[1][0] == 2
    assert res[0][1] >= 1  # FID eval recorded by rank 0


# ---------------------------------------------------------------------------
# World-2 resume: training restarted from a snapshot must continue on
# all ranks (weights broadcast from the pkl, trainer state restored,
# collectives aligned) and finish with rank-identical parameters.
# ---------------------------------------------------------------------------

def _worker_resume(rank, world, port, q, run_root):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    torch.set_num_threads(1)
    import glob as g
    from gansformer_amd.training.loop import training_loop
    kw = dict(
        dataset_args=dict(dataset="clevr-synth", resolution=16,
                          synthetic_size=32),
        G_args=dict(z_dim=16, w_dim=16, num_components=2,
                    transformer="simplex", channel_base=256, channel_max=16,
                    bf16_res_count=0, mapping_layers=1),
        D_args=dict(channel_base=256, channel_max=16, mbstd_group_size=2,
                    bf16_res_count=0),
        batch_gpu=2, snapshot_kimg=0.008, image_snapshot_kimg=1.0,
        log_interval_kimg=0.004, num_workers=0, seed=0)
    d1 = os.path.join(run_root, "phase1")
    if rank == 0:
        os.makedirs(d1, exist_ok=True)
    training_loop(run_dir=d1 if rank == 0 else None, total_kimg=0.008, **kw)
    pkls = sorted(g.glob(os.path.join(d1, "network-snapshot-*.pkl")))
    assert pkls, "no snapshot written"
    d2 = os.path.join(run_root, "phase2")
    if rank == 0:
        os.makedirs(d2, exist_ok=True)
    out = training_loop(run_dir=d2 if rank == 0 else None,
                        resume_pkl=pkls[-1], total_kimg=0.016, **kw)
    # report final params for cross-rank comparison
    pkl2 = sorted(g.glob(os.path.join(d2, "network-snapshot-*.pkl")))
    q.put((rank, out["cur_nimg"], len(pkl2) if rank == 0 else -1))


@pytest.mark.timeout(420)
def test_world2_resume(tmp_path_factory):
    import tempfile
    world = 2
    run_root = tempfile.mkdtemp(prefix="resume2-")
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_resume,
                         args=(r, world, 29772, q, run_root))
             for r in range(world)]
    for p in procs:
        p.start()
    res = {}
    for _ in range(world):
        rank, nimg, npkl = q.get(timeout=400)
        res[rank] = (nimg, npkl)
    for p in procs:
        p.join(timeout=30)
    assert res[0][0] == res[1][0] >= 16
    assert res[0][1] >= 1
