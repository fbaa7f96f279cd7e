"""Multi-process data-parallel tests over the gloo backend (CPU, world 2).
Validates the bucketed GradReducer: averaged gradients match the single-
process gradient of the combined batch, and replicas stay in lockstep."""

import os

import torch
import torch.distributed as dist
import torch.multiprocessing as mp
import pytest


def _setup(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["RANK"] = str(rank)
    dist.init_process_group("gloo", rank=rank, world_size=world)


def _worker_reducer(rank, world, port, q):
    _setup(rank, world, port)
    from dsin_amd.parallel import GradReducer
    torch.manual_seed(0)  # same params on both ranks
    model = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.ReLU(),
                                torch.nn.Linear(32, 4))
    params = list(model.parameters())
    reducer = GradReducer(params, bucket_bytes=256)  # force several buckets
    torch.manual_seed(100 + rank)  # different data per rank
    x = torch.randn(8, 16)
    reducer.prepare()
    model(x).pow(2).mean().backward()
    reducer.finalize()
    grads = [p.grad.clone() for p in params]
    q.put((rank, [g.numpy() for g in grads], x.numpy()))
    dist.destroy_process_group()


def test_grad_reducer_averages():
    import socket
    with socket.socket() as _s:
        _s.bind(("127.0.0.1", 0))
        port = _s.getsockname()[1]
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_reducer, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, grads, x = q.get(timeout=120)
        results[rank] = (grads, x)
    for p in procs:
        p.join(timeout=60)

    # reducer output must be identical across ranks
    for g0, g1 in zip(results[0][0], results[1][0]):
        assert abs(g0 - g1).max() < 1e-6

    # and equal to the manual average of per-rank gradients
    torch.manual_seed(0)
    model = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.ReLU(),
                                torch.nn.Linear(32, 4))
    accum = None
    for r in range(2):
        model.zero_grad()
        x = torch.from_numpy(results[r][1])
        model(x).pow(2).mean().backward()
        gs = [p.grad.clone() for p in model.parameters()]
        accum = gs if accum is None else [a + b for a, b in zip(accum, gs)]
    expect = [a / 2 for a in accum]
    for g, e in zip(results[0][0], expect):
        assert abs(torch.from_numpy(g) - e).max() < 1e-5


def _worker_flat_reducer(rank, world, port, q):
    _setup(rank, world, port)
    from dsin_amd.ops.adam import FusedAdam
    from dsin_amd.parallel import FlatGradReducer
    torch.manual_seed(0)  # same params on all ranks
    model = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.ReLU(),
                                torch.nn.Linear(32, 4))
    params = list(model.parameters())
    opt = FusedAdam(params, lr=0.0)  # lr 0: only the reducer matters here
    reducer = FlatGradReducer(opt, bucket_bytes=256)  # force several buckets
    torch.manual_seed(100 + rank)  # different data per rank
    x = torch.randn(8, 16)
    for p in params:
        p.grad = None
    reducer.prepare()
    model(x).pow(2).mean().backward()
    reducer.finalize()
    q.put((rank, opt.flat_g.numpy().copy(), x.numpy(),
           [list(s) for s in opt._slices]))
    dist.destroy_process_group()


def test_flat_grad_reducer_averages():
    """The fused-path overlapped reducer (the path the GPU runs) must leave
    flat_g equal to the cross-rank mean gradient."""
    import socket
    with socket.socket() as _s:
        _s.bind(("127.0.0.1", 0))
        port = _s.getsockname()[1]
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    world = 2
    procs = [ctx.Process(target=_worker_flat_reducer, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, flat_g, x, slices = q.get(timeout=120)
        results[rank] = (flat_g, x, slices)
    for p in procs:
        p.join(timeout=60)

    for r in range(1, world):
        assert abs(results[0][0] - results[r][0]).max() < 1e-6

    # equals the manual average of per-rank flat gradients
    torch.manual_seed(0)
    model = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.ReLU(),
                                torch.nn.Linear(32, 4))
    params = list(model.parameters())
    slices = results[0][2]
    accum = torch.zeros(sum(n for _, n in slices))
    for r in range(world):
        model.zero_grad()
        x = torch.from_numpy(results[r][1])
        model(x).pow(2).mean().backward()
        for p, (ofs, n) in zip(params, slices):
            accum[ofs:ofs + n] += p.grad.reshape(-1)
    expect = accum / world
    assert abs(torch.from_numpy(results[0][0]) - expect).max() < 1e-5


def _worker_trainstep(rank, world, port, q, fused):
    _setup(rank, world, port)
    from dsin_amd import config as cm
    from dsin_amd.models import DSIN
    from dsin_amd.training import Trainer
    from dsin_amd.data import SyntheticStereo
    here = os.path.dirname(os.path.abspath(__file__))
    ae, _ = cm.parse(os.path.join(here, "..", "run_configs", "ae_run_configs"))
    pc, _ = cm.parse(os.path.join(here, "..", "run_configs", "pc_run_configs"))
    ae.crop_size = (64, 96)
    ae.y_patch_size = (16, 16)
    torch.manual_seed(123 + rank)  # DIFFERENT init; broadcast must fix it
    model = DSIN(ae, pc)
    tr = Trainer(model, ae, pc, num_training_imgs=100, fused_adam=fused)
    gen = SyntheticStereo(64, 96, seed=555 + rank)
    for _ in range(2):
        x, y = gen.next_batch()
        loss, bpp = tr.train_step(x, y)
    w = model.encoder.h1.conv.weight.detach().clone()
    q.put((rank, float(loss), w.numpy()))
    dist.destroy_process_group()


def _run_trainstep_world(world, fused):
    import socket
    with socket.socket() as _s:
        _s.bind(("127.0.0.1", 0))
        port = _s.getsockname()[1]
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_trainstep,
                         args=(r, world, port, q, fused))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, loss, w = q.get(timeout=600)
        results[rank] = (loss, w)
    for p in procs:
        p.join(timeout=60)
    # identical post-step weights despite different data and different seeds
    for r in range(1, world):
        assert abs(results[0][1] - results[r][1]).max() < 1e-6


def test_flat_grad_reducer_eight_ranks():
    """8-rank flat-reducer equivalence (the driver's max GPU count)."""
    import socket
    with socket.socket() as _s:
        _s.bind(("127.0.0.1", 0))
        port = _s.getsockname()[1]
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    world = 8
    procs = [ctx.Process(target=_worker_flat_reducer, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, flat_g, x, slices = q.get(timeout=300)
        results[rank] = flat_g
    for p in procs:
        p.join(timeout=60)
    for r in range(1, world):
        assert abs(results[0] - results[r]).max() < 1e-6


def test_full_train_step_replicas_stay_synced():
    _run_trainstep_world(2, fused=None)


def test_full_train_step_fused_flat_reducer_world4():
    """4-rank fused-Adam + FlatGradReducer — the exact optimizer/reducer
    combination an 8-GPU RCCL run uses, over gloo on CPU."""
    _run_trainstep_world(4, fused=True)
