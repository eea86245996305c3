"""Multi-process (gloo, world=2) CPU tests for BucketedDataParallel: gradients
and trained params must match a single-process run on the combined batch."""
import os

import pytest
import torch
import torch.multiprocessing as mp
import torch.nn as nn

from deeplearning_amd.parallel.syncbn import all_reduce_norm


def _make_model(seed=0):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 4))


def _worker(rank, world, port, q):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK=str(rank))
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    from deeplearning_amd.parallel import BucketedDataParallel

    model = _make_model(seed=rank * 100)  # different init; broadcast must fix it
    ddp = BucketedDataParallel(model, bucket_cap_mb=0.0001)  # several tiny buckets
    opt = torch.optim.SGD(ddp.module.parameters(), lr=0.1)

    torch.manual_seed(42)
    xs = [torch.randn(8, 8) for _ in range(3)]
    ys = [torch.randn(8, 4) for _ in range(3)]
    for x, y in zip(xs, ys):
        xr = x.chunk(world)[rank]
        yr = y.chunk(world)[rank]
        loss = ((ddp(xr) - yr) ** 2).mean()
        opt.zero_grad(set_to_none=True)
        loss.backward()
        ddp.finalize()
        opt.step()
    if rank == 0:
        q.put([p.detach().numpy().copy() for p in ddp.module.parameters()])
    dist.barrier()
    dist.destroy_process_group()


def _worker_nosync(rank, world, port, q):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK=str(rank))
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    from deeplearning_amd.parallel import BucketedDataParallel

    model = _make_model()
    ddp = BucketedDataParallel(model, bucket_cap_mb=0.0001)
    torch.manual_seed(7 + rank)
    x = torch.randn(4, 8)
    with ddp.no_sync():
        loss = ddp(x).sum()
        loss.backward()
    g_local = [p.grad.clone() for p in ddp.module.parameters()]
    # second (sync) micro-batch accumulates then syncs
    loss = ddp(x).sum()
    loss.backward()
    ddp.finalize()
    g_sync = [p.grad.clone() for p in ddp.module.parameters()]
    q.put((rank, [g.numpy().copy() for g in g_local],
           [g.numpy().copy() for g in g_sync]))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.parametrize("world", [2, 4])
def test_ddp_matches_single_process(world):
    fn = _worker
    port = 29611 + world
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=fn, args=(r, world, port, q)) for r in range(world)]
    for p in procs:
        p.start()
    ddp_params = q.get()
    for p in procs:
        p.join(60)
        assert p.exitcode == 0

    # single-process run over the full batch
    model = _make_model(seed=0)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    torch.manual_seed(42)
    xs = [torch.randn(8, 8) for _ in range(3)]
    ys = [torch.randn(8, 4) for _ in range(3)]
    for x, y in zip(xs, ys):
        # DDP averages the two half-batch losses -> same as mean over full batch
        loss = ((model(x) - y) ** 2).mean()
        opt.zero_grad(set_to_none=True)
        loss.backward()
        opt.step()
    for p_ddp, p_ref in zip(ddp_params, model.parameters()):
        torch.testing.assert_close(torch.from_numpy(p_ddp), p_ref.detach(), atol=1e-5, rtol=1e-5)


def test_no_sync_accumulation():
    """no_sync micro-batch accumulates locally; the following synced backward
    must deliver grad == mean over ranks of the 2-micro-batch accumulated
    gradient (numeric check against an eager recomputation)."""
    world = 2
    port = 29613
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker_nosync, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, gl, gs = q.get()
        results[rank] = (gl, gs)
    for p in procs:
        p.join(60)
        assert p.exitcode == 0

    # eager reference: per-rank accumulated grads, then mean over ranks
    ref_model = _make_model()
    per_rank = []
    for r in range(world):
        for prm in ref_model.parameters():
            prm.grad = None
        torch.manual_seed(7 + r)
        x = torch.randn(4, 8)
        ref_model(x).sum().backward()
        ref_model(x).sum().backward()  # accumulate the 2nd micro-batch
        per_rank.append([prm.grad.clone() for prm in ref_model.parameters()])
    expected = [sum(gs) / world for gs in zip(*per_rank)]

    for r in range(world):
        gl, gs = results[r]
        # local (no_sync) grad on rank r == ONE micro-batch of rank r
        # (both micro-batches use the same x, so one = accumulated / 2)
        for g_got, g_exp in zip(gl, per_rank[r]):
            torch.testing.assert_close(torch.from_numpy(g_got), g_exp / 2,
                                       atol=1e-5, rtol=1e-5)
        for g_got, g_exp in zip(gs, expected):
            torch.testing.assert_close(torch.from_numpy(g_got), g_exp,
                                       atol=1e-5, rtol=1e-5)


def test_all_reduce_norm_single_proc():
    m = nn.Sequential(nn.Conv2d(3, 8, 3), nn.BatchNorm2d(8))
    all_reduce_norm(m)  # no-op without dist, must not raise


def test_syncbn_conversion_preserves_fused_relu():
    """Torch's converter silently drops BatchNorm2d(relu=True)'s activation;
    ours must keep it as SyncBN + ReLU with copied stats."""
    import torch.nn as nn

    from deeplearning_amd.ops import BatchNorm2d
    from deeplearning_amd.parallel.syncbn import convert_sync_batchnorm

    net = nn.Sequential(nn.Conv2d(3, 4, 3), BatchNorm2d(4, relu=True),
                        BatchNorm2d(4))
    with torch.no_grad():
        net[1].running_mean.fill_(0.5)
    conv = convert_sync_batchnorm(net)
    assert isinstance(conv[1], nn.Sequential)
    assert isinstance(conv[1][0], nn.SyncBatchNorm)
    assert isinstance(conv[1][1], nn.ReLU)
    assert isinstance(conv[2], nn.SyncBatchNorm)
    assert float(conv[1][0].running_mean[0]) == 0.5


def _launch_probe(tag):
    import torch.distributed as dist

    from deeplearning_amd.core.dist import cleanup, init_distributed
    info = init_distributed()
    assert info["world_size"] == 2, info
    t = torch.tensor([dist.get_rank() + 1.0])
    dist.all_reduce(t)
    assert float(t) == 3.0  # 1 + 2
    assert tag == "hello"
    cleanup()


def test_yolox_style_launch_spawns_world2():
    """launch() (ref YOLOX yolox/core/launch.py:39-147): free-port autodetect
    + mp.start_processes; workers join the group via init_distributed()."""
    from deeplearning_amd.core.dist import launch

    launch(_launch_probe, num_gpus_per_machine=2, args=("hello",))
