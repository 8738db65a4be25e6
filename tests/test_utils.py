"""utils/: tracing wrapper and bandwidth math (aux subsystem; the
reference has none — SURVEY.md §5)."""

import torch

from spmd import run_spmd


def _trace_worker(rank, world):
    import mpi4torch_amd as m
    from mpi4torch_amd.utils import trace

    comm = trace(m.COMM_WORLD)
    t = torch.ones(1000)
    comm.Allreduce(t, m.MPI_SUM)
    comm.Allgather(t, 0)
    recs = comm.trace_records()
    assert [r.op for r in recs] == ["Allreduce", "Allgather"]
    assert all(r.ms >= 0 for r in recs)
    assert recs[0].nbytes == 4000
    comm.clear_trace()
    assert comm.trace_records() == []

    # round-2 surface is traced too
    comm.Reducescatter(torch.ones(world * 2), 0, 2)
    comm.Alltoallv(torch.ones(world, 3), 1, 0, [1] * world, [3] * world)
    comm.AlltoallPairwise(torch.ones(world, 2), 0, [1] * world, [])
    h = comm.Iallgather(torch.ones(4))
    comm.Wait(h)
    ops = [r.op for r in comm.trace_records()]
    assert ops == ["Reducescatter", "Alltoallv", "AlltoallPairwise",
                   "Iallgather", "Wait"], ops


def test_trace_ws2():
    run_spmd(2, _trace_worker)


def test_busbw_math():
    from mpi4torch_amd.utils import busbw_gbps, algbw_gbps

    assert algbw_gbps(1e9, 1.0) == 1.0
    assert abs(busbw_gbps(1e9, 1.0, 8, "allreduce") - 2 * 7 / 8) < 1e-9
    assert busbw_gbps(1e9, 1.0, 1) == 1.0


def _ckpt_worker(rank, world, tmpdir):
    import os
    import torch
    import mpi4torch_amd as m
    from mpi4torch_amd.utils import save_checkpoint, load_checkpoint

    comm = m.COMM_WORLD
    torch.manual_seed(5)
    net = torch.nn.Linear(4, 3)
    opt = torch.optim.SGD(net.parameters(), lr=0.1)
    net(torch.randn(2, 4)).sum().backward()
    opt.step()
    path = os.path.join(tmpdir, "ckpt.pt")
    save_checkpoint(path, net, opt, extra={"step": 7})
    # ranks see the file via the shared filesystem; barrier via collective
    comm.Allreduce(torch.zeros(1), m.MPI_SUM)

    torch.manual_seed(1000 + rank)  # diverged fresh model per rank
    net2 = torch.nn.Linear(4, 3)
    opt2 = torch.optim.SGD(net2.parameters(), lr=0.1)
    extra = load_checkpoint(path, net2, opt2)
    if rank == 0:
        assert extra == {"step": 7}
    for p, q in zip(net.parameters(), net2.parameters()):
        assert torch.allclose(p, q), "checkpoint round-trip diverged"


def test_checkpoint_ws2(tmp_path_factory):
    import tempfile

    d = tempfile.mkdtemp()
    run_spmd(2, _ckpt_worker, d)


def _clip_worker(rank, world):
    import torch
    import mpi4torch_amd as m
    from mpi4torch_amd.utils import clip_grad_norm_sharded

    # shard: rank r holds a grad of known norm; global norm = sqrt(sum)
    g = torch.full((4,), float(rank + 1))
    p = torch.nn.Parameter(torch.zeros(4))
    p.grad = g.clone()
    total = clip_grad_norm_sharded([p], max_norm=1.0)
    import math

    expect = math.sqrt(sum(4 * (r + 1) ** 2 for r in range(world)))
    assert abs(float(total) - expect) < 1e-6
    # clipped to global norm 1: every slice scaled by 1/expect
    assert torch.allclose(p.grad, g / expect, atol=1e-6)

    # under the norm: untouched
    p2 = torch.nn.Parameter(torch.zeros(2))
    p2.grad = torch.full((2,), 1e-4)
    clip_grad_norm_sharded([p2], max_norm=10.0)
    assert torch.allclose(p2.grad, torch.full((2,), 1e-4))


def test_clip_sharded_ws3():
    run_spmd(3, _clip_worker)


def _sharded_ckpt_worker(rank, world, tmpdir):
    import mpi4torch_amd as m  # noqa: F401
    from mpi4torch_amd.parallel import (ShardedDataParallel,
                                        FullyShardedDataParallel,
                                        ZeroRedundancyOptimizer)
    from mpi4torch_amd.utils import (save_sharded_checkpoint,
                                     load_sharded_checkpoint)

    def train_steps(model, n):
        for i in range(n):
            x = torch.randn(6, 8)
            model(x).pow(2).sum().backward()
            model.step()
            model.zero_grad()

    # ---- ZeRO-2 round trip ----
    torch.manual_seed(11)
    net = torch.nn.Sequential(torch.nn.Linear(8, 8), torch.nn.Tanh(),
                              torch.nn.Linear(8, 4))
    model = ShardedDataParallel(net, torch.optim.Adam, bucket_cap_mb=0,
                                lr=0.05)
    torch.manual_seed(100 + rank)
    train_steps(model, 2)
    save_sharded_checkpoint(f"{tmpdir}/z2", model.sharded_state_dict())
    params_before = [p.detach().clone() for p in net.parameters()]
    train_steps(model, 2)  # diverge past the checkpoint

    torch.manual_seed(11)
    net2 = torch.nn.Sequential(torch.nn.Linear(8, 8), torch.nn.Tanh(),
                               torch.nn.Linear(8, 4))
    model2 = ShardedDataParallel(net2, torch.optim.Adam, bucket_cap_mb=0,
                                 lr=0.05)
    model2.load_sharded_state_dict(load_sharded_checkpoint(f"{tmpdir}/z2"))
    for p, q in zip(net2.parameters(), params_before):
        assert torch.allclose(p.detach(), q, atol=1e-6)

    # continued training from the restore matches Adam-state-dependent
    # trajectories only if optimizer state was restored too
    torch.manual_seed(500 + rank)
    x = torch.randn(6, 8)
    model2(x).pow(2).sum().backward()
    model2.step()

    # ---- FSDP round trip ----
    torch.manual_seed(12)
    fnet = torch.nn.Sequential(torch.nn.Linear(8, 8), torch.nn.GELU(),
                               torch.nn.Linear(8, 4))
    fsdp = FullyShardedDataParallel(fnet)
    opt = torch.optim.Adam(fsdp.shard_parameters(), lr=0.05)
    for _ in range(2):
        loss = fsdp(torch.randn(5, 8)).pow(2).sum()
        loss.backward()
        fsdp.finish_backward()
        opt.step()
        fsdp.refresh_shards()
    save_sharded_checkpoint(f"{tmpdir}/f", {
        "model": fsdp.sharded_state_dict(),
        "opt": opt.state_dict(),
    })
    shards_before = [u.shard.detach().clone() for u in fsdp._units]
    loss = fsdp(torch.randn(5, 8)).pow(2).sum()
    loss.backward()
    fsdp.finish_backward()
    opt.step()  # diverge

    blob = load_sharded_checkpoint(f"{tmpdir}/f")
    fsdp.load_sharded_state_dict(blob["model"])
    opt.load_state_dict(blob["opt"])
    for u, s in zip(fsdp._units, shards_before):
        assert torch.allclose(u.shard.detach(), s, atol=1e-7)
    # model still steps after restore
    loss = fsdp(torch.randn(5, 8)).pow(2).sum()
    loss.backward()
    fsdp.finish_backward()
    opt.step()

    # ---- ZeRO-1 round trip (state_dict already existed) ----
    torch.manual_seed(13)
    znet = torch.nn.Linear(8, 4)
    zopt = ZeroRedundancyOptimizer(znet.parameters(), torch.optim.Adam,
                                   lr=0.05)
    znet(torch.randn(4, 8)).sum().backward()
    zopt.step()
    save_sharded_checkpoint(f"{tmpdir}/z1", zopt.state_dict())
    restored = load_sharded_checkpoint(f"{tmpdir}/z1")
    zopt.load_state_dict(restored)

    # world-size mismatch raises with guidance
    try:
        load_sharded_checkpoint(f"{tmpdir}/does-not-exist")
        raise AssertionError("expected FileNotFoundError")
    except FileNotFoundError:
        pass


def test_sharded_checkpoint_ws2(tmp_path):
    run_spmd(2, _sharded_ckpt_worker, str(tmp_path))


def test_sharded_checkpoint_ws5(tmp_path):
    run_spmd(5, _sharded_ckpt_worker, str(tmp_path))
