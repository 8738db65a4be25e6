"""Chunked pack->wire pipelining (MPI4TORCH_AMD_PIPELINE_MB): the phased
K-exchange path of Alltoall/Alltoallv must be value- and adjoint-identical
to the single-exchange path. A tiny chunk size forces K=4 phases on CPU so
the gloo SPMD suite exercises the exact multi-rank phase geometry the GPU
runs (the op-layer slicing is transport-independent)."""

import os

import torch

from spmd import run_spmd


def _phased_alltoall_worker(rank, world):
    os.environ["MPI4TORCH_AMD_PIPELINE_MB"] = "0.0001"  # ~100 bytes
    import mpi4torch_amd as m

    comm = m.COMM_WORLD

    # different-axis, recv side marshaled (the EP/bench layout):
    # in [world*3, 7, 2] -> out [3, 7*world, 2]
    x = torch.arange(world * 3 * 7 * 2, dtype=torch.float64).reshape(
        world * 3, 7, 2) * (rank + 1)
    x.requires_grad_()
    y = comm.Alltoall(x, 1, 0, 3)
    assert y.shape == (3, 7 * world, 2)
    # round trip = identity (Alltoall∘Alltoall)
    z = comm.Alltoall(y, 0, 1, 7)
    assert (z.detach() == x.detach()).all()
    z.sum().backward()
    assert (x.grad == 1.0).all()

    # value check: block received from peer r along the gather axis is r's
    # slab (stamped by factor r+1)
    base = torch.arange(world * 3 * 7 * 2, dtype=torch.float64).reshape(
        world * 3, 7, 2)
    for r in range(world):
        want = base[rank * 3:(rank + 1) * 3] * (r + 1)
        got = y.detach()[:, r * 7:(r + 1) * 7]
        assert (got == want).all(), (rank, r)

    # different-axis, SEND side marshaled: gatheraxis 0, scatteraxis 1
    a = torch.randn(rank + 2, world * 4, 3, dtype=torch.float64)
    a.requires_grad_()
    b = comm.Alltoall(a, 0, 1, 4)
    total0 = sum(r + 2 for r in range(world))
    assert b.shape == (total0, 4, 3)
    c = comm.Alltoall(b, 1, 0, rank + 2)
    assert torch.allclose(c.detach(), a.detach())
    c.sum().backward()
    assert (a.grad == 1.0).all()

    # same-axis repartition with non-uniform counts and an inner axis
    # (before > 1 on both sides)
    n_old, n_new = rank + 1, world - rank
    s = torch.full((2, n_old, 3), float(rank), dtype=torch.float64)
    t = comm.Alltoall(s, 1, 1, n_new)
    assert t.shape == (2, n_new, 3)
    bounds, off = [], 0
    for r in range(world):
        bounds.append((off, off + r + 1))
        off += r + 1
    my_lo = sum(world - r for r in range(rank))
    for i in range(n_new):
        gpos = my_lo + i
        owner = next(r for r, (lo, hi) in enumerate(bounds)
                     if lo <= gpos < hi)
        assert (t[:, i] == owner).all()

    # Alltoallv (explicit counts) through the phased path, with autograd
    v = torch.randn(world * 2, 5, dtype=torch.float64).requires_grad_()
    w = comm.Alltoallv(v, 1, 0, [2] * world, [5] * world)
    ww = comm.Alltoallv(w, 0, 1, [5] * world, [2] * world)
    assert torch.allclose(ww.detach(), v.detach())
    ww.sum().backward()
    assert (v.grad == 1.0).all()


def _phased_funnel_worker(rank, world):
    os.environ["MPI4TORCH_AMD_PIPELINE_MB"] = "0.0001"  # force K=4
    import mpi4torch_amd as m

    comm = m.COMM_WORLD

    # Gather with before>1 and variable per-rank counts, fwd + adjoint
    my = rank + 1
    x = torch.full((3, my, 2), float(rank), dtype=torch.float64)
    x.requires_grad_()
    g = comm.Gather(x, 1, 0)
    total = world * (world + 1) // 2
    if rank == 0:
        assert g.shape == (3, total, 2)
        off = 0
        for r in range(world):
            assert (g[:, off:off + r + 1] == r).all()
            off += r + 1
    else:
        assert g.size(1) == 0
    # Scatter∘Gather identity through the phased paths
    back = comm.Scatter(g, 1, my, 0)
    assert (back.detach() == x.detach()).all()
    back.sum().backward()
    assert (x.grad == 1.0).all()

    # Allgather with before>1, non-uniform grads (reduce-scatter adjoint)
    y = torch.full((2, my, 3), float(rank + 1), dtype=torch.float64)
    y.requires_grad_()
    ag = comm.Allgather(y, 1)
    assert ag.shape == (2, total, 3)
    off = 0
    for r in range(world):
        assert (ag.detach()[:, off:off + r + 1] == r + 1).all()
        off += r + 1
    ag.backward(torch.full_like(ag, float(rank + 1)))
    want = sum(r + 1 for r in range(world))
    assert (y.grad == want).all()

    # equal counts too (the allgather_equal bypass must not kick in when
    # phased; values must agree regardless)
    z = torch.full((2, 4, 3), float(rank), dtype=torch.float64)
    agz = comm.Allgather(z, 1)
    for r in range(world):
        assert (agz[:, r * 4:(r + 1) * 4] == r).all()


def test_phased_funnel_ws2():
    run_spmd(2, _phased_funnel_worker)


def test_phased_funnel_ws5():
    run_spmd(5, _phased_funnel_worker)


def test_phased_funnel_ws7():
    run_spmd(7, _phased_funnel_worker)


def test_phased_alltoall_ws2():
    run_spmd(2, _phased_alltoall_worker)


def test_phased_alltoall_ws5():
    run_spmd(5, _phased_alltoall_worker)


def _phased_vs_plain_fuzz_worker(rank, world):
    # identical inputs through phased and unphased paths must agree exactly
    import mpi4torch_amd as m

    comm = m.COMM_WORLD
    torch.manual_seed(99)  # SAME tensor logic on all ranks modulo stamps
    cases = [
        ((world * 2, 3, 4), 1, 0, 2),
        ((2, world * 3, 5), 2, 1, 3),
        ((3, 4, world * 2), 0, 2, 2),
    ]
    for shape, ga, sa, ne in cases:
        x = torch.randn(shape, dtype=torch.float64) * (rank + 1)
        os.environ["MPI4TORCH_AMD_PIPELINE_MB"] = "0.0001"
        m._C.reload_config()
        y_phased = comm.Alltoall(x, ga, sa, ne)
        os.environ["MPI4TORCH_AMD_PIPELINE_MB"] = "64"
        m._C.reload_config()
        y_plain = comm.Alltoall(x, ga, sa, ne)
        assert (y_phased == y_plain).all(), (shape, ga, sa)


def test_phased_vs_plain_ws3():
    run_spmd(3, _phased_vs_plain_fuzz_worker)
