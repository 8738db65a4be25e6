"""TorchScript compatibility of the whole public surface.

The reference's distinguishing secondary capability: every
wrapper is @torch.jit.script-able, including functions taking the
communicator as an argument (reference tests/test_collectives.py:14-21).
"""

import torch

import mpi4torch_amd as m  # module-level so TorchScript resolves m.MPI_Communicator
from spmd import run_spmd


def _scripted_allreduce_worker(rank, world):
    comm = m.COMM_WORLD
    tmp = torch.rand(10, dtype=torch.double).requires_grad_()

    @torch.jit.script
    def myfunc(x: torch.Tensor, comm_: m.MPI_Communicator) -> torch.Tensor:
        return comm_.Allreduce(x, m.MPI_SUM)

    res = myfunc(tmp, comm)
    res.sum().backward()
    assert (tmp.grad == world * torch.ones(10, dtype=torch.double)).all()


def _scripted_ring_worker(rank, world):
    comm = m.COMM_WORLD

    @torch.jit.script
    def ring(x: torch.Tensor, comm_: m.MPI_Communicator, rank: int,
             world: int) -> torch.Tensor:
        req = comm_.Isend(x, (rank + 1) % world, 0)
        req2 = comm_.Irecv(
            m.JoinDummies(torch.empty_like(x), [req.dummy]),
            (rank + world - 1) % world,
            0,
        )
        res = comm_.Wait(m.JoinDummiesHandle(req, [req2.dummy]))
        res2 = comm_.Wait(m.JoinDummiesHandle(req2, [res]))
        return res2

    x = torch.full((100,), float(rank), dtype=torch.double).requires_grad_()
    got = ring(x, comm, rank, world)
    assert (got == (rank + world - 1) % world).all()
    (got * rank).sum().backward()
    assert (x.grad == ((rank + 1) % world) * torch.ones_like(x)).all()


def _scripted_axis_worker(rank, world):
    comm = m.COMM_WORLD

    @torch.jit.script
    def reshard(x: torch.Tensor, comm_: m.MPI_Communicator) -> torch.Tensor:
        g = comm_.Allgather(x, 0)
        return comm_.Alltoall(g, 1, 0, x.size(0))

    x = torch.rand(2 * world, 3, dtype=torch.double)
    out = reshard(x, comm)
    assert out.shape[0] == 2 * world and out.shape[1] == 3 * world


def _scripted_extensions_worker(rank, world):
    comm = m.COMM_WORLD

    @torch.jit.script
    def ep_roundtrip(x: torch.Tensor, comm_: m.MPI_Communicator,
                     world: int) -> torch.Tensor:
        counts = [2 for _ in range(world)]
        sizes = [5 for _ in range(world)]
        y = comm_.Alltoallv(x, 1, 0, counts, sizes)
        return comm_.Alltoallv(y, 0, 1, sizes, counts)

    @torch.jit.script
    def pairwise(x: torch.Tensor, comm_: m.MPI_Communicator,
                 world: int) -> torch.Tensor:
        sc = [1 for _ in range(world)]
        empty: list[int] = []
        return comm_.AlltoallPairwise(x, 0, sc, empty)

    @torch.jit.script
    def overlap(x: torch.Tensor, comm_: m.MPI_Communicator) -> torch.Tensor:
        h = comm_.Iallgather(x)
        return comm_.Wait(h)

    @torch.jit.script
    def minloc(x: torch.Tensor, comm_: m.MPI_Communicator) -> torch.Tensor:
        return comm_.Allreduce(x, m.MPI_MINLOC)

    x = torch.rand(world * 2, 5, dtype=torch.double).requires_grad_()
    rt = ep_roundtrip(x, comm, world)
    assert (rt.detach() == x.detach()).all()
    rt.sum().backward()
    assert (x.grad == 1.0).all()

    p = torch.full((world, 3), float(rank), dtype=torch.double)
    got = pairwise(p, comm, world)
    for r in range(world):
        assert (got[r] == r).all()

    g = overlap(torch.full((4,), float(rank + 1)), comm)
    assert g.numel() == 4 * world

    pairs = torch.stack([torch.full((3,), float(rank)),
                         torch.full((3,), float(rank * 7))], dim=-1)
    mn = minloc(pairs, comm)
    assert (mn[:, 0] == 0).all() and (mn[:, 1] == 0).all()


def test_scripted_extensions_ws3():
    run_spmd(3, _scripted_extensions_worker)


def test_scripted_allreduce_ws2():
    run_spmd(2, _scripted_allreduce_worker)


def test_scripted_ring_ws2():
    run_spmd(2, _scripted_ring_worker)


def test_scripted_axis_ws2():
    run_spmd(2, _scripted_axis_worker)


def test_scripted_local():
    # scripting + execution without a distributed context
    comm = m.COMM_WORLD

    @torch.jit.script
    def f(x: torch.Tensor, comm_: m.MPI_Communicator) -> torch.Tensor:
        return comm_.Allreduce(x, m.MPI_SUM)

    t = torch.rand(4).requires_grad_()
    out = f(t, comm)
    out.sum().backward()
    assert (t.grad == torch.ones(4)).all()
