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
