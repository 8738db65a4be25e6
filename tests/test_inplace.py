"""Bcast_/Reduce_: in-place semantics, adjoints, misuse guard.

Mirrors reference tests/test_collectives.py:23-46 (TestReduce/TestBcast),
including the negative test that reusing the input of an in-place
collective raises in backward (reference :30-36).
"""

import torch

from spmd import run_spmd


def _bcast_worker(rank, world):
    import mpi4torch_amd as m

    comm = m.COMM_WORLD
    t = torch.full((10,), float(rank), dtype=torch.double).requires_grad_()
    res = comm.Bcast_(t, 0)
    assert (res == 0.0).all()  # everyone has root's data
    res.sum().backward()
    # adjoint: grad accumulates at root, zeros elsewhere (reference :38-46)
    if rank == 0:
        assert (t.grad == world * torch.ones(10, dtype=torch.double)).all()
    else:
        assert (t.grad == torch.zeros(10, dtype=torch.double)).all()

    # broadcast from a non-zero root
    t2 = torch.full((4,), float(rank))
    res2 = comm.Bcast_(t2, world - 1)
    assert (res2 == world - 1).all()


def _reduce_worker(rank, world):
    import mpi4torch_amd as m

    comm = m.COMM_WORLD
    t = torch.rand(10, dtype=torch.double).requires_grad_()
    res = comm.Reduce_(t, m.MPI_SUM, 0)
    res.sum().backward()
    # adjoint of reduce-to-root is broadcast: grad = ones everywhere
    assert (t.grad == torch.ones(10, dtype=torch.double)).all()

    # forward value + non-root zeros
    v = torch.full((3,), float(rank + 1))
    r = comm.Reduce_(v, m.MPI_SUM, 0)
    if rank == 0:
        assert (r == world * (world + 1) / 2).all()
    else:
        assert (r == 0).all()


def _guard_worker(rank, world):
    import mpi4torch_amd as m

    comm = m.COMM_WORLD
    # reusing the INPUT of an in-place collective must raise in backward
    # (reference tests/test_collectives.py:30-36)
    tmp = 0.0 + torch.rand(10, dtype=torch.double).requires_grad_()
    res = tmp + comm.Reduce_(tmp, m.MPI_SUM, 0)
    try:
        res.sum().backward()
        raise AssertionError("expected RuntimeError from in-place misuse")
    except RuntimeError as e:
        assert "in-place" in str(e) or "inplace" in str(e).lower()


def test_bcast_ws2():
    run_spmd(2, _bcast_worker)


def test_bcast_ws5():
    run_spmd(5, _bcast_worker)


def test_reduce_ws2():
    run_spmd(2, _reduce_worker)


def test_reduce_ws5():
    run_spmd(5, _reduce_worker)


def test_inplace_guard_ws2():
    run_spmd(2, _guard_worker)
