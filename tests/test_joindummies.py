"""JoinDummies: dummies get zero grads, loopthrough grad unaffected.

Mirrors reference tests/test_joindummies.py.
"""

import torch

from spmd import run_spmd


def _worker(rank, world):
    import mpi4torch_amd as m

    comm = m.COMM_WORLD
    tmp = torch.rand(10, dtype=torch.double).requires_grad_()
    tmp2 = torch.rand(10, dtype=torch.double).requires_grad_()
    tmp3 = torch.rand(7, dtype=torch.double).requires_grad_()
    res = comm.Allreduce(tmp, m.MPI_SUM)
    res2 = m.JoinDummies(res, [tmp2, tmp3])
    res2.sum().backward()
    assert (tmp2.grad == torch.zeros(10, dtype=torch.double)).all()
    assert (tmp3.grad == torch.zeros(7, dtype=torch.double)).all()
    assert (tmp.grad == world * torch.ones(10, dtype=torch.double)).all()

    # passthrough when no dummy requires grad (reference :1030-1033)
    a = torch.rand(4).requires_grad_()
    b = torch.rand(4)  # no grad
    out = m.JoinDummies(a, [b])
    out.sum().backward()
    assert (a.grad == torch.ones(4)).all()


def test_joindummies_ws2():
    run_spmd(2, _worker)


def test_joindummies_local():
    # also valid without any distributed context (world of one)
    import mpi4torch_amd as m

    t = torch.rand(5).requires_grad_()
    d = torch.rand(3).requires_grad_()
    out = m.JoinDummies(t, [d])
    out.sum().backward()
    assert (d.grad == torch.zeros(3)).all()
    assert (t.grad == torch.ones(5)).all()
