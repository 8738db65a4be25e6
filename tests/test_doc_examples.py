"""The worked examples from doc/basic_usage.md, executed verbatim.

The 'wrong' Isend-Recv-Wait idiom must produce grad == 1 (the engine
prunes the unreferenced send chain — the documented hazard, reference
doc/basic_usage.rst ordering chapter) and the JoinDummies-wired version
must produce the correct grad == 2.
"""

import torch

from spmd import run_spmd


def _naive_worker(rank, world):
    import mpi4torch_amd as m4a

    comm = m4a.COMM_WORLD
    a = torch.tensor([1.0 + comm.rank]).requires_grad_()
    handle = comm.Isend(a, (comm.rank + 1) % comm.size, 0)
    b = comm.Recv(torch.empty_like(a),
                  (comm.rank - 1 + comm.size) % comm.size, 0)
    comm.Wait(handle)
    res = a + b
    assert res.item() == (1.0 + comm.rank) + (1.0 + (comm.rank - 1) % world)
    res.backward()
    # documented WRONG result: the send chain is pruned, grad misses the
    # neighbor's contribution
    assert a.grad.item() == 1.0, a.grad


def _fixed_worker(rank, world):
    import mpi4torch_amd as m4a

    comm = m4a.COMM_WORLD
    nxt = (comm.rank + 1) % comm.size
    prv = (comm.rank - 1 + comm.size) % comm.size
    a = torch.tensor([1.0 + comm.rank]).requires_grad_()
    handle = comm.Isend(a, nxt, 0)
    recvbuffer = m4a.JoinDummies(torch.empty_like(a), [handle.dummy])
    b = comm.Recv(recvbuffer, prv, 0)
    wait_ret = comm.Wait(m4a.JoinDummiesHandle(handle, [b]))
    res = m4a.JoinDummies(a + b, [wait_ret])
    res.backward()
    assert a.grad.item() == 2.0, a.grad


def _backward_chain_worker(rank, world):
    # the "backward-only deadlock" example, serialized with JoinDummies:
    # two allreduce chains ordered by an explicit dummy edge
    import mpi4torch_amd as m4a

    comm = m4a.COMM_WORLD
    x = torch.full((4,), float(rank + 1)).requires_grad_()
    y1 = comm.Allreduce(2 * x, m4a.MPI_SUM)
    y2 = comm.Allreduce(m4a.JoinDummies(3 * x, [y1]), m4a.MPI_SUM)
    (y1 + y2).sum().backward()
    # d/dx [sum over ranks of (2x + 3x) summed twice over world]
    assert (x.grad == 5.0 * world).all(), x.grad


def test_doc_naive_ring_ws2():
    run_spmd(2, _naive_worker)


def test_doc_fixed_ring_ws2():
    run_spmd(2, _fixed_worker)


def test_doc_fixed_ring_ws5():
    run_spmd(5, _fixed_worker)


def test_doc_backward_chain_ws3():
    run_spmd(3, _backward_chain_worker)
