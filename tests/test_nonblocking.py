"""Isend/Irecv/Wait ring exchanges with autograd through WaitHandle.

Mirrors reference tests/test_nonblocking.py: three orderings of the ring
exchange, each verifying that the gradient is routed through the REVERSED
ring: grad(tmp) == ((rank+1) % size) * ones. Tensor sizes large enough to
force real async progress on the 10M-element case (reference :9).
"""

import torch

from spmd import run_spmd


def _isend_irecv_worker(rank, world, n):
    import mpi4torch_amd as m

    comm = m.COMM_WORLD
    tmp = torch.rand(n, dtype=torch.double).requires_grad_()
    req = comm.Isend(tmp, (rank + 1) % world, 0)
    req2 = comm.Irecv(
        m.JoinDummies(torch.empty_like(tmp), [req.dummy]),
        (rank + world - 1) % world,
        0,
    )
    res = comm.Wait(m.JoinDummiesHandle(req, [req2.dummy]))
    res2 = comm.Wait(m.JoinDummiesHandle(req2, [res]))
    res3 = res2 * rank
    res3.sum().backward()
    assert (tmp.grad == ((rank + 1) % world) * torch.ones_like(tmp)).all()


def _isend_recv_worker(rank, world, n):
    import mpi4torch_amd as m

    comm = m.COMM_WORLD
    tmp = torch.rand(n, dtype=torch.double).requires_grad_()
    req = comm.Isend(tmp, (rank + 1) % world, 0)
    res = comm.Recv(
        m.JoinDummies(torch.empty_like(tmp), [req.dummy]),
        (rank + world - 1) % world,
        0,
    )
    res2 = comm.Wait(m.JoinDummiesHandle(req, [res]))
    res3 = m.JoinDummies(res, [res2]) * rank
    res3.sum().backward()
    assert (tmp.grad == ((rank + 1) % world) * torch.ones_like(tmp)).all()


def _irecv_send_worker(rank, world, n):
    import mpi4torch_amd as m

    comm = m.COMM_WORLD
    tmp = torch.rand(n, dtype=torch.double).requires_grad_()
    req = comm.Irecv(
        m.JoinDummies(torch.empty_like(tmp), [tmp]),
        (rank + world - 1) % world,
        0,
    )
    res = comm.Send(tmp, (rank + 1) % world, 0)
    res2 = comm.Wait(m.JoinDummiesHandle(req, [res]))
    res3 = res2 * rank
    res3.sum().backward()
    assert (tmp.grad == ((rank + 1) % world) * torch.ones_like(tmp)).all()


def _forward_value_worker(rank, world, n):
    import mpi4torch_amd as m

    comm = m.COMM_WORLD
    # value check: receive predecessor's rank-stamped tensor
    src = torch.full((n,), float(rank))
    req = comm.Isend(src, (rank + 1) % world, 3)
    got = comm.Recv(
        m.JoinDummies(torch.empty(n), [req.dummy]), (rank + world - 1) % world, 3
    )
    comm.Wait(m.JoinDummiesHandle(req, [got]))
    assert (got == (rank + world - 1) % world).all()


def _bifurcation_worker(rank, world):
    import mpi4torch_amd as m

    comm = m.COMM_WORLD
    # consuming the handle buffer outside Wait must be detected
    # (reference csrc/extension.cpp:1231-1237)
    t = torch.rand(10).requires_grad_()
    req = comm.Isend(t, rank, 5)  # self send (never completed)
    broken = m.WaitHandle([req._handle[0], req._handle[1] + 0.0, req._handle[2]])
    try:
        comm.Wait(broken)
        raise AssertionError("expected bifurcation detection to fire")
    except RuntimeError as e:
        assert "bifurcation" in str(e)
    # complete the dangling self pair so the request table drains
    req2 = comm.Irecv(torch.empty(10), rank, 5)
    comm.Wait(req2)
    comm.Wait(req)


def test_isend_irecv_small_ws5():
    run_spmd(5, _isend_irecv_worker, 10_000)


def test_isend_irecv_10m_ws2():
    # 10M doubles: forces true async progress (reference :9)
    run_spmd(2, _isend_irecv_worker, 10_000_000)


def test_isend_recv_ws5():
    run_spmd(5, _isend_recv_worker, 10_000)


def test_irecv_send_ws5():
    run_spmd(5, _irecv_send_worker, 10_000)


def test_forward_values_ws5():
    run_spmd(5, _forward_value_worker, 1000)


def test_bifurcation_detection_ws2():
    run_spmd(2, _bifurcation_worker)


def _channel_independence_worker(rank, world):
    import mpi4torch_amd as m

    comm = m.COMM_WORLD
    # p2p and collectives are posted in DIFFERENT orders on the two ranks —
    # legal under MPI semantics because a posted Isend does not block the
    # sender. The dedicated p2p channel (transport.hpp) preserves this;
    # a single-ordered transport would deadlock here.
    x = torch.full((1000,), float(rank))
    y = torch.ones(10)
    if rank == 0:
        h = comm.Isend(x, 1, 9)
        ar = comm.Allreduce(y, m.MPI_SUM)
        comm.Wait(h)
    else:
        ar = comm.Allreduce(y, m.MPI_SUM)
        if rank == 1:
            got = comm.Recv(torch.empty(1000), 0, 9)
            assert (got == 0.0).all()
    assert (ar == world).all()


def test_channel_independence_ws2():
    run_spmd(2, _channel_independence_worker)


def test_channel_independence_ws5():
    run_spmd(5, _channel_independence_worker)
