"""Collective-desync detector (MPI4TORCH_AMD_DEBUG=1): mismatched
collective sequences raise immediately instead of deadlocking."""

import os

import torch

from spmd import run_spmd


def _desync_worker(rank, world):
    os.environ["MPI4TORCH_AMD_DEBUG"] = "1"
    import mpi4torch_amd as m

    comm = m.COMM_WORLD
    # matched collective passes
    r = comm.Allreduce(torch.ones(4), m.MPI_SUM)
    assert (r == world).all()
    # mismatched shape across ranks must raise on every rank
    t = torch.ones(4 if rank == 0 else 5)
    try:
        comm.Allreduce(t, m.MPI_SUM)
        raise AssertionError("expected desync detection to fire")
    except RuntimeError as e:
        assert "desync" in str(e)


def _debug_off_worker(rank, world):
    os.environ.pop("MPI4TORCH_AMD_DEBUG", None)
    import mpi4torch_amd as m

    comm = m.COMM_WORLD
    r = comm.Allreduce(torch.ones(8), m.MPI_SUM)
    assert (r == world).all()


def test_desync_detector_ws2():
    run_spmd(2, _desync_worker)


def test_debug_off_ws2():
    run_spmd(2, _debug_off_worker)


def _p2p_tag_cross_worker(rank, world):
    os.environ["MPI4TORCH_AMD_DEBUG"] = "1"
    import mpi4torch_amd as m

    comm = m.COMM_WORLD
    # rank 0 posts tags (1, 2); rank 1 recvs in order (2, 1). Matching is
    # FIFO per (peer, channel), so the pairs cross — the debug handshake
    # must turn the silent payload swap into an error at Wait.
    if rank == 0:
        h1 = comm.Isend(torch.ones(8), 1, 1)
        h2 = comm.Isend(torch.ones(8) * 2, 1, 2)
        comm.Wait(h1)
        comm.Wait(h2)
    else:
        r1 = comm.Irecv(torch.empty(8), 0, 2)
        r2 = comm.Irecv(torch.empty(8), 0, 1)
        try:
            comm.Wait(r1)
            raise AssertionError("expected p2p tag-mismatch detection")
        except RuntimeError as e:
            assert "FIFO" in str(e) and "tag=1" in str(e), e
        try:
            comm.Wait(r2)  # the second pair is crossed too
            raise AssertionError("expected p2p tag-mismatch detection")
        except RuntimeError as e:
            assert "tag=2" in str(e), e


def test_p2p_tag_cross_ws2():
    run_spmd(2, _p2p_tag_cross_worker)


def _p2p_dtype_mismatch_worker(rank, world):
    os.environ["MPI4TORCH_AMD_DEBUG"] = "1"
    import mpi4torch_amd as m

    comm = m.COMM_WORLD
    # same byte count, different dtype: the wire moves bytes happily; only
    # the handshake can catch it
    if rank == 0:
        h = comm.Isend(torch.ones(8, dtype=torch.float32), 1, 0)
        comm.Wait(h)
    else:
        r = comm.Irecv(torch.empty(8, dtype=torch.int32), 0, 0)
        try:
            comm.Wait(r)
            raise AssertionError("expected p2p dtype-mismatch detection")
        except RuntimeError as e:
            assert "dtype" in str(e), e


def test_p2p_dtype_mismatch_ws2():
    run_spmd(2, _p2p_dtype_mismatch_worker)


def _p2p_matched_with_debug_worker(rank, world):
    os.environ["MPI4TORCH_AMD_DEBUG"] = "1"
    import mpi4torch_amd as m

    comm = m.COMM_WORLD
    # a correct ring still works with the handshake on
    src = torch.full((64,), float(rank))
    h = comm.Isend(src, (rank + 1) % world, 7)
    got = comm.Recv(
        m.JoinDummies(torch.empty(64), [h.dummy]), (rank + world - 1) % world, 7
    )
    comm.Wait(m.JoinDummiesHandle(h, [got]))
    assert (got == (rank + world - 1) % world).all()


def test_p2p_matched_with_debug_ws5():
    run_spmd(5, _p2p_matched_with_debug_worker)


def _timeout_worker(rank, world):
    import mpi4torch_amd as m

    # bound waits so a desynced peer raises instead of hanging
    m._C  # extension loaded; env read at first config() use in this process
    comm = m.COMM_WORLD
    if rank == 0:
        t = torch.ones(4)
        try:
            comm.Allreduce(t, m.MPI_SUM)  # rank 1 never joins
            raise AssertionError("expected timeout")
        except RuntimeError as e:
            assert "timed out" in str(e), e
    else:
        import time

        time.sleep(8)  # never issue the collective


def test_timeout_ws2():
    import os

    os.environ["MPI4TORCH_AMD_TIMEOUT_S"] = "3"
    try:
        run_spmd(2, _timeout_worker)
    finally:
        os.environ.pop("MPI4TORCH_AMD_TIMEOUT_S", None)


def _all_modes_worker(rank, world):
    # every debug/lowering mode stacked: desync detector + p2p handshake
    # + forced 4-phase pipelining + forced hierarchical lowering — the
    # interaction matrix (detector host traffic interleaves phased gloo
    # block traffic and handshake metadata)
    os.environ["MPI4TORCH_AMD_DEBUG"] = "1"
    os.environ["MPI4TORCH_AMD_PIPELINE_MB"] = "0.0001"
    os.environ["MPI4TORCH_AMD_FORCE_HIERARCHICAL"] = "1"
    import mpi4torch_amd as m

    comm = m.COMM_WORLD
    x = torch.full((3, rank + 1, 2), float(rank), dtype=torch.float64)
    g = comm.Gather(x, 1, 0)
    back = comm.Scatter(g, 1, rank + 1, 0)
    assert (back == x).all()
    comm.Allgather(x, 1)
    ti = torch.arange(17, dtype=torch.int64) * (rank + 3)
    comm.Allreduce(ti, m.MPI_BXOR)
    t8 = (torch.randn(33) * 0.2).to(torch.float8_e4m3fn)
    comm.Allreduce(t8, m.MPI_SUM)
    pairs = torch.stack([torch.randn(9), torch.full((9,), float(rank))], -1)
    comm.Allreduce(pairs, m.MPI_MAXLOC)
    prev = (rank - 1 + world) % world
    h = comm.Isend(x, (rank + 1) % world, 3)
    buf = torch.empty(3, prev + 1, 2, dtype=torch.float64)
    got = comm.Recv(m.JoinDummies(buf, [h.dummy]), prev, 3)
    comm.Wait(m.JoinDummiesHandle(h, [got]))
    assert (got == prev).all()


def test_all_modes_stacked_ws5():
    run_spmd(5, _all_modes_worker)
