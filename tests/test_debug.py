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
