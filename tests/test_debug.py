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
