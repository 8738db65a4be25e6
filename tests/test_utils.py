"""utils/: tracing wrapper and bandwidth math (aux subsystem; the
reference has none — SURVEY.md §5)."""

import torch

from spmd import run_spmd


def _trace_worker(rank, world):
    import mpi4torch_amd as m
    from mpi4torch_amd.utils import trace

    comm = trace(m.COMM_WORLD)
    t = torch.ones(1000)
    comm.Allreduce(t, m.MPI_SUM)
    comm.Allgather(t, 0)
    recs = comm.trace_records()
    assert [r.op for r in recs] == ["Allreduce", "Allgather"]
    assert all(r.ms >= 0 for r in recs)
    assert recs[0].nbytes == 4000
    comm.clear_trace()
    assert comm.trace_records() == []


def test_trace_ws2():
    run_spmd(2, _trace_worker)


def test_busbw_math():
    from mpi4torch_amd.utils import busbw_gbps, algbw_gbps

    assert algbw_gbps(1e9, 1.0) == 1.0
    assert abs(busbw_gbps(1e9, 1.0, 8, "allreduce") - 2 * 7 / 8) < 1e-9
    assert busbw_gbps(1e9, 1.0, 1) == 1.0
