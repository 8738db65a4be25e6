"""Per-collective tracing (hipEvent-based timing + payload bookkeeping).

The reference ships no observability; on MI355X, where the
headline metric is collective bandwidth over xGMI, per-op timing is a
first-class aux subsystem. Wrap a communicator:

    comm = trace(m4a.COMM_WORLD)
    ... run ...
    for rec in comm.trace_records():
        print(rec.op, rec.nbytes, rec.ms, rec.algbw_gbps)

Timing uses CUDA/HIP events on the current stream (no host sync until
trace_records() is called). Eager-mode only (the traced wrapper is not
TorchScript-scriptable; pass the underlying communicator to scripted code).
"""

from dataclasses import dataclass
from typing import List

import torch

from .timing import algbw_gbps


@dataclass
class TraceRecord:
    op: str
    nbytes: int
    ms: float
    algbw_gbps: float


class _Span:
    __slots__ = ("start", "end", "op", "nbytes", "_t0", "_t1")

    def __init__(self, op, nbytes):
        self.op = op
        self.nbytes = nbytes
        if torch.cuda.is_available():
            self.start = torch.cuda.Event(enable_timing=True)
            self.end = torch.cuda.Event(enable_timing=True)
        else:
            self.start = self.end = None

    def __enter__(self):
        if self.start is not None:
            self.start.record()
        else:
            import time

            self._t0 = time.perf_counter()
        return self

    def __exit__(self, *exc):
        if self.end is not None:
            self.end.record()
        else:
            import time

            self._t1 = time.perf_counter()

    def ms(self):
        if self.start is not None:
            return self.start.elapsed_time(self.end)
        return (self._t1 - self._t0) * 1e3


class TracedCommunicator:
    """Duck-typed communicator wrapper recording every collective."""

    def __init__(self, comm):
        self._comm = comm
        self._spans: List[_Span] = []

    @property
    def rank(self):
        return self._comm.rank

    @property
    def size(self):
        return self._comm.size

    def _record(self, op, tensor, fn):
        nbytes = tensor.numel() * tensor.element_size()
        span = _Span(op, nbytes)
        with span:
            out = fn()
        self._spans.append(span)
        return out

    def Allreduce(self, tensor, op):
        return self._record("Allreduce", tensor,
                            lambda: self._comm.Allreduce(tensor, op))

    def Bcast_(self, tensor, root):
        return self._record("Bcast_", tensor,
                            lambda: self._comm.Bcast_(tensor, root))

    def Reduce_(self, tensor, op, root):
        return self._record("Reduce_", tensor,
                            lambda: self._comm.Reduce_(tensor, op, root))

    def Gather(self, tensor, gatheraxis, root):
        return self._record("Gather", tensor,
                            lambda: self._comm.Gather(tensor, gatheraxis, root))

    def Allgather(self, tensor, gatheraxis):
        return self._record("Allgather", tensor,
                            lambda: self._comm.Allgather(tensor, gatheraxis))

    def Scatter(self, tensor, scatteraxis, numelem, root):
        return self._record(
            "Scatter", tensor,
            lambda: self._comm.Scatter(tensor, scatteraxis, numelem, root))

    def Alltoall(self, tensor, gatheraxis, scatteraxis, numelem):
        return self._record(
            "Alltoall", tensor,
            lambda: self._comm.Alltoall(tensor, gatheraxis, scatteraxis,
                                        numelem))

    def Reducescatter(self, tensor, axis, numelem):
        return self._record(
            "Reducescatter", tensor,
            lambda: self._comm.Reducescatter(tensor, axis, numelem))

    def Alltoallv(self, tensor, gatheraxis, scatteraxis, target_counts,
                  source_sizes):
        return self._record(
            "Alltoallv", tensor,
            lambda: self._comm.Alltoallv(tensor, gatheraxis, scatteraxis,
                                         target_counts, source_sizes))

    def AlltoallPairwise(self, tensor, axis, send_counts, recv_counts):
        return self._record(
            "AlltoallPairwise", tensor,
            lambda: self._comm.AlltoallPairwise(tensor, axis, send_counts,
                                                recv_counts))

    def Iallreduce(self, tensor, op):
        return self._record("Iallreduce", tensor,
                            lambda: self._comm.Iallreduce(tensor, op))

    def Ireducescatter(self, tensor, op):
        return self._record("Ireducescatter", tensor,
                            lambda: self._comm.Ireducescatter(tensor, op))

    def Iallgather(self, tensor):
        return self._record("Iallgather", tensor,
                            lambda: self._comm.Iallgather(tensor))

    def Barrier(self):
        return self._comm.Barrier()

    def Isend(self, tensor, dest, tag):
        return self._record("Isend", tensor,
                            lambda: self._comm.Isend(tensor, dest, tag))

    def Irecv(self, tensor, source, tag):
        return self._record("Irecv", tensor,
                            lambda: self._comm.Irecv(tensor, source, tag))

    def Wait(self, handle):
        return self._record("Wait", handle._handle[1],
                            lambda: self._comm.Wait(handle))

    def Send(self, tensor, dest, tag):
        return self._record("Send", tensor,
                            lambda: self._comm.Send(tensor, dest, tag))

    def Recv(self, tensor, source, tag):
        return self._record("Recv", tensor,
                            lambda: self._comm.Recv(tensor, source, tag))

    def trace_records(self) -> List[TraceRecord]:
        """Synchronizes the device and returns timing for every recorded op."""
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        out = []
        for s in self._spans:
            ms = s.ms()
            out.append(TraceRecord(s.op, s.nbytes, ms,
                                   algbw_gbps(s.nbytes, ms / 1e3)))
        return out

    def clear_trace(self):
        self._spans.clear()


def trace(comm) -> TracedCommunicator:
    """Wrap `comm` (an MPI_Communicator) with per-collective tracing."""
    return TracedCommunicator(comm)
