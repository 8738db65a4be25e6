"""hipEvent-based collective timing + bandwidth math.

The reference has no tracing/profiling subsystem; on MI355X
per-collective timing is first-class because the headline metric is
Allreduce algbw over xGMI.
"""

import torch


def algbw_gbps(nbytes: int, seconds: float) -> float:
    """Algorithmic bandwidth: bytes moved through the collective per second."""
    return nbytes / seconds / 1e9 if seconds > 0 else 0.0


def busbw_gbps(nbytes: int, seconds: float, world: int, op: str = "allreduce") -> float:
    """Bus bandwidth (NCCL convention): algbw corrected by the op's traffic
    factor — what the links actually carry. Ring allreduce moves
    2(P-1)/P bytes per byte of payload per GPU."""
    a = algbw_gbps(nbytes, seconds)
    if world <= 1:
        return a
    if op == "allreduce":
        return a * 2 * (world - 1) / world
    if op in ("allgather", "reducescatter", "alltoall"):
        return a * (world - 1) / world
    return a


class CollectiveTimer:
    """Times GPU regions with CUDA/HIP events (no host sync until query)."""

    def __init__(self):
        self._events = []  # (label, nbytes, start, end)

    def span(self, label: str, nbytes: int = 0):
        timer = self

        class _Span:
            def __enter__(self):
                self.s = torch.cuda.Event(enable_timing=True)
                self.e = torch.cuda.Event(enable_timing=True)
                self.s.record()
                return self

            def __exit__(self, *exc):
                self.e.record()
                timer._events.append((label, nbytes, self.s, self.e))

        return _Span()

    def results(self):
        """Synchronizes and returns [(label, ms, algbw_GBps)]."""
        torch.cuda.synchronize()
        out = []
        for label, nbytes, s, e in self._events:
            ms = s.elapsed_time(e)
            out.append((label, ms, algbw_gbps(nbytes, ms / 1e3)))
        return out

    def clear(self):
        self._events.clear()
