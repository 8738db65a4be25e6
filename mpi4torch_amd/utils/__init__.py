from .timing import CollectiveTimer, algbw_gbps, busbw_gbps
from .tracing import trace, TracedCommunicator, TraceRecord

__all__ = [
    "CollectiveTimer",
    "algbw_gbps",
    "busbw_gbps",
    "trace",
    "TracedCommunicator",
    "TraceRecord",
]
