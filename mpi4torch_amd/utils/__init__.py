from .timing import CollectiveTimer, algbw_gbps, busbw_gbps
from .tracing import trace, TracedCommunicator, TraceRecord
from .checkpoint import (save_checkpoint, load_checkpoint,
                         save_sharded_checkpoint, load_sharded_checkpoint)
from .clip import clip_grad_norm_sharded

__all__ = [
    "CollectiveTimer",
    "algbw_gbps",
    "busbw_gbps",
    "trace",
    "TracedCommunicator",
    "TraceRecord",
    "save_checkpoint",
    "load_checkpoint",
    "save_sharded_checkpoint",
    "load_sharded_checkpoint",
    "clip_grad_norm_sharded",
]
