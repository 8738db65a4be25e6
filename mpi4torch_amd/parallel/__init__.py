"""Parallelism strategies as thin layers over the primitives.

The reference is a primitive library; its docs demonstrate data parallelism
(doc/examples.rst) and its axis-aware Alltoall IS the Ulysses sequence<->
head reshard primitive. These modules make those patterns
reusable, tuned for one-process-per-GPU RCCL over xGMI.
"""

from .ddp import DistributedDataParallel
from .ulysses import ulysses_reshard, seq_to_head, head_to_seq
from .tp import (ColumnParallelLinear, RowParallelLinear, TensorParallelMLP,
                 copy_to_tp_region)
from .pipeline import GPipe
from .zero import ZeroRedundancyOptimizer
from .sharded_ddp import ShardedDataParallel
from .fsdp import FullyShardedDataParallel

__all__ = [
    "DistributedDataParallel",
    "ulysses_reshard",
    "seq_to_head",
    "head_to_seq",
    "ColumnParallelLinear",
    "RowParallelLinear",
    "TensorParallelMLP",
    "copy_to_tp_region",
    "GPipe",
    "ZeroRedundancyOptimizer",
    "ShardedDataParallel",
    "FullyShardedDataParallel",
]
