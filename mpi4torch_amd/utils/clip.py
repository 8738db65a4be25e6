"""Gradient clipping for sharded training.

With ZeRO-sharded gradients every rank holds a different slice, so the
global norm needs a cross-rank reduction before scaling — torch's
clip_grad_norm_ would silently clip by the LOCAL norm.
"""

from typing import Iterable

import torch

import mpi4torch_amd as m4a


def clip_grad_norm_sharded(tensors: Iterable[torch.Tensor],
                           max_norm: float, comm=None) -> torch.Tensor:
    """Clip gradients whose union is SHARDED across ranks by the true
    global L2 norm. `tensors` are this rank's gradient tensors (e.g. the
    .grad of ZeroRedundancyOptimizer/ShardedDataParallel shards, or of
    FSDP shard_parameters()). Returns the global norm."""
    comm = comm if comm is not None else m4a.COMM_WORLD
    grads = [t.grad if isinstance(t, torch.nn.Parameter) else t
             for t in tensors]
    grads = [g for g in grads if g is not None]
    if not grads:
        return torch.zeros(())
    local_sq = torch.stack([g.double().square().sum() for g in grads]).sum()
    total_sq = comm.Allreduce(local_sq.reshape(1), m4a.MPI_SUM)[0]
    total = total_sq.sqrt()
    scale = (max_norm / (total + 1e-6)).clamp(max=1.0)
    s = scale.to(grads[0].dtype)
    with torch.no_grad():
        for g in grads:
            g.mul_(s)
    return total
