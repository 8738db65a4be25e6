"""DeepSpeed-Ulysses-style sequence<->head resharding on Alltoall.

The reference's axis-aware Alltoall IS the Ulysses reshard primitive
(reference csrc/extension.cpp:917-987): moving a
[batch, seq/P, heads, dim] activation to [batch, seq, heads/P, dim] and
back is Alltoall(gatheraxis=seq, scatteraxis=heads, numelem=heads/P).
Because Alltoall is autograd-transparent, the backward reshards the
gradient the opposite way automatically — attention code using these
helpers needs no custom autograd.
"""

import torch

import mpi4torch_amd as m4a


def ulysses_reshard(t: torch.Tensor, gather_axis: int, scatter_axis: int,
                    numelem: int, comm=None) -> torch.Tensor:
    """General reshard: gather `gather_axis` across ranks, scatter
    `scatter_axis`, keeping `numelem` slices of it locally."""
    comm = comm if comm is not None else m4a.COMM_WORLD
    return comm.Alltoall(t, gather_axis, scatter_axis, numelem)


def seq_to_head(t: torch.Tensor, comm=None, seq_axis: int = 1,
                head_axis: int = 2) -> torch.Tensor:
    """[b, s/P, h, d] -> [b, s, h/P, d]: full sequence, sharded heads —
    the layout attention wants. Uses the explicit-counts Alltoallv (all
    counts are uniform and locally known), skipping the host exchanges."""
    comm = comm if comm is not None else m4a.COMM_WORLD
    heads = t.size(head_axis)
    assert heads % comm.size == 0, (
        f"head count {heads} must divide by world size {comm.size}")
    if comm.size == 1:
        return comm.Alltoall(t, seq_axis, head_axis, heads)
    target = [heads // comm.size] * comm.size
    source = [t.size(seq_axis)] * comm.size
    return comm.Alltoallv(t, seq_axis, head_axis, target, source)


def head_to_seq(t: torch.Tensor, comm=None, seq_axis: int = 1,
                head_axis: int = 2) -> torch.Tensor:
    """[b, s, h/P, d] -> [b, s/P, h, d]: back to sequence sharding."""
    comm = comm if comm is not None else m4a.COMM_WORLD
    seq = t.size(seq_axis)
    assert seq % comm.size == 0, (
        f"sequence length {seq} must divide by world size {comm.size}")
    if comm.size == 1:
        return comm.Alltoall(t, head_axis, seq_axis, seq)
    target = [seq // comm.size] * comm.size
    source = [t.size(head_axis)] * comm.size
    return comm.Alltoallv(t, head_axis, seq_axis, target, source)
