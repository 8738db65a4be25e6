"""Functional wrappers over the COMM_WORLD communicator.

Thin conveniences for code that does not want to thread a communicator
object through call sites; every function accepts an explicit ``comm=``.
"""

from typing import List, Optional

import torch

import mpi4torch_amd as _m


def _comm(comm):
    return comm if comm is not None else _m.COMM_WORLD


def allreduce(tensor: torch.Tensor, op: Optional[int] = None, comm=None):
    return _comm(comm).Allreduce(tensor, _m.MPI_SUM if op is None else op)


def bcast_(tensor: torch.Tensor, root: int = 0, comm=None):
    return _comm(comm).Bcast_(tensor, root)


def reduce_(tensor: torch.Tensor, op: Optional[int] = None, root: int = 0,
            comm=None):
    return _comm(comm).Reduce_(tensor, _m.MPI_SUM if op is None else op, root)


def gather(tensor: torch.Tensor, axis: int = 0, root: int = 0, comm=None):
    return _comm(comm).Gather(tensor, axis, root)


def allgather(tensor: torch.Tensor, axis: int = 0, comm=None):
    return _comm(comm).Allgather(tensor, axis)


def scatter(tensor: torch.Tensor, axis: int = 0, numelem: int = 1,
            root: int = 0, comm=None):
    return _comm(comm).Scatter(tensor, axis, numelem, root)


def alltoall(tensor: torch.Tensor, gatheraxis: int, scatteraxis: int,
             numelem: int, comm=None):
    return _comm(comm).Alltoall(tensor, gatheraxis, scatteraxis, numelem)


def reducescatter(tensor: torch.Tensor, axis: int = 0, numelem: int = 1,
                  comm=None):
    return _comm(comm).Reducescatter(tensor, axis, numelem)


def alltoallv(tensor: torch.Tensor, gatheraxis: int, scatteraxis: int,
              target_counts, source_sizes, comm=None):
    return _comm(comm).Alltoallv(tensor, gatheraxis, scatteraxis,
                                 list(target_counts), list(source_sizes))


def alltoall_pairwise(tensor: torch.Tensor, axis: int, send_counts,
                      recv_counts=None, comm=None):
    return _comm(comm).AlltoallPairwise(
        tensor, axis, list(send_counts),
        [] if recv_counts is None else list(recv_counts))


def iallreduce(tensor: torch.Tensor, op: Optional[int] = None, comm=None):
    return _comm(comm).Iallreduce(tensor, _m.MPI_SUM if op is None else op)


def ireducescatter(tensor: torch.Tensor, op: Optional[int] = None,
                   comm=None):
    return _comm(comm).Ireducescatter(
        tensor, _m.MPI_SUM if op is None else op)


def iallgather(tensor: torch.Tensor, comm=None):
    return _comm(comm).Iallgather(tensor)


def isend(tensor: torch.Tensor, dest: int, tag: int = 0, comm=None):
    return _comm(comm).Isend(tensor, dest, tag)


def irecv(tensor: torch.Tensor, source: int, tag: int = 0, comm=None):
    return _comm(comm).Irecv(tensor, source, tag)


def wait(handle, comm=None):
    return _comm(comm).Wait(handle)


def send(tensor: torch.Tensor, dest: int, tag: int = 0, comm=None):
    return _comm(comm).Send(tensor, dest, tag)


def recv(tensor: torch.Tensor, source: int, tag: int = 0, comm=None):
    return _comm(comm).Recv(tensor, source, tag)


def join_dummies(loopthrough: torch.Tensor, dummies: List[torch.Tensor]):
    return _m.JoinDummies(loopthrough, dummies)


def barrier(comm=None):
    """Synchronize all ranks (not in the reference API; implemented as a
    1-element allreduce, which is a barrier on both transports)."""
    import torch as _t

    _comm(comm).Allreduce(_t.zeros(1), _m.MPI_SUM)
