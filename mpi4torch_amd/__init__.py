"""mpi4torch_amd — MI355X-native autodiff-transparent collectives for PyTorch.

A from-scratch re-design of the capability set of helmholtz-analytics/
mpi4torch (reference: /root/reference, src/__init__.py) for AMD MI355X
(gfx950) nodes:

* transport is RCCL over xGMI (one process per GPU, torchrun launch) with a
  c10d/gloo path for CPU tensors — no MPI, no mpirun, no CUDA-aware shims;
* every collective is autograd-transparent: its backward issues the adjoint
  collective (Allreduce<->Allreduce, Bcast<->Reduce, Gather<->Scatter,
  Allgather->ReduceScatter, Alltoall<->Alltoall, Send<->Recv);
* axis-aware variable-count collectives marshal through hand-written CDNA4
  HIP pack/unpack kernels instead of MPI derived datatypes;
* the whole Python surface is TorchScript-scriptable, like the reference
  (reference src/__init__.py:27-240).

Public API parity (reference src/__init__.py:5-25): the 12 MPI_* reduction
constants, WaitHandle, JoinDummies, JoinDummiesHandle, MPI_Communicator,
COMM_WORLD, plus torch.distributed interop (comm_from_process_group, the
analog of comm_from_mpi4py) and the host-staging debug toggle (the analog of
deactivate_cuda_aware_mpi_support).
"""

import os
from typing import List, Optional

import torch

__version__ = "0.2.0"

from . import _C  # registers torch.classes.mpi4torch_amd.* and torch.ops.mpi4torch_amd.*

MPI_MAX: int = int(_C.MPI_MAX)
MPI_MIN: int = int(_C.MPI_MIN)
MPI_SUM: int = int(_C.MPI_SUM)
MPI_PROD: int = int(_C.MPI_PROD)
MPI_LAND: int = int(_C.MPI_LAND)
MPI_BAND: int = int(_C.MPI_BAND)
MPI_LOR: int = int(_C.MPI_LOR)
MPI_BOR: int = int(_C.MPI_BOR)
MPI_LXOR: int = int(_C.MPI_LXOR)
MPI_BXOR: int = int(_C.MPI_BXOR)
MPI_MINLOC: int = int(_C.MPI_MINLOC)
MPI_MAXLOC: int = int(_C.MPI_MAXLOC)

__all__ = [
    "MPI_MAX",
    "MPI_MIN",
    "MPI_SUM",
    "MPI_PROD",
    "MPI_LAND",
    "MPI_BAND",
    "MPI_LOR",
    "MPI_BOR",
    "MPI_LXOR",
    "MPI_BXOR",
    "MPI_MINLOC",
    "MPI_MAXLOC",
    "WaitHandle",
    "JoinDummies",
    "JoinDummiesHandle",
    "MPI_Communicator",
    "Communicator",
    "COMM_WORLD",
    "init",
    "comm_from_process_group",
    "comm_split",
    "comm_from_mpi4py",
    "force_host_staging",
    "deactivate_cuda_aware_mpi_support",
]


@torch.jit.script
class WaitHandle:
    """Wait handle returned by the non-blocking ops (Isend/Irecv).

    Same 3-slot contract as the reference (src/__init__.py:27-40,
    csrc/extension.cpp:1094-1107): [metadata, live comm buffer, input].
    """

    def __init__(self, raw_handle: List[torch.Tensor]):
        self._handle = raw_handle

    @property
    def dummy(self) -> torch.Tensor:
        """Dummy tensor usable as a JoinDummies dependency."""
        return self._handle[0]


@torch.jit.script
def JoinDummies(loopthrough: torch.Tensor, dummies: List[torch.Tensor]) -> torch.Tensor:
    """Join dummy dependencies into the autograd DAG.

    Forward: identity on ``loopthrough``. Backward: the ``dummies`` receive
    zero gradients, but the AD engine treats them as real dependencies —
    the mechanism for hand-encoding cross-rank ordering (reference
    src/__init__.py:42-67, doc/basic_usage.rst:314-457).
    """
    return torch.ops.mpi4torch_amd.JoinDummies(loopthrough, dummies)


@torch.jit.script
def JoinDummiesHandle(handle: WaitHandle, dummies: List[torch.Tensor]) -> WaitHandle:
    """JoinDummies for a WaitHandle (reference src/__init__.py:69-87)."""
    raw = handle._handle
    return WaitHandle([
        torch.ops.mpi4torch_amd.JoinDummies(raw[0], dummies),
        raw[1],
        raw[2],
    ])


@torch.jit.script
class MPI_Communicator:
    """Communicator facade (reference src/__init__.py:89-240).

    Methods with an underscore suffix are in-place. All communication is
    autograd-transparent; see the class docstrings of the native layer
    (csrc/ops.cpp) for the adjoint of each op.
    """

    def __init__(self, comm: torch.classes.mpi4torch_amd.Communicator):
        self._comm = comm

    @property
    def rank(self) -> int:
        """Rank of the local process within this communicator."""
        return self._comm.GetRank()

    @property
    def size(self) -> int:
        """Number of processes in this communicator."""
        return self._comm.GetSize()

    @property
    def group_name(self) -> str:
        """The c10d group this communicator is bound to ('' = local)."""
        return self._comm.GetGroupName()

    def Barrier(self) -> None:
        """Synchronize all ranks (host-level; not in the reference API)."""
        # literal 2 == MPI_SUM (module globals are not visible inside a
        # TorchScript class body)
        self._comm.Allreduce(torch.zeros(1), 2)

    def Allreduce(self, tensor: torch.Tensor, op: int) -> torch.Tensor:
        """Elementwise combine across all ranks; result on every rank.

        Backward (MPI_SUM only): Allreduce of the gradient — self-adjoint.
        """
        return self._comm.Allreduce(tensor, op)

    def Bcast_(self, tensor: torch.Tensor, root: int) -> torch.Tensor:
        """Broadcast from ``root`` (in place). Backward: Reduce_ to root."""
        return self._comm.Bcast_(tensor, root)

    def Reduce_(self, tensor: torch.Tensor, op: int, root: int) -> torch.Tensor:
        """Reduce to ``root`` (in place; non-root result is zeros).

        Backward (MPI_SUM only): Bcast_ from root.
        """
        return self._comm.Reduce_(tensor, op, root)

    def Gather(self, tensor: torch.Tensor, gatheraxis: int, root: int) -> torch.Tensor:
        """Concatenate per-rank tensors along ``gatheraxis`` at ``root``.

        Non-root ranks receive an empty tensor (axis size 0). Per-rank axis
        sizes may differ. Backward: Scatter.
        """
        return self._comm.Gather(tensor, gatheraxis, root)

    def Allgather(self, tensor: torch.Tensor, gatheraxis: int) -> torch.Tensor:
        """Concatenate per-rank tensors along ``gatheraxis`` on all ranks.

        Backward: reduce-scatter of the gradient (the mathematically correct
        adjoint; the reference's composite had a latent wrong-root bug,
        csrc/extension.cpp:626-628).
        """
        return self._comm.Allgather(tensor, gatheraxis)

    def Reducescatter(self, tensor: torch.Tensor, axis: int,
                      numelem: int) -> torch.Tensor:
        """Elementwise-SUM across ranks, keeping ``numelem`` slices of
        ``axis`` locally (MI355X extension; the exact adjoint pair of
        Allgather — its backward IS Allgather). The ZeRO gradient-sharding
        primitive."""
        return self._comm.Reducescatter(tensor, axis, numelem)

    def Scatter(self, tensor: torch.Tensor, scatteraxis: int, numelem: int,
                root: int) -> torch.Tensor:
        """Distribute ``root``'s tensor along ``scatteraxis``; this rank
        receives ``numelem`` slices. Non-root input tensors are ignored
        (pass any placeholder). Backward: Gather.
        """
        return self._comm.Scatter(tensor, scatteraxis, numelem, root)

    def Alltoall(self, tensor: torch.Tensor, gatheraxis: int, scatteraxis: int,
                 numelem: int) -> torch.Tensor:
        """Equivalent to Scatter(Gather(tensor, gatheraxis, 0), scatteraxis,
        numelem, 0) but implemented as one grouped RCCL exchange with fused
        pack/unpack. Supports gatheraxis == scatteraxis (repartition) with
        per-rank variable counts. Backward: Alltoall with axes swapped.
        """
        return self._comm.Alltoall(tensor, gatheraxis, scatteraxis, numelem)

    def Alltoallv(self, tensor: torch.Tensor, gatheraxis: int,
                  scatteraxis: int, target_counts: List[int],
                  source_sizes: List[int]) -> torch.Tensor:
        """Alltoall with caller-provided counts (MI355X extension): skips
        the host count exchanges when every rank already knows the routing
        (expert parallelism). ``target_counts[j]`` = slices rank j keeps
        along ``scatteraxis``; ``source_sizes[j]`` = rank j's current
        gather/partition-axis size. Backward: Alltoallv with axes and
        count vectors swapped."""
        return self._comm.Alltoallv(tensor, gatheraxis, scatteraxis,
                                    target_counts, source_sizes)

    def AlltoallPairwise(self, tensor: torch.Tensor, axis: int,
                         send_counts: List[int],
                         recv_counts: List[int]) -> torch.Tensor:
        """Arbitrary pairwise-count alltoall along ``axis`` (MI355X
        extension): this rank sends ``send_counts[j]`` slices to rank j.
        The expert-parallel token dispatch/combine primitive — the
        reference's same-axis Alltoall only repartitions contiguous global
        intervals. Pass ``recv_counts=[]`` to have them exchanged
        automatically. Backward returns every received slice to its
        sender (exact adjoint)."""
        return self._comm.AlltoallPairwise(tensor, axis, send_counts,
                                           recv_counts)

    def Iallreduce(self, tensor: torch.Tensor, op: int) -> WaitHandle:
        """Non-blocking Allreduce (no autograd): returns a WaitHandle whose
        Wait() yields the reduced tensor. The overlap primitive behind
        gradient bucketing (mpi4torch_amd.parallel.DistributedDataParallel);
        not part of the reference API."""
        return WaitHandle(self._comm.Iallreduce(tensor, op))

    def Ireducescatter(self, tensor: torch.Tensor, op: int) -> WaitHandle:
        """Non-blocking equal-count flat reduce-scatter (no autograd):
        Wait() yields this rank's block of the elementwise reduction.
        The ZeRO-2 bucket primitive; not in the reference API."""
        return WaitHandle(self._comm.Ireducescatter(tensor, op))

    def Iallgather(self, tensor: torch.Tensor) -> WaitHandle:
        """Non-blocking equal-count flat allgather (no autograd): Wait()
        yields the rank-major concatenation (size * numel). The FSDP
        parameter-prefetch primitive; not in the reference API."""
        return WaitHandle(self._comm.Iallgather(tensor))

    def Isend(self, tensor: torch.Tensor, dest: int, tag: int) -> WaitHandle:
        """Non-blocking send; complete with Wait. Backward: reverse recv.

        Matching contract (ALL transports): FIFO per (peer, channel) — the
        n-th send posted to a peer matches its n-th recv from this rank.
        Unlike MPI, ``tag`` does NOT reorder matching (RCCL has no tags);
        it is metadata, validated against the receiver's tag under
        MPI4TORCH_AMD_DEBUG=1 so crossed transfers raise instead of
        silently swapping payloads.
        """
        return WaitHandle(self._comm.Isend(tensor, dest, tag))

    def Irecv(self, tensor: torch.Tensor, source: int, tag: int) -> WaitHandle:
        """Non-blocking receive into ``tensor``'s buffer; complete with
        Wait. Backward: reverse send.

        Matching is FIFO per (peer, channel); ``tag`` is validating
        metadata only — see Isend.
        """
        return WaitHandle(self._comm.Irecv(tensor, source, tag))

    def Wait(self, waithandle: WaitHandle) -> torch.Tensor:
        """Complete a non-blocking op. On GPU this inserts a stream wait —
        no host synchronization."""
        return self._comm.Wait(waithandle._handle)

    def Send(self, tensor: torch.Tensor, dest: int, tag: int) -> torch.Tensor:
        """Blocking send = Isend + Wait (reference src/__init__.py:234-236)."""
        handle = self._comm.Isend(tensor, dest, tag)
        return self._comm.Wait(handle)

    def Recv(self, tensor: torch.Tensor, source: int, tag: int) -> torch.Tensor:
        """Blocking recv = Irecv + Wait (reference src/__init__.py:238-240)."""
        handle = self._comm.Irecv(tensor, source, tag)
        return self._comm.Wait(handle)


# Friendlier alias for new code; MPI_Communicator keeps reference parity.
Communicator = MPI_Communicator


def _default_group_gloo_name() -> str:
    """Return the name of a gloo-backed c10d group spanning the default
    world, creating a companion gloo group if the default backend has no
    CPU path. This replaces the reference's import-time MPI_Init_thread
    (csrc/extension.cpp:1306-1394) with torchrun-era bootstrap."""
    import torch.distributed as dist

    pg = dist.distributed_c10d._get_default_group()
    try:
        pg._get_backend(torch.device("cpu"))
        return pg.group_name
    except Exception:
        companion = dist.new_group(backend="gloo")
        return companion.group_name


def init() -> None:
    """Bind COMM_WORLD to the torch.distributed world.

    If torch.distributed is not yet initialized but torchrun environment
    variables are present, initializes a gloo process group (the RCCL
    communicators bootstrap lazily from it on first GPU op). Safe to call
    multiple times. Without any distributed context COMM_WORLD is a local
    single-rank communicator.
    """
    if _C._get_world_group_name():
        return
    import torch.distributed as dist

    if dist.is_available() and dist.is_initialized():
        _C._set_world_group_name(_default_group_gloo_name())
        return
    if "RANK" in os.environ and "WORLD_SIZE" in os.environ:
        dist.init_process_group(backend="gloo")
        _C._set_world_group_name(_default_group_gloo_name())


_comm_world_cache = None
_comm_world_cache_name: Optional[str] = None


def _comm_world() -> MPI_Communicator:
    global _comm_world_cache, _comm_world_cache_name
    init()
    name = _C._get_world_group_name()
    if _comm_world_cache is None or _comm_world_cache_name != name:
        _comm_world_cache = MPI_Communicator(torch.ops.mpi4torch_amd.comm_world())
        _comm_world_cache_name = name
    return _comm_world_cache


def __getattr__(name: str):
    # Lazy COMM_WORLD (PEP 562): unlike the reference we must not rendezvous
    # at import time — RCCL/c10d setup is deferred to first use.
    if name == "COMM_WORLD":
        return _comm_world()
    raise AttributeError(f"module 'mpi4torch_amd' has no attribute '{name}'")


def comm_from_process_group(pg) -> MPI_Communicator:
    """Wrap an existing torch.distributed ProcessGroup as a communicator.

    The torch.distributed analog of the reference's comm_from_mpi4py
    (src/__init__.py:247-261): sub-communicators are torch.distributed
    groups. If the group's backend has no CPU (gloo) path, a companion gloo
    group over the same ranks is created (collective over the group's
    members only — every member must reach this point).
    """
    import torch.distributed as dist

    try:
        pg._get_backend(torch.device("cpu"))
        name = pg.group_name
    except Exception:
        ranks = dist.get_process_group_ranks(pg)
        companion = dist.new_group(ranks=ranks, backend="gloo",
                                   use_local_synchronization=True)
        name = companion.group_name
    return MPI_Communicator(torch.ops.mpi4torch_amd.comm_from_group(name))


def comm_split(comm: MPI_Communicator, color: int) -> Optional[MPI_Communicator]:
    """MPI_Comm_split analog: partition `comm` into sub-communicators by
    `color`. Color < 0 (MPI_UNDEFINED style) returns None. Collective over
    `comm` (every rank must call it).

    Members are ranked by ascending global rank (torch.distributed groups
    fix this ordering, so MPI's `key` reordering is not supported).

    The reference only obtains sub-communicators through mpi4py
    (src/__init__.py:247-261); here they are torch.distributed groups
    created on the fly.
    """
    import torch.distributed as dist

    world = comm.size
    if world == 1 or not (dist.is_available() and dist.is_initialized()):
        # splitting a singleton / local communicator is the identity
        return comm if color >= 0 else None
    # dist.new_group takes GLOBAL ranks (of the default world), while this
    # comm's members may already be a subgroup with renumbered local ranks —
    # so allgather (color, global_rank) pairs over `comm` and pass the
    # global ranks through. use_local_synchronization makes new_group
    # collective over the subgroup only, so splitting a split communicator
    # works (non-members never call it).
    me_global = dist.get_rank()
    pairs_t = comm.Allgather(
        torch.tensor([float(color), float(me_global)], dtype=torch.float64), 0)
    pairs = pairs_t.view(world, 2)
    colors = [int(pairs[r, 0].item()) for r in range(world)]
    granks = [int(pairs[r, 1].item()) for r in range(world)]
    if color < 0:
        return None
    ranks = sorted(granks[r] for r in range(world) if colors[r] == color)
    mine = dist.new_group(ranks=ranks, backend="gloo",
                          use_local_synchronization=True)
    return comm_from_process_group(mine)


def comm_from_mpi4py(comm) -> MPI_Communicator:
    """Reference-parity stub (src/__init__.py:247-261): the MI355X build has
    no MPI runtime. Use comm_from_process_group with a torch.distributed
    group instead."""
    raise RuntimeError(
        "mpi4torch_amd does not use MPI; convert your communicator to a "
        "torch.distributed process group and call comm_from_process_group()"
    )


def force_host_staging(enabled: bool = True) -> None:
    """Debug toggle: route GPU tensors through host memory + gloo instead of
    RCCL/xGMI. The MI355X analog of the reference's
    deactivate_cuda_aware_mpi_support (csrc/extension.cpp:1404-1414)."""
    _C.force_host_staging(enabled)


def deactivate_cuda_aware_mpi_support() -> None:
    """Reference-parity alias for force_host_staging(True)."""
    _C.force_host_staging(True)
