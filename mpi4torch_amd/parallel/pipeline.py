"""Pipeline parallelism on autograd-transparent point-to-point ops.

The reference ships no PP but its Isend/Irecv/Wait + JoinDummies machinery
is exactly what PP needs (reference doc/basic_usage.rst:
194-457). Because Send/Recv are autograd-transparent — the backward of a
completed Send *receives* the downstream gradient over the dedicated
backward channel, and the backward of a Recv *sends* the gradient upstream
— a GPipe-style schedule needs no hand-written gradient plumbing at all:
each rank just calls backward on what it produced.

Scheduling contract: all ranks iterate microbatches forward in the same
order and backward in the same (reversed) order; the per-peer FIFO
matching of the p2p channels then pairs every transfer correctly.
"""

from typing import Callable, List, Optional, Sequence

import torch

import mpi4torch_amd as m4a


class GPipe:
    """Fill-drain (GPipe) pipeline over `comm`: rank r runs `stage` as the
    r-th segment of the model.

    Parameters
    ----------
    stage: this rank's submodule.
    recv_shape / recv_dtype: activation shape/dtype arriving from the
        previous stage (required on every rank but the first).
    comm: communicator (default COMM_WORLD).

    run(...) executes forward for every microbatch, then backward in
    reverse order, accumulating parameter gradients; it returns the list
    of per-microbatch losses on the last rank ([] elsewhere).
    """

    def __init__(self, stage: torch.nn.Module, recv_shape=None,
                 recv_dtype=torch.float32, comm=None):
        self.comm = comm if comm is not None else m4a.COMM_WORLD
        self.stage = stage
        self.recv_shape = recv_shape
        self.recv_dtype = recv_dtype
        if self.comm.rank > 0:
            assert recv_shape is not None, (
                "recv_shape is required on ranks > 0")

    @property
    def is_first(self) -> bool:
        return self.comm.rank == 0

    @property
    def is_last(self) -> bool:
        return self.comm.rank == self.comm.size - 1

    def run(self, microbatches: Optional[Sequence[torch.Tensor]] = None,
            loss_fn: Optional[Callable[[torch.Tensor, int], torch.Tensor]] = None,
            n_microbatches: Optional[int] = None) -> List[torch.Tensor]:
        comm = self.comm
        rank, world = comm.rank, comm.size
        if self.is_first:
            assert microbatches is not None
            n = len(microbatches)
        else:
            assert n_microbatches is not None
            n = n_microbatches
        if self.is_last:
            assert loss_fn is not None

        device = next(self.stage.parameters()).device

        outputs: List[torch.Tensor] = []
        losses: List[torch.Tensor] = []
        # ---- forward fill: microbatch i is labeled tag i (metadata only —
        # matching is FIFO per (peer, channel); the fill loop posts
        # microbatches in the same order on every rank, which is what
        # actually keeps them paired. MPI4TORCH_AMD_DEBUG=1 validates.)
        for i in range(n):
            if self.is_first:
                x = microbatches[i].to(device)
            else:
                # requires_grad so the Recv participates in autograd — its
                # backward is what SENDS dL/dx to the previous stage
                buf = torch.empty(self.recv_shape, dtype=self.recv_dtype,
                                  device=device, requires_grad=True)
                x = comm.Recv(buf, rank - 1, i)
            y = self.stage(x)
            if self.is_last:
                losses.append(loss_fn(y, i))
                outputs.append(y)
            else:
                # Send returns (an alias of) y with the transfer recorded in
                # its autograd history; backward on it receives dL/dy from
                # the next stage over the backward channel.
                outputs.append(comm.Send(y, rank + 1, i))

        # ---- backward drain, reverse order on every rank
        for i in reversed(range(n)):
            if self.is_last:
                losses[i].backward()
            else:
                # the seed is ignored for a completed send — the true
                # gradient arrives via the reverse transfer
                outputs[i].backward(torch.zeros_like(outputs[i]))

        return [l.detach() for l in losses] if self.is_last else []
