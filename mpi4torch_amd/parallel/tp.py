"""Tensor parallelism on the autograd-transparent primitives.

The reference never ships TP but its primitives express it (SURVEY.md
§2.5). These modules are the canonical Megatron-style pair realized purely
with autograd-transparent collectives — no hand-written autograd functions.

Gradient-scaling convention. Under SPMD autodiff the loss appears
(identically) on every rank and every rank runs backward, so each
collective's adjoint receives P submissions: a RowParallelLinear output's
Allreduce turns a gradient g into P*g on the way down. Two corrections make
everything come out at true scale:

* entry: ColumnParallelLinear passes its (replicated) input through
  ``Allreduce(x)/P`` — forward-identity on replicated data, backward
  ``Allreduce(grad)/P`` — which both sums the per-rank shard contributions
  to dL/dx and cancels the P factor, so everything UPSTREAM of the TP block
  gets the exact gradient;
* parameters: each TP weight registers a post-accumulate hook dividing its
  gradient by P, cancelling the same factor locally.

With those, TP modules compose with ordinary modules and optimizers with
no user-side scaling.
"""

import torch

import mpi4torch_amd as m4a


def _register_tp_grad_scale(module: torch.nn.Module, world: int):
    if world <= 1:
        return
    inv = 1.0 / world

    def _scale(param, _inv=inv):
        param.grad.mul_(_inv)

    for p in module.parameters():
        p.register_post_accumulate_grad_hook(_scale)


def copy_to_tp_region(x: torch.Tensor, comm) -> torch.Tensor:
    """Forward-identity on replicated input; backward sums shard
    contributions across ranks (and cancels the P-fold replication factor).
    The Megatron 'f' operator, expressed with a plain Allreduce."""
    if comm.size == 1:
        return x
    return comm.Allreduce(x, m4a.MPI_SUM) / comm.size


class ColumnParallelLinear(torch.nn.Module):
    """y_local = x @ W_local^T + b_local with W sharded on the output dim.

    With gather_output=True the full y is materialized via Allgather (whose
    backward reduce-scatters the gradient correctly — this framework's
    Allgather adjoint is a true reduce-scatter).
    """

    def __init__(self, in_features: int, out_features: int, comm=None,
                 bias: bool = True, gather_output: bool = False):
        super().__init__()
        self.comm = comm if comm is not None else m4a.COMM_WORLD
        world = self.comm.size
        assert out_features % world == 0, (
            f"out_features {out_features} must divide world size {world}")
        self.out_local = out_features // world
        self.linear = torch.nn.Linear(in_features, self.out_local, bias=bias)
        self.gather_output = gather_output
        _register_tp_grad_scale(self, world)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = copy_to_tp_region(x, self.comm)
        y = self.linear(x)
        if self.gather_output and self.comm.size > 1:
            y = self.comm.Allgather(y, y.dim() - 1)
        return y

    def shard_from_full(self, weight: torch.Tensor, bias=None):
        """Load this rank's slice of a full [out, in] weight (testing and
        checkpoint import)."""
        r = self.comm.rank
        with torch.no_grad():
            self.linear.weight.copy_(
                weight[r * self.out_local:(r + 1) * self.out_local])
            if bias is not None and self.linear.bias is not None:
                self.linear.bias.copy_(
                    bias[r * self.out_local:(r + 1) * self.out_local])


class RowParallelLinear(torch.nn.Module):
    """y = Allreduce_SUM(x_local @ W_local^T) + b with W sharded on the
    input dim. Expects input already sharded on the last dim (the layout
    ColumnParallelLinear produces with gather_output=False)."""

    def __init__(self, in_features: int, out_features: int, comm=None,
                 bias: bool = True):
        super().__init__()
        self.comm = comm if comm is not None else m4a.COMM_WORLD
        world = self.comm.size
        assert in_features % world == 0, (
            f"in_features {in_features} must divide world size {world}")
        self.in_local = in_features // world
        self.linear = torch.nn.Linear(self.in_local, out_features, bias=False)
        self.bias = (torch.nn.Parameter(torch.zeros(out_features))
                     if bias else None)
        # the /P correction applies only BELOW the output Allreduce (whose
        # adjoint multiplies upstream gradients by P). The bias adds AFTER
        # it: its gradient is already at true scale on every rank, so
        # scaling it would under-train the bias by the TP degree.
        _register_tp_grad_scale(self.linear, world)

    def forward(self, x_local: torch.Tensor) -> torch.Tensor:
        partial = self.linear(x_local)
        if self.comm.size > 1:
            partial = self.comm.Allreduce(partial, m4a.MPI_SUM)
        return partial + self.bias if self.bias is not None else partial

    def shard_from_full(self, weight: torch.Tensor, bias=None):
        r = self.comm.rank
        with torch.no_grad():
            self.linear.weight.copy_(
                weight[:, r * self.in_local:(r + 1) * self.in_local])
            if bias is not None and self.bias is not None:
                self.bias.copy_(bias)


class TensorParallelMLP(torch.nn.Module):
    """The canonical TP block: Column -> activation -> Row. One forward
    Allreduce per block; backward communication generated by the
    collectives' adjoints."""

    def __init__(self, d_model: int, d_hidden: int, comm=None,
                 activation=torch.nn.functional.gelu):
        super().__init__()
        self.comm = comm if comm is not None else m4a.COMM_WORLD
        self.up = ColumnParallelLinear(d_model, d_hidden, self.comm,
                                       gather_output=False)
        self.down = RowParallelLinear(d_hidden, d_model, self.comm)
        self.activation = activation

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.down(self.activation(self.up(x)))
