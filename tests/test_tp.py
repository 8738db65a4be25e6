"""Tensor parallelism: forward and backward must match a dense
single-process reference exactly (values AND gradients, including the
SPMD gradient-scaling corrections)."""

import torch

from spmd import run_spmd


def _tp_mlp_worker(rank, world):
    from mpi4torch_amd.parallel.tp import TensorParallelMLP

    torch.manual_seed(7)  # identical on all ranks
    d_model, d_hidden, batch = 6, 4 * world, 5
    w1 = torch.randn(d_hidden, d_model, dtype=torch.double)
    b1 = torch.randn(d_hidden, dtype=torch.double)
    w2 = torch.randn(d_model, d_hidden, dtype=torch.double)
    x = torch.randn(batch, d_model, dtype=torch.double).requires_grad_()

    mlp = TensorParallelMLP(d_model, d_hidden,
                            activation=torch.nn.functional.relu).double()
    mlp.up.shard_from_full(w1, b1)
    mlp.down.shard_from_full(w2)
    with torch.no_grad():
        mlp.down.bias.zero_()

    y = mlp(x)
    loss = (y ** 2).sum()
    loss.backward()

    # dense reference (incl. the Row bias, replicated across TP ranks —
    # it adds AFTER the output allreduce so its grad must be at TRUE
    # dense scale, not /P; regression test for the scaling-hook bug the
    # FSDPxTP hybrid test caught)
    x_ref = x.detach().clone().requires_grad_()
    w1_ref = w1.clone().requires_grad_()
    b1_ref = b1.clone().requires_grad_()
    w2_ref = w2.clone().requires_grad_()
    b2_ref = torch.zeros(d_model, dtype=torch.double, requires_grad=True)
    y_ref = torch.relu(x_ref @ w1_ref.t() + b1_ref) @ w2_ref.t() + b2_ref
    ((y_ref ** 2).sum()).backward()
    assert torch.allclose(mlp.down.bias.grad, b2_ref.grad, atol=1e-10)

    assert torch.allclose(y, y_ref, atol=1e-10), (y - y_ref).abs().max()
    # input gradient: exact (entry collective sums shard contributions)
    assert torch.allclose(x.grad, x_ref.grad, atol=1e-10)
    # weight gradients: this rank's shard of the dense gradients
    lo, hi = rank * mlp.up.out_local, (rank + 1) * mlp.up.out_local
    assert torch.allclose(mlp.up.linear.weight.grad, w1_ref.grad[lo:hi],
                          atol=1e-10)
    assert torch.allclose(mlp.up.linear.bias.grad, b1_ref.grad[lo:hi],
                          atol=1e-10)
    lo2, hi2 = rank * mlp.down.in_local, (rank + 1) * mlp.down.in_local
    assert torch.allclose(mlp.down.linear.weight.grad,
                          w2_ref.grad[:, lo2:hi2], atol=1e-10)


def _tp_column_gather_worker(rank, world):
    from mpi4torch_amd.parallel.tp import ColumnParallelLinear

    torch.manual_seed(11)
    d_in, d_out, batch = 5, 3 * world, 4
    w = torch.randn(d_out, d_in, dtype=torch.double)
    b = torch.randn(d_out, dtype=torch.double)
    x = torch.randn(batch, d_in, dtype=torch.double).requires_grad_()

    col = ColumnParallelLinear(d_in, d_out, gather_output=True).double()
    col.shard_from_full(w, b)
    y = col(x)
    y.sum().backward()

    x_ref = x.detach().clone().requires_grad_()
    w_ref = w.clone().requires_grad_()
    y_ref = x_ref @ w_ref.t() + b
    y_ref.sum().backward()

    assert torch.allclose(y, y_ref, atol=1e-10)
    assert torch.allclose(x.grad, x_ref.grad, atol=1e-10)
    lo, hi = rank * col.out_local, (rank + 1) * col.out_local
    assert torch.allclose(col.linear.weight.grad, w_ref.grad[lo:hi],
                          atol=1e-10)


def test_tp_mlp_ws2():
    run_spmd(2, _tp_mlp_worker)


def test_tp_mlp_ws5():
    run_spmd(5, _tp_mlp_worker)


def test_tp_column_gather_ws2():
    run_spmd(2, _tp_column_gather_worker)
