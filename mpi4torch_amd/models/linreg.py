"""Data-parallel linear regression on autograd-transparent collectives.

The pattern of the reference's flagship example
(examples/simple_linear_regression.py:27-53, doc/examples.rst:22-65): the
loss function itself contains two Allreduces — a parameter average that
keeps optimizer replicas in lockstep, and a loss sum — and
``loss.backward()`` produces correct distributed gradients because both
collectives are autograd nodes. BASELINE.json config #3.
"""

import torch

import mpi4torch_amd as m4a


class DistributedLinReg(torch.nn.Module):
    def __init__(self, comm, n_features: int = 8):
        super().__init__()
        self.comm = comm
        self.weight = torch.nn.Parameter(torch.zeros(n_features))
        self.bias = torch.nn.Parameter(torch.zeros(1))

    def loss(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        comm = self.comm
        # parameter averaging keeps per-rank optimizer replicas consistent
        # (rationale: reference doc/examples.rst:46-65)
        w = comm.Allreduce(self.weight, m4a.MPI_SUM) / comm.size
        b = comm.Allreduce(self.bias, m4a.MPI_SUM) / comm.size
        pred = x @ w + b
        local = ((pred - y) ** 2).sum()
        # sum of local losses across ranks — backward distributes grads
        return comm.Allreduce(local, m4a.MPI_SUM)
