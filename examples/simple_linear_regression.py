"""Data-parallel linear regression with autograd-transparent Allreduce.

The flagship usage pattern (parity with reference
examples/simple_linear_regression.py, re-written for torchrun + MI355X):
the loss function itself contains the collectives, and loss.backward()
produces correct distributed gradients.

Run:
  torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 2 \
      examples/simple_linear_regression.py
or single-process:  python examples/simple_linear_regression.py
"""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import mpi4torch_amd as m4a
from mpi4torch_amd.models.linreg import DistributedLinReg

device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
comm = m4a.COMM_WORLD

# every rank holds its own shard of the data
torch.manual_seed(42)
w_true = torch.randn(8, device=device)
torch.manual_seed(1000 + comm.rank)
x = torch.randn(4096, 8, device=device)
y = x @ w_true + 0.01 * torch.randn(4096, device=device)

model = DistributedLinReg(comm, n_features=8).to(device)
optimizer = torch.optim.LBFGS(model.parameters(), max_iter=100)


def closure():
    optimizer.zero_grad()
    loss = model.loss(x, y)  # contains the Allreduces
    loss.backward()          # adjoint Allreduces run here
    return loss


final = optimizer.step(closure)
w = comm.Allreduce(model.weight.detach(), m4a.MPI_SUM) / comm.size
if comm.rank == 0:
    print(f"loss={float(final):.6f}")
    print("max |w - w_true| =", float((w - w_true).abs().max()))
