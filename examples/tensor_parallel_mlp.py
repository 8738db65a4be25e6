"""Tensor-parallel MLP block: one forward Allreduce, adjoint backward.

Run: torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 2 \
        examples/tensor_parallel_mlp.py
"""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import mpi4torch_amd as m4a
from mpi4torch_amd.parallel import TensorParallelMLP

device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
comm = m4a.COMM_WORLD

d_model, d_hidden = 64, 64 * comm.size
mlp = TensorParallelMLP(d_model, d_hidden).to(device)
opt = torch.optim.SGD(mlp.parameters(), lr=1e-2)

torch.manual_seed(3)  # identical data on all ranks (activations replicated)
for step in range(10):
    x = torch.randn(32, d_model, device=device)
    out = mlp(x)
    loss = (out ** 2).mean()
    opt.zero_grad(set_to_none=True)
    loss.backward()  # the Row layer's adjoint Allreduce runs here
    opt.step()
    if comm.rank == 0 and step % 3 == 0:
        print(f"step {step}: loss {float(loss):.6f}")
