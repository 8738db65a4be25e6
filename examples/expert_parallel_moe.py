"""Expert-parallel MoE training: token dispatch over AlltoallPairwise.

Experts are sharded across ranks (each rank owns n_experts/P of them);
tokens travel to their expert's owner and back through two pairwise-count
Alltoalls whose backward routes gradients automatically. The router is
replicated, so only ITS gradients are averaged; expert gradients stay
rank-local by construction.

Run: torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 2 \
        examples/expert_parallel_moe.py
"""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import mpi4torch_amd as m4a
from mpi4torch_amd.models.moe import ExpertParallelMoE

device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
comm = m4a.COMM_WORLD

torch.manual_seed(4)  # router + this rank's experts (router identical everywhere)
d_model, n_experts = 32, 4 * comm.size
moe = ExpertParallelMoE(d_model, n_experts).to(device)
opt = torch.optim.AdamW(moe.parameters(), lr=1e-3)

torch.manual_seed(500 + comm.rank)  # per-rank token stream
for step in range(15):
    x = torch.randn(64, d_model, device=device)
    y = moe(x)
    loss = (y - x).square().mean()  # toy reconstruction objective
    opt.zero_grad(set_to_none=True)
    loss.backward()
    # replicated router: average gradients across ranks; sharded experts:
    # their gradients are already exactly local
    with torch.no_grad():
        for p in moe.router.parameters():
            if p.grad is not None:
                p.grad.copy_(comm.Allreduce(p.grad, m4a.MPI_SUM) / comm.size)
    opt.step()
    if comm.rank == 0 and step % 5 == 0:
        print(f"step {step}: loss {float(loss):.4f}")
