"""Distributed argmin via MPI_MINLOC (value, location) pairs.

Each rank evaluates a chunk of a search space; MINLOC finds, per search
problem, the globally best value AND which global index produced it — one
collective, no gather of the full space. This is the classic MPI pair-type
pattern (the reference maps MINLOC/MAXLOC at csrc/extension.cpp:204-252
but its dtype table could never feed them; here they are first-class:
pairs ride the last axis, ties go to the smallest location).

Run: torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 4 \
        examples/distributed_argmin.py
"""

import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import mpi4torch_amd as m4a

comm = m4a.COMM_WORLD
rank, world = comm.rank, comm.size

# 8 independent search problems; each rank scans 1000 candidates of each
problems, local_n = 8, 1000
torch.manual_seed(1234)  # same objective on every rank
centers = torch.randn(problems)

# this rank's candidate slice of the global [world * local_n] grid
xs = torch.linspace(-3, 3, world * local_n)[rank * local_n:(rank + 1) * local_n]
# objective value per (problem, candidate); keep the per-rank best
vals = (xs.unsqueeze(0) - centers.unsqueeze(1)).abs()  # [problems, local_n]
best_val, best_idx = vals.min(dim=1)
best_global_idx = (best_idx + rank * local_n).to(best_val.dtype)

pairs = torch.stack([best_val, best_global_idx], dim=-1)  # [problems, 2]
winner = comm.Allreduce(pairs, m4a.MPI_MINLOC)

if rank == 0:
    grid = torch.linspace(-3, 3, world * local_n)
    for p in range(problems):
        v, loc = winner[p, 0].item(), int(winner[p, 1].item())
        # verify against the serial answer
        serial = (grid - centers[p]).abs().argmin().item()
        assert loc == serial, (p, loc, serial)
        print(f"problem {p}: argmin at global index {loc} "
              f"(x={grid[loc]:.4f}, |x-c|={v:.5f})")
    print("distributed argmin == serial argmin for all problems")

dist.barrier()
dist.destroy_process_group()
