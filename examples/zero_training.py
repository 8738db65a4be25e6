"""Data-parallel training with ZeRO-1 optimizer-state sharding.

No DDP wrapper needed: gradients stay local after backward, and
ZeroRedundancyOptimizer.step() performs the one Reducescatter (sum+shard)
+ local Adam + one Allgather that implement sharded data parallelism.
Optimizer state is 1/P of the replicated equivalent.

Run: torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 2 \
        examples/zero_training.py
"""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import mpi4torch_amd as m4a
from mpi4torch_amd.parallel import ZeroRedundancyOptimizer

device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
comm = m4a.COMM_WORLD

torch.manual_seed(8)  # identical replicas
net = torch.nn.Sequential(
    torch.nn.Linear(128, 512), torch.nn.GELU(), torch.nn.Linear(512, 32)
).to(device)
zopt = ZeroRedundancyOptimizer(net.parameters(), torch.optim.AdamW, lr=1e-3)

torch.manual_seed(900 + comm.rank)  # per-rank shard of the data
x = torch.randn(64, 128, device=device)
y = torch.randint(0, 32, (64,), device=device)
for step in range(20):
    loss = torch.nn.functional.cross_entropy(net(x), y)
    zopt.zero_grad()
    loss.backward()
    zopt.step()  # reduce-scatter grads -> sharded AdamW -> allgather params
    if comm.rank == 0 and step % 5 == 0:
        print(f"step {step}: loss {float(loss):.4f}")
