"""Data-parallel MLP training with bucketed, overlapped gradient sync.

Run: torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 2 \
        examples/ddp_training.py
"""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import mpi4torch_amd as m4a
from mpi4torch_amd.parallel import DistributedDataParallel

device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
comm = m4a.COMM_WORLD

torch.manual_seed(10 + comm.rank)  # ranks start from different inits — DDP broadcasts rank 0's
net = torch.nn.Sequential(
    torch.nn.Linear(64, 256), torch.nn.GELU(), torch.nn.Linear(256, 10)
).to(device)
model = DistributedDataParallel(net, bucket_cap_mb=16)
opt = torch.optim.AdamW(model.parameters(), lr=1e-3)

torch.manual_seed(100 + comm.rank)  # per-rank data shard
for step in range(20):
    x = torch.randn(128, 64, device=device)
    y = torch.randint(0, 10, (128,), device=device)
    loss = torch.nn.functional.cross_entropy(model(x), y)
    opt.zero_grad(set_to_none=True)
    loss.backward()                 # bucket allreduces overlap backward
    model.finish_gradient_sync()    # wait + scatter into .grad
    opt.step()
    if comm.rank == 0 and step % 5 == 0:
        print(f"step {step}: loss {float(loss):.4f}")
