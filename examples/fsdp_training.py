"""ZeRO-3-lite (FSDP) training: parameters sharded at rest.

Each Linear block is an FSDP unit: its full parameters exist only while
its forward/backward runs (one Allgather per use); gradients leave as
reduce-scatters; the optimizer and gradient clipping operate on the 1/P
shards.

Run: torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 2 \
        examples/fsdp_training.py
"""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import mpi4torch_amd as m4a
from mpi4torch_amd.parallel import FullyShardedDataParallel
from mpi4torch_amd.utils import clip_grad_norm_sharded

device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
comm = m4a.COMM_WORLD

torch.manual_seed(12)
blocks = [torch.nn.Sequential(torch.nn.Linear(64, 256), torch.nn.GELU(),
                              torch.nn.Linear(256, 64))
          for _ in range(4)]
net = torch.nn.Sequential(*blocks).to(device)
model = FullyShardedDataParallel(net, units=blocks)
opt = torch.optim.AdamW(model.shard_parameters(), lr=1e-3)

torch.manual_seed(70 + comm.rank)
x = torch.randn(32, 64, device=device)
for step in range(15):
    loss = (model(x) - x).square().mean()
    model.zero_grad()
    loss.backward()
    model.finish_backward()
    clip_grad_norm_sharded(model.shard_parameters(), max_norm=1.0)
    opt.step()
    model.refresh_shards()
    if comm.rank == 0 and step % 5 == 0:
        print(f"step {step}: loss {float(loss):.4f}")
