"""GPipe pipeline: each rank owns one segment; gradients flow through the
p2p adjoints automatically.

Run: torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 2 \
        examples/pipeline_training.py
"""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import mpi4torch_amd as m4a
from mpi4torch_amd.parallel import GPipe

device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
comm = m4a.COMM_WORLD
rank, world = comm.rank, comm.size

d, batch, n_mb = 32, 16, 4
torch.manual_seed(21)  # same full model on all ranks; keep our segment
segments = [
    torch.nn.Sequential(torch.nn.Linear(d, d), torch.nn.Tanh())
    for _ in range(world)
]
stage = segments[rank].to(device)
pipe = GPipe(stage, recv_shape=(batch, d), recv_dtype=torch.float32)
opt = torch.optim.SGD(stage.parameters(), lr=1e-2)

torch.manual_seed(33)
for step in range(10):
    data = [torch.randn(batch, d) for _ in range(n_mb)]
    targets = [torch.randn(batch, d, device=device) for _ in range(n_mb)]
    opt.zero_grad(set_to_none=True)
    losses = pipe.run(
        microbatches=data if pipe.is_first else None,
        loss_fn=(lambda y, i: torch.nn.functional.mse_loss(y, targets[i]))
        if pipe.is_last else None,
        n_microbatches=n_mb,
    )
    opt.step()
    if pipe.is_last and step % 3 == 0:
        print(f"step {step}: loss {sum(float(l) for l in losses) / n_mb:.4f}")
