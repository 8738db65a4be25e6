"""Non-blocking ring exchange with correct dummy wiring.

Parity with reference examples/isend-recv-wait.py: each rank sends its
tensor to the next rank while receiving from the previous one; JoinDummies
encodes the ordering dependencies the autograd DAG cannot see on its own.

Run: torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 2 \
        examples/isend_irecv_wait.py
"""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import mpi4torch_amd as m4a

device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
comm = m4a.COMM_WORLD
rank, world = comm.rank, comm.size

a = torch.full((1000,), float(rank), device=device).requires_grad_()
req = comm.Isend(a, (rank + 1) % world, 0)
req2 = comm.Irecv(
    m4a.JoinDummies(torch.empty_like(a), [req.dummy]),
    (rank + world - 1) % world,
    0,
)
sent = comm.Wait(m4a.JoinDummiesHandle(req, [req2.dummy]))
received = comm.Wait(m4a.JoinDummiesHandle(req2, [sent]))

loss = (received * rank).sum()
loss.backward()
print(f"rank {rank}: received from {(rank + world - 1) % world}, "
      f"grad[0] = {float(a.grad[0])} (expected {(rank + 1) % world})")
