#!/usr/bin/env python3
"""Round-2 soak: hammer the NEW code paths (hierarchical fp8/bitwise/
pairloc allreduce, phased pipelined alltoall, deferred-request table,
Iallgather) at world 1 with force_full_path, asserting a flat memory
steady state. Usage: python tools/soak_r2.py [seconds]"""

import os
import sys
import time

os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29571")
os.environ.setdefault("RANK", "0")
os.environ.setdefault("WORLD_SIZE", "1")
os.environ["MPI4TORCH_AMD_FORCE_FULL_PATH"] = "1"
os.environ["MPI4TORCH_AMD_PIPELINE_MB"] = "0.01"  # force K=4 phases

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

dist.init_process_group("gloo", rank=0, world_size=1)
import mpi4torch_amd as m

m.init()
comm = m.COMM_WORLD
dev = torch.device("cuda:0")
secs = float(sys.argv[1]) if len(sys.argv) > 1 else 300.0

t8 = (torch.randn(1 << 20, device=dev) * 0.3).to(torch.float8_e4m3fn)
ti = torch.randint(0, 1 << 30, ((1 << 20) + 3,), device=dev,
                   dtype=torch.int32)
pairs = torch.stack([torch.randn(1 << 18, device=dev),
                     torch.arange(1 << 18, device=dev).float()], dim=-1)
x = torch.rand(8, 64, 32, device=dev, dtype=torch.bfloat16)
big = torch.rand(1 << 22, device=dev).requires_grad_()
seed = torch.ones(1 << 22, device=dev)

net = torch.nn.Linear(256, 256).to(dev)
from mpi4torch_amd.parallel import DistributedDataParallel

model = DistributedDataParallel(net, bucket_cap_mb=1)
opt = torch.optim.SGD(model.parameters(), lr=1e-4)

start = time.time()
it = 0
mem0 = None
while time.time() - start < secs:
    r8 = comm.Allreduce(t8, m.MPI_SUM)
    ri = comm.Allreduce(ti, m.MPI_BXOR)
    rp = comm.Allreduce(pairs, m.MPI_MINLOC)
    y = comm.Alltoall(x, 1, 0, 8)       # phased (K forced)
    z = comm.Alltoall(y, 0, 1, 64)
    big.grad = None
    ar = comm.Allreduce(big, m.MPI_SUM)
    ar.backward(seed)
    h1 = comm.Iallgather(t8.view(torch.uint8))
    h2 = comm.Ireducescatter(big.detach(), m.MPI_SUM)
    comm.Wait(h1), comm.Wait(h2)
    req = comm.Isend(x, 0, it % 5)
    got = comm.Recv(m.JoinDummies(torch.empty_like(x), [req.dummy]), 0,
                    it % 5)
    comm.Wait(m.JoinDummiesHandle(req, [got]))
    loss = model(torch.randn(64, 256, device=dev)).pow(2).mean()
    opt.zero_grad()
    loss.backward()
    model.finish_gradient_sync()
    opt.step()
    it += 1
    if it == 20:
        torch.cuda.synchronize()
        mem0 = torch.cuda.memory_allocated()
    if it % 200 == 0:
        torch.cuda.synchronize()
        cur = torch.cuda.memory_allocated()
        print(f"iter {it}: allocated {cur/1e6:.1f} MB "
              f"(delta vs iter20: {(cur-mem0)/1e6:+.2f} MB)", flush=True)
        assert cur <= mem0 + (32 << 20), "memory growth detected"

torch.cuda.synchronize()
cur = torch.cuda.memory_allocated()
print(f"SOAK OK: {it} iterations in {time.time()-start:.0f}s; "
      f"memory delta {(cur-mem0)/1e6:+.2f} MB", flush=True)
dist.destroy_process_group()
