#!/usr/bin/env python3
"""Extended single-GPU fuzz: random geometries through the FULL native
pipeline (force_full_path, forced 4-phase pipelining) on CUDA, checked
against the same ops computed on CPU tensors (gloo/local path) — kernel
marshaling, phased self-exchanges, hierarchical lowerings, dtype sweep.

Usage: python tools/fuzz_gpu.py [seconds]
"""

import os
import random
import sys
import time

os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29601")
os.environ.setdefault("RANK", "0")
os.environ.setdefault("WORLD_SIZE", "1")
os.environ["MPI4TORCH_AMD_FORCE_FULL_PATH"] = "1"
os.environ["MPI4TORCH_AMD_PIPELINE_MB"] = "0.0005"  # force phases
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

dist.init_process_group("gloo", rank=0, world_size=1)
import mpi4torch_amd as m

m.init()
comm = m.COMM_WORLD
secs = float(sys.argv[1]) if len(sys.argv) > 1 else 180.0
rng = random.Random(int(sys.argv[2]) if len(sys.argv) > 2 else 20260914)

DTYPES = [torch.float32, torch.float64, torch.bfloat16, torch.float16,
          torch.int32, torch.int64, torch.uint8]

start = time.time()
it = 0
while time.time() - start < secs:
    it += 1
    ndim = rng.randint(1, 4)
    shape = [rng.randint(1, 9) for _ in range(ndim)]
    axis = rng.randint(0, ndim - 1)
    dtype = rng.choice(DTYPES)
    if dtype.is_floating_point:
        x = torch.randn(shape).to(dtype)
    else:
        x = torch.randint(0, 100, shape).to(dtype)
    xg = x.cuda()

    # world-1 semantics: every op is an identity of some form; GPU result
    # must match the CPU tensor bit-for-bit
    g = comm.Gather(xg, axis, 0).cpu()
    assert (g == x).all(), ("gather", shape, axis, dtype)
    ag = comm.Allgather(xg, axis).cpu()
    assert (ag == x).all(), ("allgather", shape, axis, dtype)
    sc = comm.Scatter(xg, axis, shape[axis], 0).cpu()
    assert (sc == x).all(), ("scatter", shape, axis, dtype)
    if ndim >= 2:
        gax = axis
        sax = (axis + 1) % ndim
        a2 = comm.Alltoall(xg, gax, sax, shape[sax]).cpu()
        assert (a2 == x).all(), ("alltoall", shape, gax, sax, dtype)
    a2s = comm.Alltoall(xg, axis, axis, shape[axis]).cpu()
    assert (a2s == x).all(), ("alltoall-same", shape, axis, dtype)

    if dtype in (torch.int32, torch.int64, torch.uint8):
        for op in (m.MPI_BAND, m.MPI_BOR, m.MPI_BXOR):
            r = comm.Allreduce(xg, op).cpu()
            assert (r == x).all(), ("bitwise", op, shape, dtype)
    else:
        r = comm.Allreduce(xg, m.MPI_SUM).cpu()
        assert (r == x).all(), ("sum", shape, dtype)

    # pairloc identity on random pairs
    if dtype.is_floating_point and rng.random() < 0.3:
        n = rng.randint(1, 500)
        pr = torch.stack([torch.randn(n), torch.randint(0, 50, (n,)).float()],
                         dim=-1).to(dtype)
        got = comm.Allreduce(pr.cuda(), m.MPI_MINLOC).cpu()
        assert (got.view(torch.uint8) == pr.view(torch.uint8)).all()

    # fp8 hierarchical identity
    if rng.random() < 0.2:
        n = rng.randint(1, 5000)
        t8 = (torch.randn(n) * 0.3).to(torch.float8_e4m3fn)
        got = comm.Allreduce(t8.cuda(), m.MPI_SUM).cpu()
        assert (got.view(torch.uint8) == t8.view(torch.uint8)).all()

    if it % 500 == 0:
        torch.cuda.synchronize()
        print(f"iter {it}: ok ({time.time()-start:.0f}s)", flush=True)

torch.cuda.synchronize()
print(f"GPU FUZZ OK: {it} random geometries, all exact", flush=True)
dist.destroy_process_group()
