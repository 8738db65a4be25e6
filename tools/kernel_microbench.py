#!/usr/bin/env python3
"""CDNA4 kernel micro-benchmarks (1 GPU): effective HBM bandwidth of the
local-reduce and marshaling kernels vs the ~8 TB/s HBM3E peak.

Each kernel moves (nranks*N reads + N writes) or (N reads + N writes) —
reported GB/s counts actual bytes touched. Run on an MI355X box:
    python tools/kernel_microbench.py
"""

import os
import sys

os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29581")
os.environ.setdefault("RANK", "0")
os.environ.setdefault("WORLD_SIZE", "1")
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

dist.init_process_group("gloo", rank=0, world_size=1)
import mpi4torch_amd as m

m.init()
assert torch.cuda.is_available()


def timeit(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters  # ms


def report(name, ms, nbytes):
    print(f"{name:34s} {ms*1e3:9.1f} us  {nbytes/ms/1e6:8.1f} GB/s", flush=True)


P = 8
N = 64 << 20  # 64M elements per chunk

# fp8 local reduce: P chunks fp8 -> 1 chunk fp8, fp32 accumulation
stk8 = (torch.randn(P, N, device="cuda") * 0.2).to(torch.float8_e4m3fn)
ms = timeit(lambda: m._C._fp8_reduce(stk8, 0))
report(f"fp8_reduce  P={P} N={N>>20}Mi", ms, (P + 1) * N)

# bitwise reduce: int32
stki = torch.randint(0, 1 << 30, (P, N // 4), device="cuda",
                     dtype=torch.int32)
ms = timeit(lambda: m._C._bitwise_reduce(stki, 2))
report(f"bitwise_reduce int32 P={P}", ms, (P + 1) * N)

# pairloc reduce: fp32 pairs
npairs = N // 8
stkp = torch.stack([torch.randn(P, npairs, device="cuda"),
                    torch.randint(0, 99, (P, npairs), device="cuda").float()],
                   dim=-1)
ms = timeit(lambda: m._C._pairloc_reduce(stkp, 0))
report(f"pairloc_reduce fp32 P={P}", ms, (P + 1) * npairs * 8)

# slab pack/unpack roundtrip via debug entry (counts partition the axis):
# before=4096 rows, axis=4096, after=16 fp32 -> 1 GiB tensor, 8 blocks
x = torch.randn(4096, 4096, 16, device="cuda")
counts = [512] * 8
ms = timeit(lambda: m._C._pack_roundtrip(x, 1, counts), iters=10)
report("slab pack+unpack 4 GiB moved", ms, 4 * x.numel() * 4)

# contiguous fast_clone path (nontemporal streaming copy) via Allreduce w1
t = torch.randn(N, device="cuda")
comm = m.COMM_WORLD
ms = timeit(lambda: comm.Allreduce(t, m.MPI_SUM))
report("w1 fast-path clone 64Mi fp32", ms, 2 * N * 4)

dist.destroy_process_group()
