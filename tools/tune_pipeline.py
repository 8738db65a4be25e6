#!/usr/bin/env python3
"""Pipeline chunk-size sweep (1 GPU, world-1 force_full_path).

Even alone, the phased path overlaps the pack/unpack slab kernels
(compute stream) with the exchange copies (collective stream), so this
measures the real overlap mechanics and phase overheads; on a multi-GPU
node the wire term grows ~8x and the overlap win with it.

Measures a recv-marshaled Alltoall (the EP/bench layout, unpack-side) at
several payloads x chunk sizes, phased vs single-phase.
"""

import os
import sys

os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29591")
os.environ.setdefault("RANK", "0")
os.environ.setdefault("WORLD_SIZE", "1")
os.environ["MPI4TORCH_AMD_FORCE_FULL_PATH"] = "1"
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

dist.init_process_group("gloo", rank=0, world_size=1)
import mpi4torch_amd as m

m.init()
comm = m.COMM_WORLD


def timeit(fn, iters=15, warmup=4):
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
    return s.elapsed_time(e) / iters


print(f"{'payload':>9} {'chunkMB':>8} {'K':>2} {'ms':>8} {'GB/s(eff)':>10}")
for mib in (256, 1024, 2048):
    rows = (mib << 20) // (4096 * 4)  # fp32, inner dim 4096
    # recv-marshaled layout at P=1: gather axis 1, scatter axis 0
    x = torch.randn(rows, 4096, device="cuda")
    for chunk in ("off", 256, 128, 64, 32, 16):
        if chunk == "off":
            os.environ["MPI4TORCH_AMD_PIPELINE_MB"] = "0"
        else:
            os.environ["MPI4TORCH_AMD_PIPELINE_MB"] = str(chunk)
        m._C.reload_config()
        ms = timeit(lambda: comm.Alltoall(x, 1, 0, rows))
        k = 1 if chunk == "off" else min(4, max(1, -(-mib // int(chunk))))
        nbytes = x.numel() * 4 * 3  # view-send + copy + unpack ≈ 3x
        print(f"{mib:>7}Mi {str(chunk):>8} {k:>2} {ms:>8.3f} "
              f"{nbytes/ms/1e6:>10.1f}", flush=True)

os.environ["MPI4TORCH_AMD_PIPELINE_MB"] = "64"
m._C.reload_config()
dist.destroy_process_group()
