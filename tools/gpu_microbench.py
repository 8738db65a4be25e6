#!/usr/bin/env python3
"""MI355X microbenchmarks for the CDNA4 kernels + world-1 RCCL ops.

Measures the batched slab pack/unpack kernel (the axis-collective
marshaling path) against the HBM3E roofline (~6.3 TB/s achievable read+
write), and 1-rank RCCL allreduce latency/bandwidth. Run under gpurun.
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def timed(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29461")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    import torch.distributed as dist

    if not dist.is_initialized():
        dist.init_process_group("gloo", rank=0, world_size=1)
    import mpi4torch_amd as m

    m.init()

    print("== slab pack/unpack kernel (roundtrip = 4x tensor bytes moved) ==")
    for shape, axis, nblocks, dtype in [
        ((8, 65536, 512), 1, 8, torch.bfloat16),     # 512 MiB, strided middle axis
        ((1, 1 << 28), 1, 8, torch.bfloat16),        # 512 MiB, contiguous blocks
        ((64, 8192, 128), 1, 8, torch.float32),      # 256 MiB fp32
        ((8, 65537, 511), 1, 7, torch.bfloat16),     # odd sizes, 1B/4B paths
    ]:
        x = torch.randn(shape, device="cuda", dtype=torch.float32).to(dtype)
        n = x.size(axis)
        counts = [n // nblocks] * nblocks
        counts[-1] += n - sum(counts)
        t = timed(lambda: m._C._pack_roundtrip(x, axis, counts))
        nbytes = x.numel() * x.element_size()
        # roundtrip reads+writes the tensor twice (pack + unpack)
        print(f"shape={shape} dtype={dtype} blocks={nblocks}: "
              f"{t*1e3:.3f} ms  {4*nbytes/t/1e12:.2f} TB/s effective")

    print("== bitwise reduce kernel ==")
    stacked = torch.randint(0, 1 << 30, (8, 1 << 26), device="cuda",
                            dtype=torch.int32)  # 8 x 256 MiB
    t = timed(lambda: m._C._bitwise_reduce(stacked, 0))
    nbytes = stacked.numel() * 4 + stacked[0].numel() * 4
    print(f"8x256MiB int32 AND: {t*1e3:.3f} ms  {nbytes/t/1e12:.2f} TB/s")

    print("== fused fp8 reduce kernel (fp32 accumulation) ==")
    stacked8 = (torch.randn(8, 1 << 27, device="cuda") * 0.2).to(
        torch.float8_e4m3fn)  # 8 x 128 MiB
    t = timed(lambda: m._C._fp8_reduce(stacked8, 0))
    nbytes = stacked8.numel() + stacked8[0].numel()
    print(f"8x128MiB fp8 SUM: {t*1e3:.3f} ms  {nbytes/t/1e12:.2f} TB/s")

    print("== full-path axis collectives (world 1, pack+exchange+unpack) ==")
    m._C.force_full_path(True)
    try:
        comm1 = m.COMM_WORLD
        xb = torch.randn(8, 65536, 256, device="cuda", dtype=torch.float32).to(
            torch.bfloat16)  # 256 MiB, middle axis
        t = timed(lambda: comm1.Allgather(xb, 1))
        nbytes = xb.numel() * 2
        print(f"Allgather mid-axis 256MiB: {t*1e3:.3f} ms "
              f"{nbytes/t/1e9:.1f} GB/s payload")
        t = timed(lambda: comm1.Alltoall(xb, 1, 1, 65536))
        print(f"Alltoall same-axis 256MiB: {t*1e3:.3f} ms "
              f"{nbytes/t/1e9:.1f} GB/s payload")
    finally:
        m._C.force_full_path(False)

    print("== world-1 RCCL allreduce (copy bound) ==")
    comm = m.COMM_WORLD
    for mib in (64, 1024):
        x = torch.randn(mib * 1024 * 1024 // 2, device="cuda",
                        dtype=torch.float32).to(torch.bfloat16)
        t = timed(lambda: comm.Allreduce(x, m.MPI_SUM))
        nbytes = x.numel() * 2
        print(f"allreduce {mib} MiB bf16: {t*1e3:.3f} ms "
              f"{nbytes/t/1e9:.1f} GB/s algbw")

    print("== small-op latency (per call, current-stream wall) ==")
    for nbytes, label in ((4096, "4KB"), (1 << 20, "1MB")):
        xs = torch.randn(nbytes // 4, device="cuda")
        t = timed(lambda: comm.Allreduce(xs, m.MPI_SUM), iters=200, warmup=20)
        print(f"allreduce {label}: {t*1e6:.1f} us/op")
        h = None

        def ia():
            hh = comm.Iallreduce(xs, m.MPI_SUM)
            comm.Wait(hh)

        t = timed(ia, iters=200, warmup=20)
        print(f"iallreduce+wait {label}: {t*1e6:.1f} us/op")

    print("== allreduce fwd+bwd step (bench inner loop) ==")
    x = torch.randn(1 << 29, device="cuda", dtype=torch.float32).to(
        torch.bfloat16).requires_grad_()
    seed = torch.ones_like(x)

    def step():
        x.grad = None
        r = comm.Allreduce(x, m.MPI_SUM)
        r.backward(seed)

    t = timed(step, iters=10, warmup=3)
    print(f"1 GiB fwd+bwd: {t*1e3:.3f} ms  "
          f"{2 * x.numel() * 2 / t / 1e9:.1f} GB/s algbw")


if __name__ == "__main__":
    main()
