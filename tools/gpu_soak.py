#!/usr/bin/env python3
"""Extended single-GPU soak: continuous mixed workload (DDP training step,
axis collectives on the full pipeline, self-p2p ring, hipGraph replay)
until --seconds elapse. Watches device memory for leaks and validates
results each iteration. Run under gpurun: exercises the event pool,
request table, staging allocations and stream bracketing under sustained
pressure."""

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--seconds", type=float, default=120)
    args = p.parse_args()

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29471")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    import torch.distributed as dist

    if not dist.is_initialized():
        dist.init_process_group("gloo", rank=0, world_size=1)
    import mpi4torch_amd as m
    from mpi4torch_amd.models.moe import ExpertParallelMoE
    from mpi4torch_amd.parallel import (DistributedDataParallel,
                                        ShardedDataParallel)

    m.init()
    comm = m.COMM_WORLD
    device = torch.device("cuda:0")

    net = torch.nn.Sequential(
        torch.nn.Linear(256, 1024), torch.nn.GELU(), torch.nn.Linear(1024, 64)
    ).to(device)
    model = DistributedDataParallel(net, bucket_cap_mb=4)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-4)
    znet = torch.nn.Sequential(
        torch.nn.Linear(128, 512), torch.nn.GELU(), torch.nn.Linear(512, 32)
    ).to(device)
    zmodel = ShardedDataParallel(znet, torch.optim.AdamW, bucket_cap_mb=1,
                                 lr=1e-4)
    moe = ExpertParallelMoE(64, 4).to(device)
    moe_opt = torch.optim.AdamW(moe.parameters(), lr=1e-4)

    m._C.force_full_path(True)
    torch.cuda.synchronize()
    base_mem = torch.cuda.memory_allocated()
    t0 = time.perf_counter()
    it = 0
    while time.perf_counter() - t0 < args.seconds:
        it += 1
        # DDP step
        x = torch.randn(64, 256, device=device)
        loss = model(x).square().mean()
        opt.zero_grad(set_to_none=True)
        loss.backward()
        model.finish_gradient_sync()
        opt.step()

        # ZeRO-2 step + expert-parallel MoE step
        zloss = zmodel(torch.randn(32, 128, device=device)).square().mean()
        zmodel.zero_grad()
        zloss.backward()
        zmodel.step()
        mloss = moe(torch.randn(48, 64, device=device)).square().mean()
        moe_opt.zero_grad()
        mloss.backward()
        moe_opt.step()

        # axis collectives, full pipeline, with value checks
        t = torch.rand(4, 1000 + (it % 7), 8, device=device)
        assert torch.equal(comm.Allgather(t, 1), t)
        assert torch.equal(comm.Alltoall(t, 1, 1, t.size(1)), t)

        # autograd allreduce + self ring
        g = torch.rand(1 << 18, device=device).requires_grad_()
        r = comm.Allreduce(g, m.MPI_SUM)
        r.backward(torch.ones_like(r))
        assert bool((g.grad == 1).all())
        h = comm.Isend(g.detach(), 0, it)
        h2 = comm.Irecv(torch.empty_like(g), 0, it)
        got = comm.Wait(h2)
        comm.Wait(h)
        assert torch.equal(got, g.detach())

        if it % 50 == 0:
            torch.cuda.synchronize()
            grown = torch.cuda.memory_allocated() - base_mem
            print(f"iter {it}: mem growth {grown/1e6:.1f} MB, "
                  f"{(time.perf_counter()-t0):.0f}s", flush=True)
            assert grown < 256 * 1024 * 1024, "memory leak"
    torch.cuda.synchronize()
    m._C.force_full_path(False)
    print(f"SOAK OK: {it} iterations in {time.perf_counter()-t0:.0f}s, "
          f"final mem growth {(torch.cuda.memory_allocated()-base_mem)/1e6:.1f} MB")


if __name__ == "__main__":
    main()
