"""SPMD worker for the multi-GPU test battery (launched by torchrun from
tests/test_gpu_multi.py, one rank per GPU over RCCL/xGMI).

Every rank runs the identical battery, asserting forward values and
closed-form adjoints per the reference's SPMD test style
(reference tests/test_collectives.py, tests/test_nonblocking.py), on
cuda:LOCAL_RANK when GPUs are present. With CUDA unavailable the same
battery runs on CPU/gloo — that is the launch-path dry-run the CPU CI
exercises, so the only untested delta on a GPU box is the RCCL transport
itself.

Run directly:
  python -m torch.distributed.run --standalone --local-addr 127.0.0.1 \
      --nproc-per-node N tests/gpu_spmd_worker.py
"""

import os
import sys

import torch
import torch.distributed as dist


def log(rank, msg):
    if rank == 0:
        print(f"[gpu_spmd] {msg}", flush=True)


def section_allreduce(m, comm, device, dtype):
    rank, world = comm.rank, comm.size
    t = torch.full((1 << 16,), float(rank + 1), device=device,
                   dtype=dtype).requires_grad_()
    r = comm.Allreduce(t, m.MPI_SUM)
    want = world * (world + 1) / 2
    assert (r.detach().float() == want).all(), (dtype, r[0].item(), want)
    # closed-form adjoint (reference tests/test_collectives.py:8-12)
    r.backward(torch.ones_like(r))
    assert (t.grad.float() == world).all()
    # MAX/MIN forward values
    assert (comm.Allreduce(t.detach(), m.MPI_MAX).float() == world).all()
    assert (comm.Allreduce(t.detach(), m.MPI_MIN).float() == 1).all()


def section_bcast_reduce(m, comm, device):
    rank, world = comm.rank, comm.size
    t = torch.full((257,), float(rank), device=device).requires_grad_()
    b = comm.Bcast_(t, 1 % world)
    assert (b.detach() == 1 % world).all()
    b.backward(torch.full_like(b, 2.0))
    # adjoint of bcast is reduce-to-root: root sees world*2, others 0
    want = 2.0 * world if rank == 1 % world else 0.0
    assert (t.grad == want).all(), (rank, t.grad[0].item(), want)

    u = torch.full((63,), 1.0, device=device).requires_grad_()
    rr = comm.Reduce_(u, m.MPI_SUM, 0)
    if rank == 0:
        assert (rr.detach() == world).all()
    else:
        assert (rr.detach() == 0).all()
    rr.sum().backward()
    # adjoint of reduce is bcast: every rank gets root's seed (= ones)
    assert (u.grad == 1.0).all()


def section_gather_scatter(m, comm, device):
    rank, world = comm.rank, comm.size
    # variable axis sizes: rank r contributes r+2 slices
    my = rank + 2
    x = torch.full((3, my, 2), float(rank), device=device).requires_grad_()
    g = comm.Gather(x, 1, 0)
    total = sum(r + 2 for r in range(world))
    if rank == 0:
        assert g.shape == (3, total, 2)
        off = 0
        for r in range(world):
            assert (g[:, off : off + r + 2] == r).all()
            off += r + 2
    else:
        assert g.size(1) == 0
    # Scatter∘Gather identity (reference tests/test_collectives.py:92-100)
    back = comm.Scatter(g, 1, my, 0)
    assert (back.detach() == x.detach()).all()
    back.sum().backward()
    assert (x.grad == 1.0).all()


def section_allgather_adjoint(m, comm, device):
    rank, world = comm.rank, comm.size
    my = rank + 1
    x = torch.full((2, my), float(rank + 1), device=device).requires_grad_()
    ag = comm.Allgather(x, 1)
    total = world * (world + 1) // 2
    assert ag.shape == (2, total)
    # NON-uniform gradient seed: catches the reference's wrong-root adjoint
    # bug (reference csrc/extension.cpp:626-628) — grad for my slice must
    # be the SUM over ranks of their seed at my slice positions.
    seed = torch.full_like(ag, float(rank + 1))
    ag.backward(seed)
    want = sum(r + 1 for r in range(world))
    assert (x.grad == want).all(), (rank, x.grad.flatten()[0].item(), want)


def section_alltoall(m, comm, device):
    rank, world = comm.rank, comm.size
    # Alltoall ∘ Alltoall = identity (reference tests/test_collectives.py:137)
    x = torch.rand(world * 2, 5, device=device).requires_grad_()
    y = comm.Alltoall(x, 1, 0, 2)
    z = comm.Alltoall(y, 0, 1, 5)
    assert (z.detach() == x.detach()).all()
    z.sum().backward()
    assert (x.grad == 1.0).all()
    # same-axis repartition with per-rank counts (reference :127-135)
    n_old = rank + 1
    n_new = world - rank
    a = torch.full((n_old, 3), float(rank), device=device)
    b = comm.Alltoall(a, 0, 0, n_new)
    assert b.shape == (n_new, 3)
    # global axis = blocks of sizes 1..world stamped by owner rank
    bounds = []
    off = 0
    for r in range(world):
        bounds.append((off, off + r + 1))
        off += r + 1
    my_lo = sum(world - r for r in range(rank))
    for i in range(n_new):
        gpos = my_lo + i
        owner = next(r for r, (lo, hi) in enumerate(bounds) if lo <= gpos < hi)
        assert (b[i] == owner).all()


def section_ring_large(m, comm, device, nbytes):
    # the deadlock scenario the deferred grouped p2p exists for: every rank
    # Isends before Irecving, payload far beyond RCCL's internal buffering
    rank, world = comm.rank, comm.size
    n = nbytes // 4
    t = torch.full((n,), float(rank), device=device).requires_grad_()
    req = comm.Isend(t, (rank + 1) % world, 0)
    req2 = comm.Irecv(
        m.JoinDummies(torch.empty_like(t), [req.dummy]),
        (rank + world - 1) % world, 0)
    res = comm.Wait(m.JoinDummiesHandle(req, [req2.dummy]))
    res2 = comm.Wait(m.JoinDummiesHandle(req2, [res]))
    assert (res2.detach() == (rank + world - 1) % world).all()
    (res2 * rank).sum().backward()
    # gradient routed through the REVERSED ring
    assert (t.grad == (rank + 1) % world).all()


def section_isend_recv_orderings(m, comm, device):
    rank, world = comm.rank, comm.size
    n = 10_000
    # Isend then blocking Recv (reference tests/test_nonblocking.py:19-25)
    tmp = torch.rand(n, device=device).requires_grad_()
    req = comm.Isend(tmp, (rank + 1) % world, 0)
    res = comm.Recv(
        m.JoinDummies(torch.empty_like(tmp), [req.dummy]),
        (rank + world - 1) % world, 0)
    res2 = comm.Wait(m.JoinDummiesHandle(req, [res]))
    (m.JoinDummies(res, [res2]) * rank).sum().backward()
    assert (tmp.grad == (rank + 1) % world).all()
    # Irecv then blocking Send (reference :28-34)
    tmp2 = torch.rand(n, device=device).requires_grad_()
    req = comm.Irecv(
        m.JoinDummies(torch.empty_like(tmp2), [tmp2]),
        (rank + world - 1) % world, 1)
    res = comm.Send(tmp2, (rank + 1) % world, 1)
    res2 = comm.Wait(m.JoinDummiesHandle(req, [res]))
    (res2 * rank).sum().backward()
    assert (tmp2.grad == (rank + 1) % world).all()


def section_iallreduce_overlap(m, comm, device):
    rank, world = comm.rank, comm.size
    a = torch.full((1 << 20,), float(rank + 1), device=device)
    h = comm.Iallreduce(a, m.MPI_SUM)
    # overlapped local work
    b = torch.rand(512, 512, device=device)
    c = b @ b
    r = comm.Wait(h)
    assert (r == world * (world + 1) / 2).all()
    assert c.shape == (512, 512)
    h2 = comm.Ireducescatter(
        torch.full((world * 128,), float(rank + 1), device=device), m.MPI_SUM)
    r2 = comm.Wait(h2)
    assert r2.numel() == 128 and (r2 == world * (world + 1) / 2).all()


def section_fp8_bitwise_pairloc(m, comm, device):
    rank, world = comm.rank, comm.size
    # fp8 hierarchical allreduce: fp32-accumulated, quantized once.
    # Non-divisible size exercises the variable-count allgather tail.
    n = (1 << 14) + 5
    base = torch.randn(n, device=device)  # rank-dependent
    t8 = (base * 0.25).to(torch.float8_e4m3fn)
    got = comm.Allreduce(t8, m.MPI_SUM)
    # reference: sequential fp32 sum over ranks of the QUANTIZED inputs
    allf = comm.Allgather(t8.float().reshape(1, n), 0)
    ref = allf[0]
    for r in range(1, world):
        ref = ref + allf[r]
    ref8 = ref.to(torch.float8_e4m3fn)
    assert (got.view(torch.uint8) == ref8.view(torch.uint8)).all()

    # bitwise
    ti = ((torch.arange(n, device=device) * (rank + 3)) % (1 << 20)).to(
        torch.int32)
    alli = comm.Allgather(ti.reshape(1, n), 0)
    for op, fn in ((m.MPI_BAND, torch.bitwise_and),
                   (m.MPI_BOR, torch.bitwise_or),
                   (m.MPI_BXOR, torch.bitwise_xor)):
        got = comm.Allreduce(ti, op)
        ref = alli[0]
        for r in range(1, world):
            ref = fn(ref, alli[r])
        assert (got == ref).all(), op

    # minloc/maxloc with ties
    vals = torch.tensor([float((i + rank) % world) for i in range(33)],
                        device=device)
    locs = torch.full((33,), float(rank * 10), device=device)
    pairs = torch.stack([vals, locs], dim=-1)
    gmin = comm.Allreduce(pairs, m.MPI_MINLOC)
    av = comm.Allgather(vals.reshape(1, -1), 0)
    al = comm.Allgather(locs.reshape(1, -1), 0)
    for i in range(33):
        col = av[:, i]
        mn = col.min()
        wl = al[:, i][col == mn].min()
        assert gmin[i, 0] == mn and gmin[i, 1] == wl, (i, gmin[i])


def section_phased_pipeline(m, comm, device):
    # chunked pack->wire pipelining at world > 1: force K=4 phases and
    # check values + identities + adjoints against closed forms
    rank, world = comm.rank, comm.size
    os.environ["MPI4TORCH_AMD_PIPELINE_MB"] = "0.0001"
    m._C.reload_config()
    try:
        # recv side marshaled (EP/bench layout)
        x = (torch.arange(world * 3 * 7 * 2, dtype=torch.float32,
                          device=device).reshape(world * 3, 7, 2)
             * (rank + 1)).requires_grad_()
        y = comm.Alltoall(x, 1, 0, 3)
        base = torch.arange(world * 3 * 7 * 2, dtype=torch.float32,
                            device=device).reshape(world * 3, 7, 2)
        for r in range(world):
            want = base[rank * 3:(rank + 1) * 3] * (r + 1)
            assert (y.detach()[:, r * 7:(r + 1) * 7] == want).all()
        z = comm.Alltoall(y, 0, 1, 7)
        assert (z.detach() == x.detach()).all()
        z.sum().backward()
        assert (x.grad == 1.0).all()
        # send side marshaled + variable sizes; rank 0 has extent 1, so its
        # LOCAL slab before is 1 while peers marshal — the shared-flag
        # phasing decision (not the local one) must drive every rank
        a = torch.randn(rank + 1, world * 4, 3, device=device).requires_grad_()
        b = comm.Alltoall(a, 0, 1, 4)
        c = comm.Alltoall(b, 1, 0, rank + 1)
        assert torch.allclose(c.detach(), a.detach())
        c.sum().backward()
        assert (a.grad == 1.0).all()
        # phased funnel collectives (Gather/Scatter/Allgather, before>1)
        my = rank + 1
        total = world * (world + 1) // 2
        fx = torch.full((3, my, 2), float(rank), device=device)
        fg = comm.Gather(fx, 1, 0)
        if rank == 0:
            off = 0
            for r in range(world):
                assert (fg[:, off:off + r + 1] == r).all()
                off += r + 1
        fb = comm.Scatter(fg, 1, my, 0)
        assert (fb == fx).all()
        fa = comm.Allgather(fx, 1)
        assert fa.shape == (3, total, 2)
    finally:
        os.environ["MPI4TORCH_AMD_PIPELINE_MB"] = "64"
        m._C.reload_config()


def section_reducescatter(m, comm, device):
    rank, world = comm.rank, comm.size
    x = torch.full((world * 4, 3), 1.0, device=device).requires_grad_()
    r = comm.Reducescatter(x, 0, 4)
    assert r.shape == (4, 3) and (r.detach() == world).all()
    r.sum().backward()
    assert (x.grad == 1.0).all()  # adjoint = allgather of ones


def section_alltoall_pairwise(m, comm, device):
    rank, world = comm.rank, comm.size
    # EP-style dispatch: rank r sends j+1 rows to rank j
    send_counts = [j + 1 for j in range(world)]
    rows = sum(send_counts)
    x = torch.cat([
        torch.full((j + 1, 4), float(rank * 100 + j), device=device)
        for j in range(world)
    ])
    assert x.shape[0] == rows
    x.requires_grad_()
    out = comm.AlltoallPairwise(x, 0, send_counts, [])
    # I receive rank+1 rows from every peer, stamped r*100 + my rank
    assert out.shape[0] == world * (rank + 1)
    off = 0
    for r in range(world):
        blk = out[off : off + rank + 1]
        assert (blk.detach() == r * 100 + rank).all()
        off += rank + 1
    out.sum().backward()
    assert (x.grad == 1.0).all()


def section_comm_split(m, comm, device):
    rank, world = comm.rank, comm.size
    if world < 2:
        return
    color = rank % 2
    sub = m.comm_split(comm, color)
    members = [r for r in range(world) if r % 2 == color]
    assert sub.size == len(members) and sub.rank == members.index(rank)
    t = torch.full((16,), float(rank), device=device)
    s = sub.Allreduce(t, m.MPI_SUM)
    assert (s == sum(members)).all()


def section_hipgraph_multirank(m, comm, device):
    # OPT-IN (M4A_TEST_HIPGRAPH=1): multi-rank hipGraph capture+replay of
    # blocking collectives. Every rank captures the identical sequence
    # (the ProcessGroupNCCL pattern); kept out of the default battery
    # until validated on hardware (TODO.md item 3).
    if os.environ.get("M4A_TEST_HIPGRAPH") != "1" or device.type != "cuda":
        return
    rank, world = comm.rank, comm.size
    static_in = torch.rand(1 << 16, device=device)
    x0 = static_in.clone()
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(3):
            comm.Allreduce(static_in, m.MPI_SUM)
    torch.cuda.current_stream().wait_stream(s)
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        static_out = comm.Allreduce(static_in, m.MPI_SUM)
    for i in range(3):
        static_in.copy_(x0 * (i + 1))
        g.replay()
        torch.cuda.synchronize()
        assert torch.allclose(static_out, x0 * (i + 1) * world)


def section_hybrid_tp_dp(m, comm, device):
    # TP x DP over comm_split: TensorParallelMLP (tp) inside DDP (dp),
    # gradients vs the dense reference averaged over DP shards
    from mpi4torch_amd.parallel import DistributedDataParallel
    from mpi4torch_amd.parallel.tp import TensorParallelMLP

    rank, world = comm.rank, comm.size
    if world < 4 or world % 2 != 0:
        return
    tp_size = 2
    dp_size = world // tp_size
    tp_color, dp_color = rank // tp_size, rank % tp_size
    tp_comm = m.comm_split(comm, tp_color)
    dp_comm = m.comm_split(comm, dp_color)

    torch.manual_seed(7)
    d, h, b = 16, 32, 8
    w1 = torch.randn(h, d, dtype=torch.double, device=device)
    b1 = torch.randn(h, dtype=torch.double, device=device)
    w2 = torch.randn(d, h, dtype=torch.double, device=device)
    mlp = TensorParallelMLP(d, h, comm=tp_comm,
                            activation=torch.relu).double().to(device)
    mlp.up.shard_from_full(w1, b1)
    mlp.down.shard_from_full(w2)
    with torch.no_grad():
        mlp.down.bias.zero_()
    model = DistributedDataParallel(mlp, comm=dp_comm, bucket_cap_mb=0)
    torch.manual_seed(1000 + tp_color)
    x = torch.randn(b, d, dtype=torch.double, device=device)
    (model(x) ** 2).sum().backward()
    model.finish_gradient_sync()

    grads = []
    for s in range(dp_size):
        torch.manual_seed(1000 + s)
        xs = torch.randn(b, d, dtype=torch.double, device=device)
        w1r = w1.clone().requires_grad_()
        b1r = b1.clone().requires_grad_()
        w2r = w2.clone().requires_grad_()
        b2r = torch.zeros(d, dtype=torch.double, device=device,
                          requires_grad=True)
        ((torch.relu(xs @ w1r.t() + b1r) @ w2r.t() + b2r) ** 2
         ).sum().backward()
        grads.append((w1r.grad, b1r.grad, w2r.grad, b2r.grad))
    g1 = sum(g[0] for g in grads) / dp_size
    gb = sum(g[1] for g in grads) / dp_size
    g2 = sum(g[2] for g in grads) / dp_size
    gb2 = sum(g[3] for g in grads) / dp_size
    tpr = tp_comm.rank
    lo, hi = tpr * mlp.up.out_local, (tpr + 1) * mlp.up.out_local
    lo2, hi2 = tpr * mlp.down.in_local, (tpr + 1) * mlp.down.in_local
    assert torch.allclose(mlp.up.linear.weight.grad, g1[lo:hi], atol=1e-9)
    assert torch.allclose(mlp.up.linear.bias.grad, gb[lo:hi], atol=1e-9)
    assert torch.allclose(mlp.down.linear.weight.grad, g2[:, lo2:hi2],
                          atol=1e-9)
    assert torch.allclose(mlp.down.bias.grad, gb2, atol=1e-9)


def section_ddp(m, comm, device):
    from mpi4torch_amd.parallel import DistributedDataParallel

    rank, world = comm.rank, comm.size
    torch.manual_seed(1000 + rank)  # deliberately different init
    net = torch.nn.Sequential(torch.nn.Linear(32, 64), torch.nn.GELU(),
                              torch.nn.Linear(64, 8)).to(device)
    model = DistributedDataParallel(net, bucket_cap_mb=1)
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    torch.manual_seed(7 + rank)
    for _ in range(3):
        x = torch.randn(16, 32, device=device)
        loss = model(x).pow(2).mean()
        opt.zero_grad()
        loss.backward()
        model.finish_gradient_sync()
        opt.step()
    for p in net.parameters():
        ref = comm.Bcast_(p.data.clone(), 0)
        assert torch.allclose(p.data, ref, atol=1e-6), "replicas diverged"


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)
    else:
        device = torch.device("cpu")

    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import mpi4torch_amd as m

    m.init()
    comm = m.COMM_WORLD
    assert comm.size == world and comm.rank == rank

    log(rank, f"world={world} device={device} — allreduce")
    section_allreduce(m, comm, device, torch.float32)
    section_allreduce(m, comm, device,
                      torch.bfloat16 if use_cuda else torch.float64)
    log(rank, "bcast/reduce adjoints")
    section_bcast_reduce(m, comm, device)
    log(rank, "gather/scatter identity")
    section_gather_scatter(m, comm, device)
    log(rank, "allgather reduce-scatter adjoint")
    section_allgather_adjoint(m, comm, device)
    log(rank, "alltoall identities")
    section_alltoall(m, comm, device)
    log(rank, "large ring exchange (grouped p2p)")
    section_ring_large(m, comm, device,
                       nbytes=(64 << 20) if use_cuda else (4 << 20))
    log(rank, "isend/recv orderings")
    section_isend_recv_orderings(m, comm, device)
    log(rank, "iallreduce overlap")
    section_iallreduce_overlap(m, comm, device)
    log(rank, "fp8 / bitwise / pairloc")
    section_fp8_bitwise_pairloc(m, comm, device)
    log(rank, "phased pipelining (forced K=4)")
    section_phased_pipeline(m, comm, device)
    log(rank, "reducescatter")
    section_reducescatter(m, comm, device)
    log(rank, "alltoall pairwise (EP)")
    section_alltoall_pairwise(m, comm, device)
    log(rank, "comm_split subgroups")
    section_comm_split(m, comm, device)
    log(rank, "hybrid TP x DP")
    section_hybrid_tp_dp(m, comm, device)
    section_hipgraph_multirank(m, comm, device)  # opt-in, see TODO.md
    log(rank, "DDP end-to-end")
    section_ddp(m, comm, device)

    if use_cuda:
        torch.cuda.synchronize()
    dist.barrier()
    log(rank, "ALL SECTIONS PASSED")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
