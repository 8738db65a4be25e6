"""GPU (MI355X) tests: RCCL world-of-one semantics, CDNA4 kernel numerics,
stream-ordered p2p, host-staging toggle.

These run on a single GPU (gpurun gives one MI355X); multi-rank GPU behavior
is covered by the CPU gloo SPMD tests (same op-layer code paths) and by the
driver's round-end multi-GPU bench. Kernel numerics compare against plain
PyTorch fp32/reference ops per the test policy.
"""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def world1():
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29551")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    import torch.distributed as dist

    if not dist.is_initialized():
        dist.init_process_group("gloo", rank=0, world_size=1)
    import mpi4torch_amd as m

    m.init()
    return m


def test_native_extension_loaded(world1):
    m = world1
    # the in-tree extension must be the loaded one (no site-packages copy)
    assert m._C.__file__.endswith("mpi4torch_amd/_C.so")
    assert m._C._rccl_version() > 0


def test_allreduce_gpu_fwd_bwd(world1):
    m = world1
    comm = m.COMM_WORLD
    for dtype in (torch.float32, torch.bfloat16, torch.float16, torch.float64):
        t = torch.rand(1 << 20, dtype=dtype, device="cuda").requires_grad_()
        r = comm.Allreduce(t, m.MPI_SUM)
        assert r.is_cuda and r.dtype == dtype
        torch.testing.assert_close(r, t.detach())  # world of one: identity
        r.backward(torch.ones_like(r))
        torch.testing.assert_close(t.grad, torch.ones_like(t))
        t.grad = None


def test_allreduce_gpu_int_and_fp8(world1):
    m = world1
    comm = m.COMM_WORLD
    ti = torch.randint(0, 100, (4096,), device="cuda", dtype=torch.int32)
    torch.testing.assert_close(comm.Allreduce(ti, m.MPI_SUM), ti)
    ts = ti.to(torch.int16)
    torch.testing.assert_close(comm.Allreduce(ts, m.MPI_SUM), ts)
    t8 = torch.rand(4096, device="cuda").to(torch.float8_e4m3fn)
    r8 = comm.Allreduce(t8, m.MPI_SUM)
    assert r8.dtype == torch.float8_e4m3fn
    torch.testing.assert_close(r8.float(), t8.float())


def test_collectives_gpu_world1(world1):
    m = world1
    comm = m.COMM_WORLD
    x = torch.rand(2, 5, 4, 3, device="cuda", dtype=torch.float32)
    torch.testing.assert_close(comm.Gather(x, 2, 0), x)
    torch.testing.assert_close(comm.Allgather(x, 2), x)
    torch.testing.assert_close(comm.Scatter(x, 2, 4, 0), x)
    torch.testing.assert_close(comm.Alltoall(x, 1, 2, 4), x)
    y = x.clone().requires_grad_()
    r = comm.Bcast_(y, 0)
    torch.testing.assert_close(r, x)
    r.sum().backward()
    torch.testing.assert_close(y.grad, torch.ones_like(x))


def test_pack_kernel_roundtrip(world1):
    m = world1
    # batched CDNA4 slab copy vs identity, several geometries:
    # middle axis (strided, 16B-aligned), odd tail (1B path), leading axis
    cases = [
        ((2, 5, 12, 2, 4), 2, [3, 4, 5], torch.float32),
        ((3, 7, 5), 1, [1, 2, 4], torch.bfloat16),   # unaligned rows
        ((64, 33), 0, [10, 20, 34], torch.float32),
        ((2, 9, 3), 1, [9], torch.float8_e4m3fn),    # 1-byte dtype
        ((8, 1024, 256), 1, [256, 512, 256], torch.bfloat16),  # big, vector path
        # odd phases: exercise the byte-window (shift) path
        ((8, 65537, 511), 1, [8192, 30000, 27345], torch.bfloat16),
        ((3, 1001, 7), 1, [137, 500, 364], torch.float8_e4m3fn),
        ((1, 999983), 1, [1, 2, 999980], torch.bfloat16),
    ]
    for shape, axis, counts, dtype in cases:
        x = (torch.randn(shape, device="cuda", dtype=torch.float32)).to(dtype)
        out = m._C._pack_roundtrip(x, axis, counts)
        torch.cuda.synchronize()
        assert (out.view(torch.uint8) == x.view(torch.uint8)).all(), (
            shape, axis, counts, dtype)


def test_pack_kernel_matches_cpu(world1):
    m = world1
    x = torch.randn(4, 30, 7, device="cuda")
    out_gpu = m._C._pack_roundtrip(x, 1, [7, 11, 12])
    out_cpu = m._C._pack_roundtrip(x.cpu(), 1, [7, 11, 12])
    torch.cuda.synchronize()
    torch.testing.assert_close(out_gpu.cpu(), out_cpu)


def test_bitwise_reduce_kernel(world1):
    m = world1
    for dtype in (torch.int32, torch.int64, torch.uint8):
        stacked = torch.randint(0, 1 << 16, (5, 4097), device="cuda").to(dtype)
        for op, fn in ((0, torch.bitwise_and), (1, torch.bitwise_or),
                       (2, torch.bitwise_xor)):
            out = m._C._bitwise_reduce(stacked, op)
            ref = stacked[0]
            for i in range(1, 5):
                ref = fn(ref, stacked[i])
            torch.cuda.synchronize()
            assert (out == ref).all(), (dtype, op)


def test_self_ring_gpu(world1):
    m = world1
    comm = m.COMM_WORLD
    t = torch.rand(1 << 20, device="cuda").requires_grad_()
    req = comm.Isend(t, 0, 0)
    req2 = comm.Irecv(
        m.JoinDummies(torch.empty_like(t), [req.dummy]), 0, 0
    )
    res = comm.Wait(m.JoinDummiesHandle(req, [req2.dummy]))
    res2 = comm.Wait(m.JoinDummiesHandle(req2, [res]))
    torch.testing.assert_close(res2, t.detach())
    (res2 * 2).sum().backward()
    torch.testing.assert_close(t.grad, 2 * torch.ones_like(t))


def test_host_staging_toggle(world1):
    m = world1
    comm = m.COMM_WORLD
    t = torch.rand(4096, device="cuda").requires_grad_()
    try:
        m.force_host_staging(True)
        assert m._C.host_staging_forced()
        r = comm.Allreduce(t, m.MPI_SUM)
        assert r.is_cuda
        torch.testing.assert_close(r, t.detach())
        r.sum().backward()
        torch.testing.assert_close(t.grad, torch.ones_like(t))
        # staged variants of the axis/extension ops
        x = torch.rand(2, 6, 3, device="cuda")
        torch.testing.assert_close(comm.Gather(x, 1, 0), x)
        torch.testing.assert_close(comm.Alltoallv(x, 1, 1, [6], [6]), x)
        torch.testing.assert_close(comm.Reducescatter(x, 1, 6), x)
        torch.testing.assert_close(
            comm.AlltoallPairwise(x, 1, [6], []), x)
    finally:
        m.force_host_staging(False)


def test_allreduce_numerics_vs_fp32(world1):
    m = world1
    comm = m.COMM_WORLD
    # bf16 path vs plain fp32 reference of the same op (world-1: identity)
    x32 = torch.randn(1 << 16, device="cuda")
    r = comm.Allreduce(x32.to(torch.bfloat16), m.MPI_SUM)
    torch.testing.assert_close(
        r.float(), x32.to(torch.bfloat16).float(), rtol=0, atol=0
    )


def test_fp8_reduce_kernel(world1):
    m = world1
    # fused fp8 reduce (fp32 accumulation) vs plain PyTorch fp32 reference
    for enc in (torch.float8_e4m3fn, torch.float8_e5m2):
        for n in (4096, 4104, 4099):  # 16B-vector, 8-not-16, scalar paths
            stacked = (torch.randn(5, n, device="cuda") * 0.25).to(enc)
            ref_in = stacked.float()
            cases = [
                (0, lambda a, b: a + b),
                (1, lambda a, b: a * b),
                (2, torch.minimum),
                (3, torch.maximum),
            ]
            for op, fn in cases:
                out = m._C._fp8_reduce(stacked, op)
                ref = ref_in[0]
                for i in range(1, 5):
                    ref = fn(ref, ref_in[i])
                ref8 = ref.to(enc)
                torch.cuda.synchronize()
                assert (out.view(torch.uint8) == ref8.view(torch.uint8)).all(), (
                    enc, n, op)


def test_fp8_allreduce_path(world1):
    m = world1
    comm = m.COMM_WORLD
    # world-1: fused path must still be an exact identity
    for enc in (torch.float8_e4m3fn, torch.float8_e5m2):
        t = (torch.randn(1 << 16, device="cuda") * 0.5).to(enc)
        r = comm.Allreduce(t, m.MPI_SUM)
        assert (r.view(torch.uint8) == t.view(torch.uint8)).all()
        r2 = comm.Allreduce(t, m.MPI_MAX)
        assert (r2.view(torch.uint8) == t.view(torch.uint8)).all()


def test_pairloc_reduce_kernel(world1):
    m = world1
    # CDNA4 pair arg-reduce vs the torch-composite reference (same function
    # the CPU transport path uses)
    for dtype in (torch.float32, torch.float64, torch.float16,
                  torch.bfloat16, torch.int32, torch.int64, torch.int16,
                  torch.int8, torch.uint8):
        for n in (1024, 1027):
            if dtype.is_floating_point:
                vals = (torch.randn(6, n) * 4).to(dtype)
            else:
                vals = torch.randint(0, 50, (6, n)).to(dtype)
            locs = torch.randint(0, 90, (6, n)).to(dtype)
            stacked = torch.stack([vals, locs], dim=-1)
            for op in (0, 1):  # minloc, maxloc
                ref = m._C._pairloc_reduce(stacked, op)  # CPU composite
                got = m._C._pairloc_reduce(stacked.cuda(), op).cpu()
                torch.cuda.synchronize()
                assert (got == ref).all(), (dtype, n, op)


def test_pairloc_allreduce_full_path(world1):
    m = world1
    comm = m.COMM_WORLD
    # world-1 force_full_path drives the whole hierarchical machinery
    # (self block exchange -> arg-reduce kernel -> in-place allgather)
    m._C.force_full_path(True)
    try:
        pairs = torch.stack(
            [torch.randn(4097, device="cuda"),
             torch.randint(0, 100, (4097,), device="cuda").float()], dim=-1)
        for op in (m.MPI_MINLOC, m.MPI_MAXLOC):
            r = comm.Allreduce(pairs, op)
            torch.testing.assert_close(r, pairs)  # P=1: identity
    finally:
        m._C.force_full_path(False)


def test_hierarchical_fp8_bitwise_full_path(world1):
    m = world1
    comm = m.COMM_WORLD
    # the hierarchical (exchange + fused kernel + in-place allgather) paths
    # at P=1 must be exact identities, including non-divisible sizes
    m._C.force_full_path(True)
    try:
        for n in (1 << 16, (1 << 16) + 3):
            t8 = (torch.randn(n, device="cuda") * 0.5).to(torch.float8_e4m3fn)
            r8 = comm.Allreduce(t8, m.MPI_SUM)
            assert (r8.view(torch.uint8) == t8.view(torch.uint8)).all()
            ti = torch.randint(0, 1 << 30, (n,), device="cuda",
                               dtype=torch.int32)
            for op in (m.MPI_BAND, m.MPI_BOR, m.MPI_BXOR):
                ri = comm.Allreduce(ti, op)
                assert (ri == ti).all()
    finally:
        m._C.force_full_path(False)


def test_phased_pipeline_gpu(world1):
    m = world1
    comm = m.COMM_WORLD
    # force K=4 phases and the full path at world 1: phased slab pack ->
    # grouped iexchange (self) -> phased unpack, on the CDNA4 kernels
    os.environ["MPI4TORCH_AMD_PIPELINE_MB"] = "0.0001"
    m._C.reload_config()
    m._C.force_full_path(True)
    try:
        x = torch.rand(6, 8, 4, device="cuda", dtype=torch.bfloat16)
        y = comm.Alltoall(x, 1, 0, 6)
        torch.testing.assert_close(y.float(), x.float())
        z = torch.rand(3, 10, 2, device="cuda").requires_grad_()
        w = comm.Alltoall(z, 2, 1, 10)
        torch.testing.assert_close(w, z.detach())
        w.sum().backward()
        torch.testing.assert_close(z.grad, torch.ones_like(z))
        # funnel collectives through the phased path (before > 1)
        f = torch.rand(5, 6, 3, device="cuda", dtype=torch.bfloat16)
        torch.testing.assert_close(comm.Gather(f, 1, 0).float(), f.float())
        torch.testing.assert_close(comm.Allgather(f, 1).float(), f.float())
        torch.testing.assert_close(
            comm.Scatter(f, 1, 6, 0).float(), f.float())
        fg = torch.rand(4, 7, 2, device="cuda").requires_grad_()
        comm.Gather(fg, 1, 0).sum().backward()
        torch.testing.assert_close(fg.grad, torch.ones_like(fg))
    finally:
        os.environ["MPI4TORCH_AMD_PIPELINE_MB"] = "64"
        m._C.reload_config()
        m._C.force_full_path(False)


def test_hipgraph_capture(world1):
    m = world1
    comm = m.COMM_WORLD
    static_in = torch.rand(1 << 18, device="cuda")
    x0 = static_in.clone()
    # warmup on a side stream (torch graph-capture protocol)
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(3):
            comm.Allreduce(static_in, m.MPI_SUM)
    torch.cuda.current_stream().wait_stream(s)

    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        static_out = comm.Allreduce(static_in, m.MPI_SUM)
        static_out2 = comm.Gather(static_out, 0, 0)
    for i in range(3):
        static_in.copy_(x0 * (i + 1))
        g.replay()
        torch.cuda.synchronize()
        torch.testing.assert_close(static_out2, x0 * (i + 1))


def test_soak_no_leaks(world1):
    m = world1
    comm = m.COMM_WORLD
    torch.cuda.synchronize()
    torch.cuda.empty_cache()
    base = torch.cuda.memory_allocated()
    x = torch.rand(2, 64, 8, device="cuda")
    for i in range(200):
        t = torch.rand(1 << 14, device="cuda").requires_grad_()
        r = comm.Allreduce(t, m.MPI_SUM)
        r.backward(torch.ones_like(r))
        comm.Gather(x, 1, 0)
        comm.Alltoall(x, 0, 1, 64)
        h = comm.Isend(x, 0, i)
        h2 = comm.Irecv(torch.empty_like(x), 0, i)
        comm.Wait(h2)
        comm.Wait(h)
        hh = comm.Iallreduce(x.reshape(-1), m.MPI_SUM)
        comm.Wait(hh)
    torch.cuda.synchronize()
    grown = torch.cuda.memory_allocated() - base
    # event pool + request table must not accumulate device memory
    assert grown < 32 * 1024 * 1024, f"leaked {grown} bytes over 200 iters"


def test_zero_numel_ops(world1):
    m = world1
    comm = m.COMM_WORLD
    e = torch.empty(0, device="cuda")
    assert comm.Allreduce(e, m.MPI_SUM).numel() == 0
    assert comm.Bcast_(e.clone(), 0).numel() == 0


def test_full_path_axis_collectives(world1):
    m = world1
    comm = m.COMM_WORLD
    # run the real pack -> exchange(self) -> unpack pipeline at world 1
    m._C.force_full_path(True)
    try:
        x = torch.rand(2, 5, 12, 3, device="cuda", dtype=torch.bfloat16)
        torch.testing.assert_close(comm.Gather(x, 2, 0), x)
        torch.testing.assert_close(comm.Allgather(x, 2), x)
        torch.testing.assert_close(comm.Scatter(x, 2, 12, 0), x)
        torch.testing.assert_close(comm.Alltoall(x, 1, 2, 12), x)
        torch.testing.assert_close(comm.Alltoall(x, 2, 2, 12), x)  # same axis
        t = torch.rand(1 << 16, device="cuda").requires_grad_()
        r = comm.Allreduce(t, m.MPI_SUM)  # true ncclAllReduce at world 1
        torch.testing.assert_close(r, t.detach())
        r.backward(torch.ones_like(r))
        torch.testing.assert_close(t.grad, torch.ones_like(t))
        # autograd through the full gather/scatter pipeline
        y = torch.rand(2, 7, 3, device="cuda", dtype=torch.float32).requires_grad_()
        comm.Gather(y, 1, 0).sum().backward()
        torch.testing.assert_close(y.grad, torch.ones_like(y))
    finally:
        m._C.force_full_path(False)


def test_alltoall_pairwise_gpu(world1):
    m = world1
    comm = m.COMM_WORLD
    x = torch.rand(2, 9, 3, device="cuda", dtype=torch.bfloat16)
    out = comm.AlltoallPairwise(x, 1, [9], [])
    torch.testing.assert_close(out.float(), x.float())
    m._C.force_full_path(True)
    try:
        y = torch.rand(4, 12, device="cuda").requires_grad_()
        out2 = comm.AlltoallPairwise(y, 0, [4], [4])
        torch.testing.assert_close(out2, y.detach())
        out2.sum().backward()
        torch.testing.assert_close(y.grad, torch.ones_like(y))
    finally:
        m._C.force_full_path(False)


def test_reducescatter_gpu(world1):
    m = world1
    comm = m.COMM_WORLD
    x = torch.rand(4, 16, device="cuda", dtype=torch.bfloat16)
    out = comm.Reducescatter(x, 1, 16)
    torch.testing.assert_close(out.float(), x.float())
    m._C.force_full_path(True)
    try:
        y = torch.rand(2, 8, 3, device="cuda").requires_grad_()
        out2 = comm.Reducescatter(y, 1, 8)  # real ncclReduceScatter, P=1
        torch.testing.assert_close(out2, y.detach())
        out2.sum().backward()
        torch.testing.assert_close(y.grad, torch.ones_like(y))
    finally:
        m._C.force_full_path(False)


def test_gpu_full_path_fuzz(world1):
    """Random geometries through the full pack->exchange->unpack pipeline
    (world-1 self-loops) vs the CPU implementation of the same ops —
    stresses every slab-kernel path (granules, phases, heads/tails)."""
    import random

    m = world1
    comm = m.COMM_WORLD
    rng = random.Random(777)
    m._C.force_full_path(True)
    try:
        for it in range(30):
            ndim = rng.randint(1, 4)
            shape = [rng.randint(1, 9) for _ in range(ndim)]
            axis = rng.randint(0, ndim - 1)
            shape[axis] = rng.randint(1, 40)
            dtype = rng.choice(
                [torch.float32, torch.bfloat16, torch.float8_e4m3fn])
            x32 = torch.randn(shape, device="cuda")
            x = x32.to(dtype)
            n = shape[axis]

            g = comm.Gather(x, axis, 0)
            assert (g.view(torch.uint8) == x.view(torch.uint8)).all(), (
                "gather", it, shape, axis, dtype)
            ag = comm.Allgather(x, axis)
            assert (ag.view(torch.uint8) == x.view(torch.uint8)).all(), (
                "allgather", it, shape, axis, dtype)
            sc = comm.Scatter(x, axis, n, 0)
            assert (sc.view(torch.uint8) == x.view(torch.uint8)).all(), (
                "scatter", it, shape, axis, dtype)
            a2a = comm.Alltoall(x, axis, axis, n)
            assert (a2a.view(torch.uint8) == x.view(torch.uint8)).all(), (
                "alltoall", it, shape, axis, dtype)
            pw = comm.AlltoallPairwise(x, axis, [n], [n])
            assert (pw.view(torch.uint8) == x.view(torch.uint8)).all(), (
                "pairwise", it, shape, axis, dtype)
    finally:
        m._C.force_full_path(False)


def test_ireducescatter_gpu(world1):
    m = world1
    comm = m.COMM_WORLD
    t = torch.rand(1 << 16, device="cuda")
    h = comm.Ireducescatter(t, m.MPI_SUM)
    out = comm.Wait(h)
    torch.testing.assert_close(out, t)  # world of one


def test_zero_family_gpu(world1):
    # ZeRO-1/2/3 on CUDA at world 1 must match a plain local Adam exactly
    # (exercises CUDA storage-resize, Ireducescatter and allgather paths)
    from mpi4torch_amd.parallel import (FullyShardedDataParallel,
                                        ShardedDataParallel,
                                        ZeroRedundancyOptimizer)

    def make_net():
        torch.manual_seed(11)
        return torch.nn.Sequential(
            torch.nn.Linear(16, 32), torch.nn.Tanh(), torch.nn.Linear(32, 8)
        ).double().cuda()

    def data(step):
        torch.manual_seed(100 + step)
        return torch.randn(6, 16, dtype=torch.double, device="cuda")

    ref = make_net()
    ref_opt = torch.optim.Adam(ref.parameters(), lr=0.02)
    for s in range(3):
        ref_opt.zero_grad()
        ref(data(s)).square().sum().backward()
        ref_opt.step()

    # ZeRO-1
    net1 = make_net()
    z1 = ZeroRedundancyOptimizer(net1.parameters(), torch.optim.Adam, lr=0.02)
    for s in range(3):
        z1.zero_grad()
        net1(data(s)).square().sum().backward()
        z1.step()
    for p, q in zip(net1.parameters(), ref.parameters()):
        torch.testing.assert_close(p, q, rtol=0, atol=1e-12)

    # ZeRO-2
    net2 = make_net()
    m2 = ShardedDataParallel(net2, torch.optim.Adam, bucket_cap_mb=1, lr=0.02)
    for s in range(3):
        m2.zero_grad()
        m2(data(s)).square().sum().backward()
        m2.step()
    for p, q in zip(net2.parameters(), ref.parameters()):
        torch.testing.assert_close(p, q, rtol=0, atol=1e-12)

    # ZeRO-3 (FSDP) — CUDA storage resize + materialization
    net3 = make_net()
    m3 = FullyShardedDataParallel(net3, units=[net3[0], net3[2]])
    opt3 = torch.optim.Adam(m3.shard_parameters(), lr=0.02)
    for s in range(3):
        m3.zero_grad()
        m3(data(s)).square().sum().backward()
        m3.finish_backward()
        opt3.step()
        m3.refresh_shards()
        for u in m3._units:
            assert u.flat.untyped_storage().size() == 0  # sharded at rest
    for u in m3._units:
        u.materialize()
    for p, q in zip(net3.parameters(), ref.parameters()):
        torch.testing.assert_close(p, q, rtol=0, atol=1e-12)


def test_nondefault_stream_ordering(world1):
    """Collectives issued from a user (non-default) stream must be ordered
    by the event bracket: producer kernel -> collective -> consumer kernel
    across three different streams, with no host syncs in between."""
    m = world1
    comm = m.COMM_WORLD
    s1 = torch.cuda.Stream()
    s2 = torch.cuda.Stream()
    n = 1 << 20
    with torch.cuda.stream(s1):
        x = torch.ones(n, device="cuda")
        x = x * 3  # producer on s1
        r = comm.Allreduce(x, m.MPI_SUM)  # collective from s1
    s2.wait_stream(s1)
    with torch.cuda.stream(s2):
        y = r + 1  # consumer on s2
    torch.cuda.current_stream().wait_stream(s2)
    torch.cuda.synchronize()
    assert (y == 4).all()

    # p2p from a side stream; Wait issued from the default stream
    with torch.cuda.stream(s1):
        t = torch.full((n,), 7.0, device="cuda")
        h = comm.Isend(t, 0, 1)
        h2 = comm.Irecv(torch.empty(n, device="cuda"), 0, 1)
    torch.cuda.current_stream().wait_stream(s1)
    got = comm.Wait(h2)
    comm.Wait(h)
    torch.cuda.synchronize()
    assert (got == 7).all()


def test_multithreaded_collectives(world1):
    """Concurrent collective issue from several host threads (the autograd
    engine uses its own threads in real training): per-communicator mutex
    + event pool must serialize enqueues without loss or deadlock."""
    import concurrent.futures as cf

    m = world1
    comm = m.COMM_WORLD

    def job(i):
        t = torch.full((4096,), float(i), device="cuda")
        r = comm.Allreduce(t, m.MPI_SUM)
        g = comm.Gather(t, 0, 0)
        torch.cuda.synchronize()
        return bool((r == i).all()) and bool((g == i).all())

    with cf.ThreadPoolExecutor(max_workers=4) as ex:
        results = list(ex.map(job, range(64)))
    assert all(results)
