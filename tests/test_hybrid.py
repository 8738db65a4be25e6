"""Hybrid parallelism: TP x DP over comm_split sub-communicators.

The production topology no single-communicator test covers: world 4
splits into two tensor-parallel groups ({0,1}, {2,3} — one model replica
each) and two data-parallel groups ({0,2}, {1,3} — pairing corresponding
TP shards). A TensorParallelMLP runs over the TP communicator, wrapped
in DistributedDataParallel over the DP communicator, so BOTH
communicators issue collectives in the same forward/backward. Gradients
must equal the dense single-process reference averaged over the DP data
shards, sliced per TP rank (fp64, 1e-10).
"""

import torch

from spmd import run_spmd


def _hybrid_worker(rank, world, tp_size, h):
    import mpi4torch_amd as m
    from mpi4torch_amd.parallel import DistributedDataParallel
    from mpi4torch_amd.parallel.tp import TensorParallelMLP

    comm = m.COMM_WORLD
    dp_size = world // tp_size
    tp_color = rank // tp_size   # consecutive ranks form one replica
    dp_color = rank % tp_size    # same TP position across replicas
    tp_comm = m.comm_split(comm, tp_color)
    dp_comm = m.comm_split(comm, dp_color)
    assert tp_comm.size == tp_size and dp_comm.size == dp_size

    torch.manual_seed(7)
    d, b = 6, 4
    w1 = torch.randn(h, d, dtype=torch.double)
    b1 = torch.randn(h, dtype=torch.double)
    w2 = torch.randn(d, h, dtype=torch.double)

    mlp = TensorParallelMLP(d, h, comm=tp_comm,
                            activation=torch.relu).double()
    mlp.up.shard_from_full(w1, b1)
    mlp.down.shard_from_full(w2)
    with torch.no_grad():
        mlp.down.bias.zero_()
    model = DistributedDataParallel(mlp, comm=dp_comm, bucket_cap_mb=0)

    # data shard per DP replica (= per TP group); identical inside one
    # TP group, different across replicas
    torch.manual_seed(1000 + tp_color)
    x = torch.randn(b, d, dtype=torch.double)

    loss = (model(x) ** 2).sum()
    loss.backward()
    model.finish_gradient_sync()

    # dense reference: per-shard gradients, then the DP average
    grads = []
    for s in range(dp_size):
        torch.manual_seed(1000 + s)
        xs = torch.randn(b, d, dtype=torch.double)
        w1r = w1.clone().requires_grad_()
        b1r = b1.clone().requires_grad_()
        w2r = w2.clone().requires_grad_()
        yr = torch.relu(xs @ w1r.t() + b1r) @ w2r.t()
        (yr ** 2).sum().backward()
        grads.append((w1r.grad, b1r.grad, w2r.grad))
    g1 = sum(g[0] for g in grads) / dp_size
    gb = sum(g[1] for g in grads) / dp_size
    g2 = sum(g[2] for g in grads) / dp_size

    tpr = tp_comm.rank
    lo, hi = tpr * mlp.up.out_local, (tpr + 1) * mlp.up.out_local
    assert torch.allclose(mlp.up.linear.weight.grad, g1[lo:hi], atol=1e-10)
    assert torch.allclose(mlp.up.linear.bias.grad, gb[lo:hi], atol=1e-10)
    lo2, hi2 = tpr * mlp.down.in_local, (tpr + 1) * mlp.down.in_local
    assert torch.allclose(mlp.down.linear.weight.grad, g2[:, lo2:hi2],
                          atol=1e-10)

    # one optimizer step: all four ranks' shards stay consistent with the
    # dense trajectory (TP shards per group, identical across DP pairs)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    opt.step()
    got = dp_comm.Bcast_(mlp.up.linear.weight.data.clone(), 0)
    assert torch.allclose(mlp.up.linear.weight.data, got, atol=1e-12), \
        "DP pair shards diverged after step"


def _hybrid_fsdp_worker(rank, world):
    # FSDP over the DP communicator with TP modules (over the TP
    # communicator) inside its units: parameter shards at rest ride dp,
    # forward/backward collectives ride tp, gradient reduce-scatters ride
    # dp — three communicators live at once (world + tp + dp)
    import mpi4torch_amd as m
    from mpi4torch_amd.parallel import FullyShardedDataParallel
    from mpi4torch_amd.parallel.tp import TensorParallelMLP

    comm = m.COMM_WORLD
    tp_size, dp_size = 2, world // 2
    tp_color, dp_color = rank // tp_size, rank % tp_size
    tp_comm = m.comm_split(comm, tp_color)
    dp_comm = m.comm_split(comm, dp_color)

    torch.manual_seed(7)
    d, h, b = 6, 8, 4
    w1 = torch.randn(h, d, dtype=torch.double)
    b1 = torch.randn(h, dtype=torch.double)
    w2 = torch.randn(d, h, dtype=torch.double)

    mlp = TensorParallelMLP(d, h, comm=tp_comm,
                            activation=torch.relu).double()
    mlp.up.shard_from_full(w1, b1)
    mlp.down.shard_from_full(w2)
    with torch.no_grad():
        mlp.down.bias.zero_()
    model = FullyShardedDataParallel(mlp, units=[mlp.up, mlp.down],
                                     comm=dp_comm)

    torch.manual_seed(1000 + tp_color)
    x = torch.randn(b, d, dtype=torch.double)
    loss = (model(x) ** 2).sum()
    loss.backward()
    model.finish_backward()

    # dense reference per DP shard, averaged (incl. the Row bias, which
    # is full-size and replicated across TP)
    grads = []
    for s in range(dp_size):
        torch.manual_seed(1000 + s)
        xs = torch.randn(b, d, dtype=torch.double)
        w1r = w1.clone().requires_grad_()
        b1r = b1.clone().requires_grad_()
        w2r = w2.clone().requires_grad_()
        b2r = torch.zeros(d, dtype=torch.double, requires_grad=True)
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
    # per-parameter dense expectations, keyed by identity (the unit's flat
    # layout follows module.parameters() order — direct Parameters come
    # BEFORE child modules', e.g. RowParallel's bias precedes its weight)
    want_by_param = {
        mlp.up.linear.weight: g1[lo:hi],
        mlp.up.linear.bias: gb[lo:hi],
        mlp.down.linear.weight: g2[:, lo2:hi2],
        mlp.down.bias: gb2,
    }
    for u in model._units:
        full = dp_comm.Allgather(u.shard.grad.detach(), 0)
        off = 0
        for p, n in zip(u.params, u.numels):
            got = full[off:off + n].view_as(p)
            want = want_by_param[p]
            assert torch.allclose(got, want, atol=1e-10), (p.shape, off)
            off += n


def test_hybrid_fsdp_tp_ws4():
    run_spmd(4, _hybrid_fsdp_worker)


def _hybrid_moe_dp_worker(rank, world):
    # EP x DP: expert-parallel MoE over the EP communicator inside DDP
    # over the DP communicator — the production MoE layout. Experts that
    # receive no tokens on a replica produce no gradients, so this is
    # also the real-world consumer of find_unused_parameters=True.
    import mpi4torch_amd as m
    from mpi4torch_amd.models.moe import ExpertParallelMoE
    from mpi4torch_amd.parallel import DistributedDataParallel

    comm = m.COMM_WORLD
    ep_size = 2
    ep_color, dp_color = rank // ep_size, rank % ep_size
    ep_comm = m.comm_split(comm, ep_color)
    dp_comm = m.comm_split(comm, dp_color)
    dp_size = dp_comm.size

    torch.manual_seed(3)
    d_model, n_experts, N = 8, 2 * ep_size, 12
    moe = ExpertParallelMoE(d_model, n_experts, d_hidden=16,
                            comm=ep_comm).double()
    model = DistributedDataParallel(moe, comm=dp_comm, bucket_cap_mb=0,
                                    find_unused_parameters=True)

    torch.manual_seed(50 + ep_color * 100 + ep_comm.rank)
    x = torch.randn(N, d_model, dtype=torch.double)
    y = model(x)
    y.square().sum().backward()

    # per-replica dense forward reference (same construction as the
    # single-comm MoE parity test)
    torch.manual_seed(3)
    full = ExpertParallelMoE(d_model, n_experts, d_hidden=16,
                             comm=ep_comm).double()
    logits = full.router(x)
    gates = torch.softmax(logits, dim=-1)
    expert = torch.argmax(gates, dim=-1)
    gate = gates.gather(1, expert.unsqueeze(1)).squeeze(1)
    y_ref = torch.zeros_like(x)
    for gid in range(n_experts):
        mask = expert == gid
        if bool(mask.any()):
            idx = mask.nonzero(as_tuple=True)[0]
            mod = full.experts[gid % full.experts_per_rank]
            y_ref = y_ref.index_copy(0, idx,
                                     mod(x.index_select(0, idx)))
    y_ref = y_ref * gate.unsqueeze(1)
    assert torch.allclose(y.detach(), y_ref, atol=1e-9)

    model.finish_gradient_sync()

    # DP-paired ranks (same EP position, different replicas) must agree
    # on every present gradient after the find_unused sync
    for p in moe.parameters():
        if p.grad is None:
            presence = dp_comm.Allreduce(
                torch.zeros(1, dtype=torch.double), m.MPI_MAX)
            assert (presence == 0).all(), "grad present on peer but None here"
            continue
        presence = dp_comm.Allreduce(
            torch.ones(1, dtype=torch.double), m.MPI_MAX)
        g_ref = dp_comm.Bcast_(p.grad.clone(), 0)
        assert torch.allclose(p.grad, g_ref, atol=1e-9), "dp grads diverged"

    # one step: replicas stay consistent
    opt = torch.optim.SGD([p for p in moe.parameters()], lr=0.01)
    opt.step()
    for p in moe.parameters():
        ref = dp_comm.Bcast_(p.data.clone(), 0)
        assert torch.allclose(p.data, ref, atol=1e-12)


def test_hybrid_moe_dp_ws4():
    run_spmd(4, _hybrid_moe_dp_worker)


def _hybrid_ulysses_dp_worker(rank, world):
    # SP x DP: Ulysses sequence-parallel attention (Alltoall reshards over
    # the SP communicator) inside DDP over the DP communicator — sequence
    # parallelism's alltoalls and DDP's allreduces run in one backward.
    import copy

    import mpi4torch_amd as m
    from mpi4torch_amd.models.transformer import UlyssesTransformerBlock
    from mpi4torch_amd.parallel import DistributedDataParallel

    comm = m.COMM_WORLD
    sp_size = 2
    sp_color, dp_color = rank // sp_size, rank % sp_size
    sp_comm = m.comm_split(comm, sp_color)
    dp_comm = m.comm_split(comm, dp_color)

    torch.manual_seed(21)
    d_model, n_heads, B, S = 8, 4, 2, 6  # S per rank (sequence-sharded)
    blk = UlyssesTransformerBlock(d_model, n_heads, comm=sp_comm).double()
    solo = copy.deepcopy(blk)  # same weights, no DDP
    model = DistributedDataParallel(blk, comm=dp_comm, bucket_cap_mb=0)

    # sequence shard per (replica, sp position); identical across the DP
    # pair so the solo reference sees the same data
    torch.manual_seed(300 + sp_color * 10 + sp_comm.rank)
    x = torch.randn(B, S, d_model, dtype=torch.double)

    y = model(x)
    y.square().sum().backward()
    model.finish_gradient_sync()

    # reference: the SAME block without DDP on the same sp_comm, then a
    # manual dp average of its gradients
    ys = solo(x)
    ys.square().sum().backward()
    assert torch.allclose(y.detach(), ys.detach(), atol=1e-9)
    for p, q in zip(blk.parameters(), solo.parameters()):
        want = dp_comm.Allreduce(q.grad, m.MPI_SUM) / dp_comm.size
        assert torch.allclose(p.grad, want, atol=1e-9)


def test_hybrid_ulysses_dp_ws4():
    run_spmd(4, _hybrid_ulysses_dp_worker)


def _hybrid_pp_dp_worker(rank, world):
    # PP x DP: a 2-stage GPipe over the PP communicator, replicated twice;
    # stage gradients average over the DP communicator after the drain.
    # Pipeline p2p (fwd + adjoint channels) and DP collectives share the
    # process; gradients must equal the dense 2-segment chain's, averaged
    # over the DP data shards.
    import mpi4torch_amd as m
    from mpi4torch_amd.parallel import GPipe

    comm = m.COMM_WORLD
    pp_size = 2
    pp_color, dp_color = rank // pp_size, rank % pp_size
    pp_comm = m.comm_split(comm, pp_color)   # replicas {0,1}, {2,3}
    dp_comm = m.comm_split(comm, dp_color)   # stage pairs {0,2}, {1,3}
    dp_size = dp_comm.size

    d, batch, n_mb = 5, 3, 4
    torch.manual_seed(31)
    segments = [
        torch.nn.Sequential(torch.nn.Linear(d, d, dtype=torch.double),
                            torch.nn.Tanh())
        for _ in range(pp_size)
    ]
    stage = segments[pp_comm.rank]
    pipe = GPipe(stage, recv_shape=(batch, d), recv_dtype=torch.double,
                 comm=pp_comm)

    def make_data(shard):
        torch.manual_seed(99 + shard)
        data = [torch.randn(batch, d, dtype=torch.double)
                for _ in range(n_mb)]
        tg = [torch.randn(batch, d, dtype=torch.double)
              for _ in range(n_mb)]
        return data, tg

    data, targets = make_data(pp_color)

    def loss_fn(y, i):
        return ((y - targets[i]) ** 2).sum()

    pipe.run(
        microbatches=data if pipe.is_first else None,
        loss_fn=loss_fn if pipe.is_last else None,
        n_microbatches=n_mb,
    )
    # DP gradient sync of this stage's accumulated gradients
    with torch.no_grad():
        for p in stage.parameters():
            p.grad = dp_comm.Allreduce(p.grad, m.MPI_SUM) / dp_size

    # dense reference, averaged over the DP shards
    torch.manual_seed(31)
    ref_segments = [
        torch.nn.Sequential(torch.nn.Linear(d, d, dtype=torch.double),
                            torch.nn.Tanh())
        for _ in range(pp_size)
    ]
    full = torch.nn.Sequential(*ref_segments)
    for shard in range(dp_size):
        sdata, stg = make_data(shard)
        total = torch.zeros((), dtype=torch.double)
        for i in range(n_mb):
            total = total + ((full(sdata[i]) - stg[i]) ** 2).sum()
        total.backward()
    for p, q in zip(stage.parameters(),
                    ref_segments[pp_comm.rank].parameters()):
        want = q.grad / dp_size
        assert torch.allclose(p.grad, want, atol=1e-9), (
            rank, (p.grad - want).abs().max())


def test_hybrid_pp_dp_ws4():
    run_spmd(4, _hybrid_pp_dp_worker)


def test_hybrid_tp2_dp2_ws4():
    run_spmd(4, _hybrid_worker, 2, 8)


def test_hybrid_tp3_dp2_ws6():
    run_spmd(6, _hybrid_worker, 3, 9)
