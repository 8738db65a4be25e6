"""torch.distributed interop: communicators from existing process groups
(the torch.distributed analog of reference tests/test_mpi4pyinterop.py) and
communicator serialization (reference csrc/extension.cpp:1283-1297 — whose
deserializer had an inverted condition; ours round-trips).
"""

import torch

from spmd import run_spmd


def _pg_worker(rank, world):
    import torch.distributed as dist
    import mpi4torch_amd as m

    # wrap the default group
    comm = m.comm_from_process_group(dist.distributed_c10d._get_default_group())
    assert comm.rank == dist.get_rank()
    assert comm.size == dist.get_world_size()
    t = torch.ones(5, dtype=torch.double).requires_grad_()
    res = comm.Allreduce(t, m.MPI_SUM)
    assert (res == world).all()
    res.sum().backward()
    assert (t.grad == world).all()


def _subgroup_worker(rank, world):
    import torch.distributed as dist
    import mpi4torch_amd as m

    evens = list(range(0, world, 2))
    odds = list(range(1, world, 2))
    # every rank must call new_group for both groups (collective contract)
    g_even = dist.new_group(ranks=evens, backend="gloo")
    g_odd = dist.new_group(ranks=odds, backend="gloo")
    my_group = g_even if rank % 2 == 0 else g_odd
    my_ranks = evens if rank % 2 == 0 else odds

    comm = m.comm_from_process_group(my_group)
    assert comm.size == len(my_ranks)
    assert comm.rank == my_ranks.index(rank)
    # sum of global ranks within the subgroup
    t = torch.full((3,), float(rank), dtype=torch.double)
    res = comm.Allreduce(t, m.MPI_SUM)
    assert (res == sum(my_ranks)).all()


def _pickle_worker(rank, world):
    import io
    import mpi4torch_amd as m

    comm = m.COMM_WORLD

    # serialize the custom-class communicator through TorchScript pickling
    # inside a module attribute (the supported custom-class pickle path)
    class Holder(torch.nn.Module):
        def __init__(self, c):
            super().__init__()
            self.c = c

        def forward(self) -> int:
            return self.c.GetSize()

    holder = torch.jit.script(Holder(comm._comm))
    buf = io.BytesIO()
    torch.jit.save(holder, buf)
    buf.seek(0)
    loaded = torch.jit.load(buf)
    assert loaded() == world  # deserialized communicator resolves the group


def test_pg_interop_ws2():
    run_spmd(2, _pg_worker)


def test_pg_interop_ws5():
    run_spmd(5, _pg_worker)


def test_subgroups_ws5():
    run_spmd(5, _subgroup_worker)


def test_pickle_ws2():
    run_spmd(2, _pickle_worker)


def _split_worker(rank, world):
    import mpi4torch_amd as m

    comm = m.COMM_WORLD
    # three colors; key reverses the order inside each color
    color = rank % 3
    sub = m.comm_split(comm, color)
    members = sorted(r for r in range(world) if r % 3 == color)
    assert sub.size == len(members)
    assert sub.rank == members.index(rank), (rank, sub.rank)
    t = torch.full((2,), float(rank), dtype=torch.double)
    s = sub.Allreduce(t, m.MPI_SUM)
    assert (s == sum(members)).all()

    # undefined color drops out (but must still participate in the call)
    sub2 = m.comm_split(comm, -1 if rank == 0 else 7)
    if rank == 0:
        assert sub2 is None
    else:
        assert sub2 is not None and sub2.size == world - 1


def test_comm_split_ws5():
    run_spmd(5, _split_worker)


def _grandchild_split_worker(rank, world):
    import mpi4torch_amd as m

    comm = m.COMM_WORLD
    # first split: {0,1,2,3} vs {4,5,6}
    c1 = 0 if rank < 4 else 1
    child = m.comm_split(comm, c1)
    members1 = [r for r in range(world) if (0 if r < 4 else 1) == c1]
    assert child.size == len(members1)
    assert child.rank == members1.index(rank)

    # split the CHILD: global-rank parity. Round 1's bug passed child-local
    # ranks to dist.new_group (which wants GLOBAL ranks), silently grouping
    # the wrong processes for any comm that is not the world.
    c2 = rank % 2
    grand = m.comm_split(child, c2)
    members2 = [r for r in members1 if r % 2 == c2]
    assert grand.size == len(members2), (rank, grand.size, members2)
    assert grand.rank == members2.index(rank), (rank, grand.rank)
    t = torch.full((3,), float(rank), dtype=torch.double)
    s = grand.Allreduce(t, m.MPI_SUM)
    assert (s == sum(members2)).all(), (rank, s[0].item(), members2)
    # and the grandchild is a working communicator for axis collectives too
    g = grand.Allgather(torch.full((1,), float(rank), dtype=torch.double), 0)
    assert [int(v.item()) for v in g] == members2


def test_comm_split_grandchild_ws7():
    run_spmd(7, _grandchild_split_worker)
