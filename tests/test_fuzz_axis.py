"""Seeded randomized sweep over axis-collective geometries.

For random (ndim, shape, axis, per-rank counts), check Gather/Allgather/
Scatter/Alltoall against references computed locally from the allgathered
inputs. Shakes out marshaling corner cases (odd sizes, axis extremes,
zero counts, mixed dtypes) beyond the hand-picked tests.
"""

import random

import torch

from spmd import run_spmd


def _rand_shape(rng, ndim):
    return [rng.randint(1, 6) for _ in range(ndim)]


def _fuzz_worker(rank, world, n_iters):
    import mpi4torch_amd as m

    comm = m.COMM_WORLD
    rng = random.Random(1234)  # SAME stream on all ranks (SPMD decisions)
    for it in range(n_iters):
        ndim = rng.randint(1, 4)
        shape = _rand_shape(rng, ndim)
        axis = rng.randint(0, ndim - 1)
        dtype = rng.choice([torch.float64, torch.float32, torch.bfloat16])
        # per-rank axis sizes (may include 0)
        axis_sizes = [rng.randint(0, 5) for _ in range(world)]
        if sum(axis_sizes) == 0:
            axis_sizes[rng.randint(0, world - 1)] = 1

        myshape = list(shape)
        myshape[axis] = axis_sizes[rank]
        torch.manual_seed(10_000 * it + rank)
        mine = torch.randn(myshape).to(dtype)

        # every rank reconstructs everyone's tensors (same seeds)
        all_tensors = []
        for r in range(world):
            s = list(shape)
            s[axis] = axis_sizes[r]
            torch.manual_seed(10_000 * it + r)
            all_tensors.append(torch.randn(s).to(dtype))
        full = torch.cat(all_tensors, dim=axis)

        # Gather
        g = comm.Gather(mine, axis, it % world)
        if rank == it % world:
            assert torch.equal(g, full), f"gather it={it}"
        else:
            assert g.shape[axis] == 0

        # Allgather
        ag = comm.Allgather(mine, axis)
        assert torch.equal(ag, full), f"allgather it={it}"

        # Scatter back from the gathered full tensor
        root = (it + 1) % world
        src = full if rank == root else torch.zeros(1, dtype=dtype)
        sc = comm.Scatter(src, axis, axis_sizes[rank], root)
        assert torch.equal(sc, mine), f"scatter it={it}"

        # same-axis Alltoall: repartition to a rotated count layout
        tgt = [axis_sizes[(r + 1) % world] for r in range(world)]
        a2a = comm.Alltoall(mine, axis, axis, tgt[rank])
        off = sum(tgt[:rank])
        expect = full.narrow(axis, off, tgt[rank])
        assert torch.equal(a2a, expect), f"same-axis alltoall it={it}"

        # different-axes Alltoall (needs ndim >= 2 and a divisible axis)
        if ndim >= 2:
            gaxis = axis
            saxis = (axis + 1) % ndim
            # rebuild inputs whose scatter axis is the global partition
            scounts = [rng.randint(0, 3) for _ in range(world)]
            if sum(scounts) == 0:
                scounts[0] = 1
            s2 = list(shape)
            s2[saxis] = sum(scounts)
            s2[gaxis] = axis_sizes[rank]
            torch.manual_seed(77_000 + 10_000 * it + rank)
            mine2 = torch.randn(s2).to(dtype)
            out = comm.Alltoall(mine2, gaxis, saxis, scounts[rank])
            # reference: concat everyone's slice of my scatter interval
            all2 = []
            for r in range(world):
                sr = list(s2)
                sr[gaxis] = axis_sizes[r]
                torch.manual_seed(77_000 + 10_000 * it + r)
                all2.append(torch.randn(sr).to(dtype))
            lo = sum(scounts[:rank])
            expect2 = torch.cat(
                [t.narrow(saxis, lo, scounts[rank]) for t in all2], dim=gaxis)
            assert torch.equal(out, expect2), f"alltoall it={it}"


def _fuzz_phased_worker(rank, world, n_iters):
    # identical sweep through the phased (chunked) pipeline: 100-byte
    # chunks force K=4 phases on every geometry that marshals
    import os

    os.environ["MPI4TORCH_AMD_PIPELINE_MB"] = "0.0001"
    _fuzz_worker(rank, world, n_iters)


def test_fuzz_ws2():
    run_spmd(2, _fuzz_worker, 25)


def test_fuzz_ws3():
    run_spmd(3, _fuzz_worker, 25)


def test_fuzz_phased_ws3():
    run_spmd(3, _fuzz_phased_worker, 25)


def test_fuzz_phased_ws5():
    run_spmd(5, _fuzz_phased_worker, 15)
