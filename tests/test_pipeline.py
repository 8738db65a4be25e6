"""GPipe pipeline on autograd-transparent p2p: losses and parameter
gradients must match a dense single-process run exactly."""

import torch

from spmd import run_spmd


def _pipeline_worker(rank, world):
    from mpi4torch_amd.parallel.pipeline import GPipe

    torch.manual_seed(31)  # identical full model on all ranks
    d = 8
    segments = [
        torch.nn.Sequential(torch.nn.Linear(d, d, dtype=torch.double),
                            torch.nn.Tanh())
        for _ in range(world)
    ]
    n_mb, batch = 3, 4
    torch.manual_seed(99)
    data = [torch.randn(batch, d, dtype=torch.double) for _ in range(n_mb)]
    targets = [torch.randn(batch, d, dtype=torch.double) for _ in range(n_mb)]

    stage = segments[rank]
    pipe = GPipe(stage, recv_shape=(batch, d), recv_dtype=torch.double)

    def loss_fn(y, i):
        return ((y - targets[i]) ** 2).sum()

    losses = pipe.run(
        microbatches=data if rank == 0 else None,
        loss_fn=loss_fn if rank == world - 1 else None,
        n_microbatches=n_mb,
    )

    # dense reference: same segments chained on one process
    torch.manual_seed(31)
    ref_segments = [
        torch.nn.Sequential(torch.nn.Linear(d, d, dtype=torch.double),
                            torch.nn.Tanh())
        for _ in range(world)
    ]
    full = torch.nn.Sequential(*ref_segments)
    total = torch.zeros((), dtype=torch.double)
    for i in range(n_mb):
        total = total + ((full(data[i]) - targets[i]) ** 2).sum()
    total.backward()

    if rank == world - 1:
        assert len(losses) == n_mb
        ref_losses = [((full(data[i]) - targets[i]) ** 2).sum() for i in
                      range(n_mb)]
        for got, ref in zip(losses, ref_losses):
            assert torch.allclose(got, ref.detach(), atol=1e-9), (got, ref)

    # this rank's stage gradients == the dense segment's gradients
    for p, q in zip(stage.parameters(), ref_segments[rank].parameters()):
        assert q.grad is not None and p.grad is not None
        assert torch.allclose(p.grad, q.grad, atol=1e-9), (
            rank, (p.grad - q.grad).abs().max())


def test_pipeline_ws2():
    run_spmd(2, _pipeline_worker)


def test_pipeline_ws5():
    run_spmd(5, _pipeline_worker)
