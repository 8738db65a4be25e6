"""SPMD test harness: run a worker function on a spawned gloo world.

Mirrors the reference's test style (tests executed identically on every
rank under mpirun, reference .github/workflows/test.yml:64-84) with
torch.multiprocessing.spawn + gloo standing in for mpirun. World sizes 2/5/7
deliberately include primes to shake out divisibility assumptions, like the
reference CI matrix.
"""

import os
import socket

import torch.multiprocessing as mp


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _entry(rank, world_size, port, fn, args):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        fn(rank, world_size, *args)
    finally:
        dist.destroy_process_group()


def run_spmd(world_size, fn, *args):
    """Spawn `world_size` ranks, each running fn(rank, world_size, *args).

    Retries (3 attempts): rapid-fire test sessions can race on the
    just-freed rendezvous port (TOCTOU between _free_port and gloo's bind).
    """
    last = None
    for _ in range(3):
        port = _free_port()
        try:
            mp.spawn(
                _entry,
                args=(world_size, port, fn, args),
                nprocs=world_size,
                join=True,
            )
            return
        except Exception as e:  # noqa: BLE001 - retried once, then re-raised
            last = e
    raise last
