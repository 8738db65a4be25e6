"""Distributed checkpoint helpers.

The reference has no checkpoint story beyond (broken) communicator
pickling. These helpers cover the common SPMD cases:
rank-0 saving of replicated state and broadcast-on-load, built on the
framework's own collectives so they work for CPU and GPU tensors alike.
"""

import os
from typing import Optional

import torch

import mpi4torch_amd as m4a


def save_checkpoint(path: str, model: torch.nn.Module,
                    optimizer: Optional[torch.optim.Optimizer] = None,
                    comm=None, extra: Optional[dict] = None) -> None:
    """Save replicated (data-parallel) training state from rank 0.

    Ranks holding identical replicas (the DDP/linreg pattern) write one
    file; other ranks return immediately. A barrier-equivalent is NOT
    implied — callers who need the file visible everywhere should follow
    with their own synchronization.
    """
    comm = comm if comm is not None else m4a.COMM_WORLD
    if comm.rank != 0:
        return
    state = {"model": model.state_dict()}
    if optimizer is not None:
        state["optimizer"] = optimizer.state_dict()
    if extra:
        state["extra"] = extra
    tmp = path + ".tmp"
    torch.save(state, tmp)
    os.replace(tmp, path)


def load_checkpoint(path: str, model: torch.nn.Module,
                    optimizer: Optional[torch.optim.Optimizer] = None,
                    comm=None, map_location="cpu") -> Optional[dict]:
    """Load on rank 0 and broadcast parameters to all ranks.

    Returns the `extra` dict (on every rank: broadcast via collectives is
    tensor-only, so `extra` is returned on rank 0 and None elsewhere).
    """
    comm = comm if comm is not None else m4a.COMM_WORLD
    extra = None
    if comm.rank == 0:
        state = torch.load(path, map_location=map_location,
                           weights_only=False)
        model.load_state_dict(state["model"])
        if optimizer is not None and "optimizer" in state:
            optimizer.load_state_dict(state["optimizer"])
        extra = state.get("extra")
    if comm.size > 1:
        with torch.no_grad():
            for p in model.parameters():
                p.data.copy_(comm.Bcast_(p.data.clone(), 0))
            for b in model.buffers():
                b.data.copy_(comm.Bcast_(b.data.clone(), 0))
    return extra


def save_sharded_checkpoint(path: str, state: dict, comm=None) -> None:
    """Save per-rank sharded state (ZeRO-1/2/3): every rank writes its own
    `path.shard<rank>-of-<size>.pt`, stamped with the world size. Use with
    ZeroRedundancyOptimizer.state_dict(), ShardedDataParallel /
    FullyShardedDataParallel .sharded_state_dict(), or any rank-local
    dict. Reload requires the SAME world size (no resharding)."""
    comm = comm if comm is not None else m4a.COMM_WORLD
    fname = f"{path}.shard{comm.rank:05d}-of-{comm.size:05d}.pt"
    tmp = fname + ".tmp"
    torch.save({"world_size": comm.size, "rank": comm.rank,
                "state": state}, tmp)
    os.replace(tmp, fname)


def load_sharded_checkpoint(path: str, comm=None,
                            map_location="cpu") -> dict:
    """Load this rank's shard written by save_sharded_checkpoint. Raises
    with a clear message when the world size differs from the one that
    saved (sharded layouts are world-size-specific)."""
    comm = comm if comm is not None else m4a.COMM_WORLD
    fname = f"{path}.shard{comm.rank:05d}-of-{comm.size:05d}.pt"
    if not os.path.exists(fname):
        import glob as _glob

        found = sorted(_glob.glob(f"{path}.shard*-of-*.pt"))
        hint = (f" (found {len(found)} shard files, e.g. {found[0]!r} — "
                "sharded checkpoints must be reloaded at the world size "
                "that saved them)") if found else ""
        raise FileNotFoundError(f"no shard file {fname!r}{hint}")
    blob = torch.load(fname, map_location=map_location, weights_only=False)
    if blob["world_size"] != comm.size or blob["rank"] != comm.rank:
        raise RuntimeError(
            f"sharded checkpoint {fname!r} was written by rank "
            f"{blob['rank']}/{blob['world_size']} but is being loaded by "
            f"rank {comm.rank}/{comm.size}; resharding across world sizes "
            "is not supported")
    return blob["state"]
