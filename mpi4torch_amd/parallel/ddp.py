"""Bucketed data-parallel gradient synchronization over xGMI.

A light DDP built on mpi4torch_amd's non-blocking Allreduce: gradients are
flattened into buckets as backward produces them and each full bucket's
allreduce is launched immediately on the collective stream, overlapping the
rest of backward. Waits are deferred to finish_gradient_sync() (call it
between loss.backward() and optimizer.step()).

Sizing rationale (MI355X): intra-node xGMI is 7 point-to-point links per
GPU at ~153 GB/s each; ring allreduce is per-link bound, and RCCL needs
buckets large enough to stripe across its channels — default 128 MiB
(larger than CUDA-era defaults; 288 GB HBM3E makes big buckets free).

Unused parameters: by default every parameter must receive a gradient each
backward; a partially-filled bucket raises loudly at finish_gradient_sync
(a silent skip would leave those gradients rank-local — divergence, not a
crash). For models with data-dependent control flow pass
``find_unused_parameters=True``: bucket launches are deferred to
finish_gradient_sync, which first agrees on the globally-used parameter
set over the communicator and zero-fills locally-missing gradients, so
every rank issues the identical collective sequence by construction.
"""

from typing import Dict, List, Optional

import torch

import mpi4torch_amd as m4a


class _Bucket:
    def __init__(self, params: List[torch.nn.Parameter]):
        self.params = params
        self.pending = 0
        self.flat: Optional[torch.Tensor] = None
        self.handle = None


class DistributedDataParallel(torch.nn.Module):
    """Wrap a module for data-parallel training.

    Usage::

        model = DistributedDataParallel(module)
        loss = model(x).sum()
        loss.backward()
        model.finish_gradient_sync()   # waits overlapped allreduces
        optimizer.step()
    """

    def __init__(self, module: torch.nn.Module, comm=None,
                 bucket_cap_mb: int = 128, average: bool = True,
                 find_unused_parameters: bool = False):
        super().__init__()
        self.module = module
        self.comm = comm if comm is not None else m4a.COMM_WORLD
        self.average = average
        self.find_unused_parameters = find_unused_parameters
        self._sync_enabled = True

        # broadcast initial parameters from rank 0 so replicas agree.
        # Bcast_ is in-place only when no internal copy was needed (it
        # returns a fresh tensor for non-contiguous inputs or under
        # force_host_staging) — always copy the result back.
        if self.comm.size > 1:
            with torch.no_grad():
                for p in self.module.parameters():
                    res = self.comm.Bcast_(p.data, 0)
                    if res.data_ptr() != p.data.data_ptr():
                        p.data.copy_(res)

        # buckets in reverse parameter order (grads arrive roughly in
        # reverse forward order during backward)
        params = [p for p in self.module.parameters() if p.requires_grad]
        cap = bucket_cap_mb * 1024 * 1024
        self._buckets: List[_Bucket] = []
        cur: List[torch.nn.Parameter] = []
        size = 0
        for p in reversed(params):
            cur.append(p)
            size += p.numel() * p.element_size()
            if size >= cap:
                self._buckets.append(_Bucket(cur))
                cur, size = [], 0
        if cur:
            self._buckets.append(_Bucket(cur))

        self._params = params
        self._param_bucket = {}
        self._fired: Dict[torch.nn.Parameter, bool] = {}
        for b in self._buckets:
            for p in b.params:
                self._param_bucket[p] = b
        for p in params:
            p.register_post_accumulate_grad_hook(self._grad_ready)

        self._reset_pending()

    def _reset_pending(self):
        for b in self._buckets:
            b.pending = len(b.params)
            b.flat = None
            b.handle = None
        self._fired = {p: False for p in self._params}

    def _grad_ready(self, p: torch.nn.Parameter):
        if not self._sync_enabled or self.comm.size == 1:
            return
        self._fired[p] = True
        b = self._param_bucket[p]
        b.pending -= 1
        if b.pending == 0 and not self.find_unused_parameters:
            # eager launch overlaps the rest of backward. With
            # find_unused_parameters the launch is deferred: a bucket that
            # is complete HERE may be incomplete on a peer, and eagerly
            # posting would desynchronize the collective sequence.
            with torch.no_grad():
                b.flat = torch.cat(
                    [q.grad.reshape(-1) for q in b.params]
                ).contiguous()
                b.handle = self.comm.Iallreduce(b.flat, m4a.MPI_SUM)

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    def finish_gradient_sync(self):
        """Wait all in-flight bucket allreduces and scatter results back
        into .grad. Call between backward() and optimizer.step()."""
        if self.comm.size == 1:
            return
        if self.find_unused_parameters:
            self._sync_with_unused()
            self._reset_pending()
            return
        fired = [b for b in self._buckets if b.handle is not None]
        missing = [q for b in self._buckets if b.handle is None
                   for q in b.params if not self._fired[q]]
        if fired and missing:
            names = {id(p): n for n, p in self.module.named_parameters()}
            shown = ", ".join(names.get(id(q), "<param>") for q in missing[:5])
            raise RuntimeError(
                "mpi4torch_amd DDP: "
                f"{len(missing)} parameter(s) received no gradient this "
                f"backward (e.g. {shown}) while other buckets already "
                "launched their allreduce — the skipped gradients would "
                "stay rank-local and replicas would silently diverge. "
                "Construct DistributedDataParallel with "
                "find_unused_parameters=True for models with "
                "data-dependent control flow."
            )
        scale = 1.0 / self.comm.size if self.average else 1.0
        with torch.no_grad():
            for b in self._buckets:
                if b.handle is None:
                    continue
                reduced = self.comm.Wait(b.handle)
                if self.average:
                    reduced = reduced * scale
                off = 0
                for q in b.params:
                    n = q.numel()
                    q.grad.copy_(reduced[off : off + n].view_as(q.grad))
                    off += n
        self._reset_pending()

    def _sync_with_unused(self):
        """find_unused_parameters path: agree on the globally-used set,
        zero-fill locally-missing gradients, then reduce every bucket that
        is used anywhere — the collective sequence is identical on all
        ranks by construction."""
        with torch.no_grad():
            local = torch.tensor(
                [1.0 if self._fired[p] else 0.0 for p in self._params],
                dtype=torch.float64,
            )
            used = self.comm.Allreduce(local, m4a.MPI_MAX)
            used_set = {
                p for p, u in zip(self._params, used.tolist()) if u > 0.5
            }
            scale = 1.0 / self.comm.size if self.average else 1.0
            handles = []
            for b in self._buckets:
                bp = [q for q in b.params if q in used_set]
                if not bp:
                    continue
                flat = torch.cat([
                    q.grad.reshape(-1) if self._fired[q]
                    else torch.zeros(q.numel(), dtype=q.dtype,
                                     device=q.device)
                    for q in bp
                ]).contiguous()
                handles.append((b, bp, self.comm.Iallreduce(flat, m4a.MPI_SUM)))
            for b, bp, h in handles:
                reduced = self.comm.Wait(h)
                if self.average:
                    reduced = reduced * scale
                off = 0
                for q in bp:
                    n = q.numel()
                    if q.grad is None:
                        q.grad = torch.zeros_like(q)
                    q.grad.copy_(reduced[off : off + n].view_as(q.grad))
                    off += n

    def no_sync(self):
        """Context manager: skip gradient sync (gradient accumulation)."""
        ddp = self

        class _NoSync:
            def __enter__(self):
                ddp._sync_enabled = False

            def __exit__(self, *exc):
                ddp._sync_enabled = True

        return _NoSync()
