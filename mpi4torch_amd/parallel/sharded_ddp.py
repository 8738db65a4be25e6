"""ZeRO-2 data parallelism: bucketed gradient reduce-scatter overlapped
with backward + sharded optimizer + parameter allgather.

Per step and bucket, the wire carries exactly what DDP's allreduce would
(reduce-scatter + allgather = one allreduce), but each rank keeps only
1/P of the gradients (full per-bucket gradients are freed as soon as the
bucket's reduce-scatter is enqueued) and 1/P of the optimizer state.

Usage::

    model = ShardedDataParallel(module, torch.optim.AdamW, lr=1e-3)
    loss = model(x).sum()
    loss.backward()          # bucket reduce-scatters overlap backward
    model.step()             # wait -> sharded optimizer -> allgather params
"""

from typing import List

import torch

import mpi4torch_amd as m4a


class _Bucket:
    def __init__(self, params: List[torch.nn.Parameter], shard_len: int):
        self.params = params
        self.numels = [p.numel() for p in params]
        self.shard_len = shard_len  # padded_len // P
        self.pending = 0
        self.handle = None
        self.shard: torch.nn.Parameter = None  # this rank's slice


class ShardedDataParallel(torch.nn.Module):
    def __init__(self, module: torch.nn.Module, optimizer_cls, comm=None,
                 bucket_cap_mb: int = 64, average: bool = True,
                 master_dtype=None, **optim_kwargs):
        super().__init__()
        self.module = module
        self.comm = comm if comm is not None else m4a.COMM_WORLD
        self.average = average
        self._master_dtype = master_dtype  # None = model dtype
        self._sync_enabled = True
        P = self.comm.size

        if P > 1:
            with torch.no_grad():
                for p in self.module.parameters():
                    # Bcast_ returns a fresh tensor when an internal copy
                    # was needed (non-contiguous / host staging): copy back
                    res = self.comm.Bcast_(p.data, 0)
                    if res.data_ptr() != p.data.data_ptr():
                        p.data.copy_(res)

        params = [p for p in self.module.parameters() if p.requires_grad]
        assert params, "no trainable parameters"
        cap = bucket_cap_mb * 1024 * 1024
        self._buckets: List[_Bucket] = []
        cur: List[torch.nn.Parameter] = []
        size = 0
        for p in reversed(params):  # roughly backward completion order
            cur.append(p)
            size += p.numel() * p.element_size()
            if size >= cap:
                self._buckets.append(self._make_bucket(cur))
                cur, size = [], 0
        if cur:
            self._buckets.append(self._make_bucket(cur))

        self._param_bucket = {}
        for b in self._buckets:
            for p in b.params:
                self._param_bucket[p] = b
        for p in params:
            p.register_post_accumulate_grad_hook(self._grad_ready)
        self._reset_pending()

        self.optimizer = optimizer_cls([b.shard for b in self._buckets],
                                       **optim_kwargs)

    def _make_bucket(self, params: List[torch.nn.Parameter]) -> _Bucket:
        P = self.comm.size
        total = sum(p.numel() for p in params)
        shard_len = (total + P - 1) // P
        b = _Bucket(list(params), shard_len)
        # initialize this rank's parameter shard from the replicated params
        with torch.no_grad():
            flat = torch.zeros(shard_len * P, dtype=params[0].dtype,
                               device=params[0].device)
            torch.cat([p.reshape(-1) for p in params], out=flat[:total])
            lo = self.comm.rank * shard_len
            shard = flat[lo:lo + shard_len].clone()
            if self._master_dtype is not None:
                shard = shard.to(self._master_dtype)  # fp32 master shard
            b.shard = shard.requires_grad_()
        return b

    def _reset_pending(self):
        for b in self._buckets:
            b.pending = len(b.params)
            b.handle = None

    def _grad_ready(self, p: torch.nn.Parameter):
        if not self._sync_enabled:
            return  # accumulation microbatch: keep local grads
        b = self._param_bucket[p]
        b.pending -= 1
        if b.pending == 0:
            with torch.no_grad():
                P = self.comm.size
                total = sum(b.numels)
                flat = torch.zeros(b.shard_len * P, dtype=p.dtype,
                                   device=p.device)
                torch.cat([q.grad.reshape(-1) for q in b.params],
                          out=flat[:total])
                for q in b.params:
                    q.grad = None  # ZeRO-2: full grads freed immediately
                b.handle = (self.comm.Ireducescatter(flat, m4a.MPI_SUM)
                            if P > 1 else None)
                if P == 1:
                    b.shard.grad = flat.to(b.shard.dtype)

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    @torch.no_grad()
    def step(self):
        """Wait bucket reductions, run the sharded optimizer, allgather the
        updated shards back into the module parameters."""
        P = self.comm.size
        # a bucket that never filled would silently skip its reduce-scatter
        # while peers (whose graphs used those params) issue theirs —
        # divergence or a hang, so fail loudly. ZeRO-2 has no
        # find_unused_parameters mode; use DistributedDataParallel for
        # models with data-dependent control flow.
        fired = [b for b in self._buckets
                 if b.handle is not None or (P == 1 and b.shard.grad is not None)]
        if fired and self._sync_enabled and any(
                b.pending > 0 for b in self._buckets):
            n_miss = sum(b.pending for b in self._buckets)
            raise RuntimeError(
                "mpi4torch_amd ShardedDataParallel: "
                f"{n_miss} parameter(s) received no gradient this backward "
                "while other buckets already launched their reduce-scatter "
                "— replicas would diverge. Every parameter must get a "
                "gradient each backward under ZeRO-2; for data-dependent "
                "control flow use DistributedDataParallel("
                "find_unused_parameters=True)."
            )
        for b in self._buckets:
            if b.handle is not None:
                g = self.comm.Wait(b.handle)
                if self.average:
                    g = g / P
                b.shard.grad = (g.to(b.shard.dtype)
                                if g.dtype != b.shard.dtype else g)
        self.optimizer.step()
        for b in self._buckets:
            b.shard.grad = None
            local = b.shard.detach()
            if local.dtype != b.params[0].dtype:
                local = local.to(b.params[0].dtype)  # model dtype on the wire
            full = self.comm.Allgather(local, 0) if P > 1 else local
            off = 0
            for q, n in zip(b.params, b.numels):
                q.data.copy_(full[off:off + n].view_as(q))
                off += n
        self._reset_pending()

    def sharded_state_dict(self):
        """This rank's shard of the training state (bucket parameter
        shards + the sharded inner optimizer). Pair with
        utils.checkpoint.save_sharded_checkpoint; reload requires the
        SAME world size."""
        return {
            "buckets": [b.shard.detach().cpu() for b in self._buckets],
            "inner": self.optimizer.state_dict(),
        }

    @torch.no_grad()
    def load_sharded_state_dict(self, state):
        for b, s in zip(self._buckets, state["buckets"]):
            b.shard.copy_(s.to(b.shard.device))
        self.optimizer.load_state_dict(state["inner"])
        # rebroadcast full parameters from the restored shards
        P = self.comm.size
        for b in self._buckets:
            local = b.shard.detach()
            if local.dtype != b.params[0].dtype:
                local = local.to(b.params[0].dtype)
            full = self.comm.Allgather(local, 0) if P > 1 else local
            off = 0
            for q, n in zip(b.params, b.numels):
                q.data.copy_(full[off:off + n].view_as(q))
                off += n

    def no_sync(self):
        """Context manager for gradient-accumulation microbatches: local
        gradients accumulate in .grad; the final backward OUTSIDE the
        context reduces the accumulated values."""
        sdp = self

        class _NoSync:
            def __enter__(self):
                sdp._sync_enabled = False

            def __exit__(self, *exc):
                sdp._sync_enabled = True

        return _NoSync()

    def zero_grad(self, set_to_none: bool = True):
        for p in self.module.parameters():
            p.grad = None
