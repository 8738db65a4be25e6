"""ZeRO-3-lite (FSDP): parameters sharded at rest, materialized per unit.

Each wrapped unit's parameters live as views into one flat buffer whose
storage is freed (resized to 0) whenever the unit is idle; every rank
keeps only a 1/P shard. Around a unit's forward and backward the full
buffer is re-materialized with one Allgather; after backward the unit's
gradients leave as one reduce-scatter and everything full-sized is freed
again. Peak parameter memory is one unit, not the model.

The storage-resize trick mirrors torch FSDP: autograd's saved tensors are
views of the SAME storage, so restoring the storage before backward
revalidates them in place.

Usage::

    model = FullyShardedDataParallel(module, units=[m1, m2, ...])
    opt = torch.optim.AdamW(model.shard_parameters(), lr=...)
    loss = model(x).sum(); loss.backward()
    model.finish_backward()      # waits grad reduce-scatters
    opt.step(); model.refresh_shards()
"""

from typing import List, Optional, Sequence

import torch

import mpi4torch_amd as m4a


class _Unit:
    def __init__(self, module: torch.nn.Module, comm, master_dtype=None):
        self.module = module
        self.comm = comm
        self.params: List[torch.nn.Parameter] = [
            p for p in module.parameters() if p.requires_grad
        ]
        assert self.params, "FSDP unit has no trainable parameters"
        P = comm.size
        self.numels = [p.numel() for p in self.params]
        total = sum(self.numels)
        self.shard_len = (total + P - 1) // P
        self.padded = self.shard_len * P

        p0 = self.params[0]
        with torch.no_grad():
            flat = torch.zeros(self.padded, dtype=p0.dtype, device=p0.device)
            torch.cat([p.reshape(-1) for p in self.params], out=flat[:total])
            lo = comm.rank * self.shard_len
            # persistent 1/P shard (the optimizer's parameter); optionally a
            # wider master dtype (fp32 masters for a bf16 model)
            shard = flat[lo:lo + self.shard_len].clone()
            if master_dtype is not None:
                shard = shard.to(master_dtype)
            self.shard = torch.nn.Parameter(shard)
            # the full buffer the module computes with; params become views
            self.flat = flat
            off = 0
            for p, n in zip(self.params, self.numels):
                p.data = flat[off:off + n].view_as(p)
                off += n
        self.materialized = True
        self.grad_handle = None
        self.prefetch_handle = None
        self.pending = 0

    @torch.no_grad()
    def free(self):
        if self.materialized:
            self.flat.untyped_storage().resize_(0)
            self.materialized = False

    @torch.no_grad()
    def start_materialize(self):
        """Prefetch: launch this unit's parameter allgather without waiting
        (Iallgather skips the returning stream bracket), so it overlaps the
        CURRENT unit's compute. materialize() consumes the handle."""
        if (self.materialized or self.prefetch_handle is not None
                or self.comm.size == 1):
            return
        local = self.shard.detach()
        if local.dtype != self.flat.dtype:
            local = local.to(self.flat.dtype)
        self.prefetch_handle = self.comm.Iallgather(local)

    @torch.no_grad()
    def materialize(self):
        if self.materialized:
            return
        self.flat.untyped_storage().resize_(
            self.padded * self.flat.element_size())
        if self.prefetch_handle is not None:
            full = self.comm.Wait(self.prefetch_handle)
            self.prefetch_handle = None
            self.flat.copy_(full)
            self.materialized = True
            return
        local = self.shard.detach()
        if local.dtype != self.flat.dtype:
            local = local.to(self.flat.dtype)  # model dtype on the wire
        if self.comm.size > 1:
            full = self.comm.Allgather(local, 0)
            self.flat.copy_(full)
        else:
            self.flat[:self.shard_len].copy_(local)
        self.materialized = True

    @torch.no_grad()
    def start_grad_reduce(self):
        total = sum(self.numels)
        gflat = torch.zeros(self.padded, dtype=self.flat.dtype,
                            device=self.flat.device)
        off = 0
        for p, n in zip(self.params, self.numels):
            if p.grad is not None:
                gflat[off:off + n].copy_(p.grad.reshape(-1))
            p.grad = None
            off += n
        if self.comm.size > 1:
            self.grad_handle = self.comm.Ireducescatter(gflat, m4a.MPI_SUM)
        else:
            self.shard.grad = (gflat[:self.shard_len].clone()
                               .to(self.shard.dtype))
        self.free()  # full params not needed past this unit's backward

    @torch.no_grad()
    def finish_grad_reduce(self, average: bool):
        if self.grad_handle is not None:
            g = self.comm.Wait(self.grad_handle)
            if average:
                g = g / self.comm.size
            self.shard.grad = (g.to(self.shard.dtype)
                               if g.dtype != self.shard.dtype else g)
            self.grad_handle = None


class FullyShardedDataParallel(torch.nn.Module):
    def __init__(self, module: torch.nn.Module,
                 units: Optional[Sequence[torch.nn.Module]] = None,
                 comm=None, average: bool = True, master_dtype=None,
                 prefetch: int = 1):
        super().__init__()
        self.module = module
        self.comm = comm if comm is not None else m4a.COMM_WORLD
        self.average = average
        # how many upcoming units to allgather ahead of use (0 disables;
        # each in-flight prefetch holds one unit's full buffer, so the
        # memory high-water mark is (1 + prefetch) units)
        self.prefetch = max(0, int(prefetch))
        self._sync_enabled = True
        if self.comm.size > 1:
            with torch.no_grad():
                for p in module.parameters():
                    # Bcast_ returns a fresh tensor when an internal copy
                    # was needed (non-contiguous / host staging): copy back
                    res = self.comm.Bcast_(p.data, 0)
                    if res.data_ptr() != p.data.data_ptr():
                        p.data.copy_(res)
        unit_modules = list(units) if units is not None else [
            m for m in module.children()
            if any(p.requires_grad for p in m.parameters())
        ]
        assert unit_modules, "no FSDP units found"
        # Parameters shared ACROSS units (weight tying between units) would
        # be flattened into two shards and silently diverge; detect and
        # refuse. Tying WITHIN one unit is fine: nn.Module.parameters()
        # deduplicates, so the unit flattens one copy that both uses view.
        seen = {}
        for m in unit_modules:
            for name, p in m.named_parameters():
                if not p.requires_grad:
                    continue
                if p in seen:
                    raise RuntimeError(
                        "mpi4torch_amd FSDP: parameter is shared between "
                        f"units ('{seen[p]}' and '{name}'). Tied parameters "
                        "must live in the SAME unit — pass units=[...] "
                        "grouping the tied modules together."
                    )
                seen[p] = name
        self._units = [_Unit(m, self.comm, master_dtype)
                       for m in unit_modules]
        self._by_module = {u.module: u for u in self._units}
        self._by_param = {p: u for u in self._units for p in u.params}

        for u in self._units:
            u.module.register_forward_pre_hook(self._pre_forward)
            u.module.register_forward_hook(self._post_forward)
            u.module.register_full_backward_pre_hook(self._pre_backward)
            for p in u.params:
                p.register_post_accumulate_grad_hook(self._grad_ready)
        # params at rest are sharded
        for u in self._units:
            u.free()

    # ---- hooks -----------------------------------------------------------
    def _pre_forward(self, module, inputs):
        u = self._by_module[module]
        u.materialize()
        # prefetch the next `prefetch` units' allgathers so they overlap
        # this unit's forward compute (unit order = construction order)
        i = self._units.index(u)
        for d in range(1, self.prefetch + 1):
            if i + d < len(self._units):
                self._units[i + d].start_materialize()

    def _post_forward(self, module, inputs, output):
        u = self._by_module[module]
        u.pending = len(u.params)
        u.free()  # re-materialized by the pre-backward hook when training
        return output

    def _pre_backward(self, module, grad_output):
        u = self._by_module[module]
        u.materialize()
        # backward visits units in reverse: prefetch the PREVIOUS units
        i = self._units.index(u)
        for d in range(1, self.prefetch + 1):
            if i - d >= 0:
                self._units[i - d].start_materialize()

    def _grad_ready(self, p):
        if not self._sync_enabled:
            return  # accumulation: keep full local grads; unit stays freed
        u = self._by_param[p]
        u.pending -= 1
        if u.pending == 0:
            u.start_grad_reduce()

    # ---- public ----------------------------------------------------------
    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    def shard_parameters(self):
        return [u.shard for u in self._units]

    def finish_backward(self):
        """Wait all gradient reduce-scatters (call after loss.backward())."""
        # a unit that only partially produced gradients never launched its
        # reduce-scatter while peers may have — fail loudly instead of
        # silently diverging (FSDP requires every unit parameter to get a
        # gradient each backward; for data-dependent control flow use
        # DistributedDataParallel(find_unused_parameters=True))
        if self._sync_enabled:
            fired = any(u.grad_handle is not None or
                        (self.comm.size == 1 and u.shard.grad is not None)
                        for u in self._units)
            missing = sum(u.pending for u in self._units
                          if 0 < u.pending <= len(u.params))
            if fired and missing:
                raise RuntimeError(
                    "mpi4torch_amd FSDP: "
                    f"{missing} parameter(s) received no gradient this "
                    "backward while other units already launched their "
                    "reduce-scatter — replicas would diverge. Every unit "
                    "parameter must get a gradient each backward."
                )
        for u in self._units:
            u.finish_grad_reduce(self.average)

    @torch.no_grad()
    def refresh_shards(self):
        """No-op placeholder for symmetry: the optimizer updates the shards
        in place; full parameters re-materialize lazily at next use."""
        for u in self._units:
            u.shard.grad = None

    def sharded_state_dict(self):
        """This rank's parameter shards (one per unit). The optimizer the
        caller runs over shard_parameters() carries its own state_dict.
        Pair with utils.checkpoint.save_sharded_checkpoint; reload
        requires the SAME world size."""
        return {"units": [u.shard.detach().cpu() for u in self._units]}

    @torch.no_grad()
    def load_sharded_state_dict(self, state):
        for u, s in zip(self._units, state["units"]):
            if u.prefetch_handle is not None:
                self.comm.Wait(u.prefetch_handle)  # drain stale prefetch
                u.prefetch_handle = None
            u.shard.copy_(s.to(u.shard.device))
            # drop any materialized full buffer so the next use
            # re-allgathers from the restored shards
            u.free()

    def no_sync(self):
        """Gradient-accumulation context: local .grad accumulates; the
        final backward outside the context reduces the sums. (Parameters
        still materialize/free around each microbatch's forward/backward.)"""
        f = self

        class _NoSync:
            def __enter__(self):
                f._sync_enabled = False

            def __exit__(self, *exc):
                f._sync_enabled = True

        return _NoSync()

    def zero_grad(self, set_to_none: bool = True):
        for p in self.module.parameters():
            p.grad = None
