"""Expert-parallel mixture-of-experts layer on AlltoallPairwise.

The BASELINE config #5 pattern ("expert-parallel alltoall") as a working
layer: experts are sharded across ranks; each token is routed (top-1,
softmax-gated) to its expert's owner with one pairwise-count Alltoall,
processed, and combined back with the reverse Alltoall. Both transfers are
autograd-transparent, so router and expert gradients need no custom
backward — the dispatch's adjoint returns token gradients to their
senders automatically.
"""

import torch

import mpi4torch_amd as m4a


class ExpertParallelMoE(torch.nn.Module):
    def __init__(self, d_model: int, n_experts: int, comm=None,
                 d_hidden: int = None):
        super().__init__()
        self.comm = comm if comm is not None else m4a.COMM_WORLD
        P = self.comm.size
        assert n_experts % P == 0, (
            f"n_experts {n_experts} must divide world size {P}")
        self.n_experts = n_experts
        self.experts_per_rank = n_experts // P
        d_hidden = d_hidden or 4 * d_model
        self.router = torch.nn.Linear(d_model, n_experts)
        # this rank's experts
        self.experts = torch.nn.ModuleList([
            torch.nn.Sequential(
                torch.nn.Linear(d_model, d_hidden),
                torch.nn.GELU(),
                torch.nn.Linear(d_hidden, d_model),
            )
            for _ in range(self.experts_per_rank)
        ])

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """x: [N, D] local tokens -> [N, D]."""
        N, D = x.shape
        P = self.comm.size
        logits = self.router(x)
        gates = torch.softmax(logits, dim=-1)
        expert = torch.argmax(gates, dim=-1)          # [N], global expert id
        gate = gates.gather(1, expert.unsqueeze(1)).squeeze(1)  # [N]
        owner = expert // self.experts_per_rank       # destination rank

        # sort tokens by destination rank (stable: preserves order per dest)
        order = torch.argsort(owner, stable=True)
        inverse = torch.empty_like(order)
        inverse[order] = torch.arange(N, device=x.device)
        x_sorted = x.index_select(0, order)
        send_counts = torch.bincount(owner, minlength=P).tolist()

        # one count-matrix exchange serves both transfers explicitly
        recv_counts = self._recv_counts(send_counts)

        # dispatch tokens and (gradient-free) expert ids to the owners
        dispatched = self.comm.AlltoallPairwise(x_sorted, 0, send_counts,
                                                recv_counts)
        eid_sorted = expert.index_select(0, order).to(torch.float64)
        eid = self.comm.AlltoallPairwise(
            eid_sorted.unsqueeze(1), 0, send_counts,
            recv_counts).squeeze(1).long()
        local_eid = eid - self.comm.rank * self.experts_per_rank

        # run this rank's experts on their token groups
        out = torch.zeros_like(dispatched)
        for e in range(self.experts_per_rank):
            mask = local_eid == e
            if bool(mask.any()):
                idx = mask.nonzero(as_tuple=True)[0]
                out = out.index_copy(
                    0, idx, self.experts[e](dispatched.index_select(0, idx)))

        # combine: the reverse pairwise transfer (counts transpose)
        combined = self.comm.AlltoallPairwise(out, 0, recv_counts,
                                              send_counts)

        y = combined.index_select(0, inverse)
        return y * gate.unsqueeze(1)

    def _recv_counts(self, send_counts):
        """Dispatch recv counts = my column of the P x P count matrix;
        one exchange serves dispatch, id transfer and combine (counts are
        data-dependent — recomputed every forward)."""
        P = self.comm.size
        if P == 1:
            return list(send_counts)
        t = torch.tensor(send_counts, dtype=torch.float64)
        mat = self.comm.Allgather(t.unsqueeze(0), 0)  # [P, P]
        return [int(v) for v in mat[:, self.comm.rank].tolist()]
