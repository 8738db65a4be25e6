"""Sequence-parallel (Ulysses) transformer block.

A composition showcase for the framework's primitives: activations stay
sequence-sharded [B, S/P, D] everywhere except inside attention, where two
autograd-transparent Alltoalls (seq_to_head / head_to_seq) expose the full
sequence to P-sharded heads. Weights are replicated — pair with
parallel.DistributedDataParallel (or an explicit grad allreduce) for
training.

The gradient story needs no extra code: each rank backwards the loss of
its own sequence shard, and the Alltoall adjoints route the cross-shard
attention contributions so x_local.grad equals the slice of the global
loss gradient exactly (verified against a dense single-process reference
in tests/test_models.py).
"""

import torch

import mpi4torch_amd as m4a
from mpi4torch_amd.parallel.ulysses import head_to_seq


class UlyssesTransformerBlock(torch.nn.Module):
    def __init__(self, d_model: int, n_heads: int, comm=None,
                 mlp_ratio: int = 4):
        super().__init__()
        self.comm = comm if comm is not None else m4a.COMM_WORLD
        assert n_heads % self.comm.size == 0, (
            f"n_heads {n_heads} must divide world size {self.comm.size}")
        assert d_model % n_heads == 0
        self.n_heads = n_heads
        self.head_dim = d_model // n_heads
        self.ln1 = torch.nn.LayerNorm(d_model)
        self.ln2 = torch.nn.LayerNorm(d_model)
        self.qkv = torch.nn.Linear(d_model, 3 * d_model)
        self.proj = torch.nn.Linear(d_model, d_model)
        self.mlp = torch.nn.Sequential(
            torch.nn.Linear(d_model, mlp_ratio * d_model),
            torch.nn.GELU(),
            torch.nn.Linear(mlp_ratio * d_model, d_model),
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """x: [B, S/P, D] (sequence-sharded) -> same shape."""
        B, S_local, D = x.shape
        h = self.ln1(x)
        qkv = self.qkv(h).view(B, S_local, 3, self.n_heads, self.head_dim)
        if self.comm.size > 1:
            # [B, S/P, 3, H, d] -> [B, S, 3, H/P, d]: full sequence,
            # sharded heads (one Alltoallv for q, k and v together; counts
            # are uniform and locally known — no host exchange)
            P = self.comm.size
            qkv = self.comm.Alltoallv(qkv, 1, 3, [self.n_heads // P] * P,
                                      [S_local] * P)
        q, k, v = qkv.unbind(2)
        attn = torch.nn.functional.scaled_dot_product_attention(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2)
        ).transpose(1, 2).contiguous()
        if self.comm.size > 1:
            # [B, S, H/P, d] -> [B, S/P, H, d]
            attn = head_to_seq(attn, self.comm)
        x = x + self.proj(attn.reshape(B, S_local, D))
        x = x + self.mlp(self.ln2(x))
        return x
