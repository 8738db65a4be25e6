"""Ulysses-style sequence-parallel attention via Alltoall resharding.

Each rank holds a sequence shard [B, S/P, H, D]; the two Alltoalls move it
to full-sequence/sharded-heads for attention and back. Both reshards are
autograd-transparent, so backward reshards gradients automatically.

Run: torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 2 \
        examples/ulysses_attention.py
"""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import mpi4torch_amd as m4a
from mpi4torch_amd.parallel import seq_to_head, head_to_seq

device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
comm = m4a.COMM_WORLD

B, S, H, D = 2, 16 * comm.size, 4 * comm.size, 32
local = torch.randn(B, S // comm.size, H, D, device=device).requires_grad_()

q = seq_to_head(local)          # [B, S, H/P, D]
k, v = q, q
attn = torch.nn.functional.scaled_dot_product_attention(
    q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2)
).transpose(1, 2)
out = head_to_seq(attn.contiguous())   # back to [B, S/P, H, D]

out.sum().backward()
print(f"rank {comm.rank}: out {tuple(out.shape)}, grad norm "
      f"{float(local.grad.norm()):.3f}")
