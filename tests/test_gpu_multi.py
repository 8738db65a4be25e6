"""Multi-GPU SPMD battery: launches tests/gpu_spmd_worker.py under torchrun
with one rank per GPU over RCCL/xGMI.

- `pytest -m gpu` on a multi-GPU box runs the full battery (forward
  values, closed-form adjoints, Scatter∘Gather / Alltoall∘Alltoall
  identities, the 64 MiB grouped-p2p ring, fp8/bitwise/pairloc
  hierarchical allreduce, Iallreduce overlap, comm_split, DDP) on CUDA
  tensors through the RCCL transport. On a single-GPU box it skips.
- The plain CPU suite runs the SAME battery and the SAME torchrun launch
  path at nproc=2 on gloo (test_gpu_multi_launch_cpu_dryrun), so the only
  untested delta before a multi-GPU box is the RCCL transport itself.
"""

import os
import socket
import subprocess
import sys

import pytest
import torch

WORKER = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      "gpu_spmd_worker.py")


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _launch(nproc: int, timeout: int, extra_env=None):
    env = os.environ.copy()
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    # a fresh store port; the container hostname may not resolve
    env.pop("MASTER_ADDR", None)
    env.pop("MASTER_PORT", None)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    env.pop("LOCAL_RANK", None)
    if extra_env:
        env.update(extra_env)
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", f"--nproc-per-node={nproc}",
        "--master-addr", "127.0.0.1",
        "--master-port", str(_free_port()),
        WORKER,
    ]
    proc = subprocess.run(cmd, env=env, timeout=timeout,
                          capture_output=True, text=True)
    if proc.returncode != 0:
        raise AssertionError(
            f"gpu_spmd_worker failed (rc={proc.returncode})\n"
            f"--- stdout ---\n{proc.stdout[-8000:]}\n"
            f"--- stderr ---\n{proc.stderr[-8000:]}"
        )
    assert "ALL SECTIONS PASSED" in proc.stdout, proc.stdout[-4000:]


@pytest.mark.gpu
def test_gpu_multi_spmd():
    n = torch.cuda.device_count()
    if n < 2:
        pytest.skip("multi-GPU battery needs >=2 GPUs on this box")
    _launch(min(n, 8), timeout=1200)


def test_gpu_multi_launch_cpu_dryrun():
    # same worker, same torchrun launch path, CPU/gloo at nproc=2 — the
    # de-risking dry run for the multi-GPU battery
    _launch(2, timeout=600)
