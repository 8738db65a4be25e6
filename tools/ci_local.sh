#!/usr/bin/env bash
# Local mirror of .github/workflows/test.yml — the same steps the hosted CI
# would run, executable in the offline container (and on a GPU box with
# M4A_CI_GPU=1). This is the exercised form of the CI contract; the hosted
# workflow re-uses these steps verbatim once a runner exists.
set -euo pipefail
cd "$(dirname "$0")/.."

echo "=== [ci] build extension (gfx950 cross-compile) ==="
python tools/build_ext.py

echo "=== [ci] CPU SPMD suite (gloo worlds 2/5/7, pytest -m 'not gpu') ==="
python -m pytest tests -q -m "not gpu"

echo "=== [ci] torchrun example smoke (2 ranks) ==="
torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 2 \
    examples/isend_irecv_wait.py

echo "=== [ci] torchrun example smoke (3 ranks, linreg) ==="
torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 3 \
    examples/simple_linear_regression.py

if [ "${M4A_CI_GPU:-0}" = "1" ]; then
  echo "=== [ci] GPU suite (pytest -m gpu) ==="
  python -m pytest tests -q -m gpu
  echo "=== [ci] bench (1 GPU) ==="
  python bench.py --gpus 1 --steps 20 --warmup 5
fi

echo "=== [ci] ALL STEPS PASSED ==="
