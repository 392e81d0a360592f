#!/usr/bin/env bash
# One-command local verification: build the gfx950 extension, run the CPU
# suite, and smoke the bench contract. GPU-tier checks (pytest -m gpu,
# bench.py on device) run on an MI355X box.
set -euo pipefail
cd "$(dirname "$0")/.."

echo "== build (gfx950 cross-compile) =="
PYTORCH_ROCM_ARCH=gfx950 python -c "import __graft_entry__ as g; g.build()"

echo "== CPU test suite =="
python -m pytest tests/ -x -q -m "not gpu"

echo "== bench contract (CPU smoke + torchrun world-2) =="
python bench.py --cpu --steps 2 --warmup 1 | tail -1
python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 --master-port 29759 \
    bench.py --gpus 2 --steps 2 --warmup 1 --cpu | tail -1

echo "selfcheck OK"
