#!/bin/bash
# One-command validation: everything that must hold before shipping.
# CPU-only parts run anywhere; GPU parts run when a GPU is visible.
#   bash tools/release_check.sh
set -e
cd "$(dirname "$0")/.."

echo "== build (hipcc gfx950 cross-compile + dispatcher) =="
python -c "import __graft_entry__ as g; g.build()"

echo "== CPU test suite =="
python -m pytest tests -x -q -m "not gpu" -p no:warnings

echo "== bench contracts (engine, 2-rank gloo, full stack) =="
python bench.py | tail -1
python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 --master-port 29741 bench.py --gpus 2 \
    --steps 8 --warmup 2 2>/dev/null | grep '"metric"'
python bench.py --stack --steps 24 --warmup 4 | tail -1

if python -c "import torch,sys; sys.exit(0 if torch.cuda.is_available() else 1)"; then
  echo "== GPU test suite + smoke =="
  python -m pytest tests -x -q -m gpu -p no:warnings
  python -c "import __graft_entry__ as g; g.smoke(); print('smoke OK')"
  echo "== GPU bench =="
  python bench.py --steps 25 | tail -1
fi
echo "RELEASE CHECK OK"
