#!/usr/bin/env bash
# One-command validation: build -> CPU tier -> (on a GPU box) GPU tier + smoke
# + quick benches. CPU-only machines stop after the CPU tier.
set -euo pipefail
cd "$(dirname "$0")/.."

python tools/build_ext.py
python -m pytest tests -q -m "not gpu"

if python -c "import torch,sys; sys.exit(0 if torch.cuda.is_available() else 1)"; then
  python -m pytest tests -q -m gpu
  python __graft_entry__.py smoke
  python bench.py --steps 20 --warmup 5 | tail -1
  python bench.py --steps 10 --warmup 3 --model resnet18 --batch 128 | tail -1
else
  echo "(no GPU: stopped after CPU tier; run the rest via gpurun)"
fi
