#!/usr/bin/env bash
# Benchmark matrix (BASELINE.json configs) on one node. The driver runs
# bench.py itself for BENCH/SCALE artifacts; this script is the human-run
# equivalent sweep. All rendezvous on 127.0.0.1.
set -euo pipefail
cd "$(dirname "$0")/.."

run() { echo "### $*"; "$@" | tail -1; }

# config 2/4: LeNet single GPU, fp32 + bf16 (batch 8192)
run python bench.py --gpus 1 --steps 50 --warmup 10 --dtype fp32
run python bench.py --gpus 1 --steps 50 --warmup 10 --dtype bf16

# config 3/4: DDP over RCCL/xGMI at 2/4/8 GPUs (weak scaling, 8192/GPU)
for n in 2 4 8; do
  run python -m torch.distributed.run --nnodes=1 --nproc-per-node $n \
      --master-addr 127.0.0.1 bench.py --gpus $n --steps 50 --warmup 10
done

# config 5: ResNet-18 synthetic 3x224x224
run python bench.py --gpus 1 --steps 20 --warmup 5 --model resnet18
for n in 2 4 8; do
  run python -m torch.distributed.run --nnodes=1 --nproc-per-node $n \
      --master-addr 127.0.0.1 bench.py --gpus $n --steps 20 --warmup 5 --model resnet18
done
