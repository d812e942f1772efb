#!/usr/bin/env bash
# DDP bucket-size sweep: measures images/s of
# the multi-bucket ResNet-18 config across bucket sizes and comm backends
# on an N-GPU node. The LeNet config is one 87 KB bucket at every setting
# (latency-bound collective), so the sweep matters for ResNet only.
#
#   bash benchmarks/bucket_sweep.sh [NGPUS]
#
# xGMI reasoning for the candidate set: a ring all-reduce is per-link bound
# (7 p2p links x ~153 GB/s); ResNet-18's 46.8 MB fp32 gradient at 8 ranks
# moves 2*(N-1)/N * 46.8 MB per link — bucket sizes below ~4 MB go
# latency-bound per collective, above ~50 MB lose backward overlap.
set -euo pipefail
cd "$(dirname "$0")/.."
N="${1:-8}"

run() { echo "### $*"; "$@" 2>/dev/null | tail -1; }

for mb in 4 8 16 25 50 100; do
  run python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
      --master-addr 127.0.0.1 --master-port 2971 bench.py --gpus "$N" \
      --steps 20 --warmup 5 --model resnet18 --bucket-mb "$mb"
done

# native C++ RCCL core vs torch ProcessGroupNCCL at the default bucket
for comm in torch native; do
  run python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
      --master-addr 127.0.0.1 --master-port 2971 bench.py --gpus "$N" \
      --steps 20 --warmup 5 --model resnet18 --comm "$comm"
  run python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
      --master-addr 127.0.0.1 --master-port 2971 bench.py --gpus "$N" \
      --steps 30 --warmup 8 --comm "$comm"
done
