#!/bin/bash
# The reference's headline A/B, reproduced on an 8-GPU MI355X node
# (reference batch_dist_mpi.sh:2: MG-WFBP vs WFBP vs single-group).
# Run from the repo root on a box with 8 GPUs:
#   bash benchmarks/ab_8gpu.sh [model] [dataset] [batch]
set -e
model="${1:-resnet50}"
dataset="${2:-imagenet}"
batch="${3:-128}"
for merge in mgwfbp wfbp single; do
  for n in 8 4 2 1; do
    echo "=== $model $merge N=$n ==="
    python -m torch.distributed.run --nnodes=1 --nproc-per-node $n \
      --master-addr 127.0.0.1 --master-port 29533 \
      bench.py --gpus $n --steps 40 --warmup 10 \
      --model "$model" --dataset "$dataset" --batch-size "$batch" \
      --merge $merge 2>/dev/null | grep '^{'
  done
done
