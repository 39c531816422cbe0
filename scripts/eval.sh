#!/bin/bash
# Checkpoint evaluation (reference scripts/eval.sh):
#   ./scripts/eval.sh ./weights/<prefix>/resnet20-n8-bs32-lr0.1000 cifar10
exec python -m mgwfbp_amd.evaluate --weights-dir "$1" --dataset "${2:-cifar10}" --data-dir "${3:-}"
