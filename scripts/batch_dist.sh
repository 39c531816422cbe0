#!/bin/bash
# A/B sweep (reference batch_dist_mpi.sh): MG-WFBP vs single-group
# (threshold=536870912 elems) vs WFBP (threshold=0), over worker counts.
for dnn in resnet50 vgg16i resnet20; do
  for nworkers in 8 4 2; do
    # MG-WFBP (solver)
    dnn=$dnn nworkers=$nworkers ./scripts/dist.sh --max-iters 200
    # WFBP (per-layer)
    MGX_ADAPTIVE_MERGE=0 threshold=0 dnn=$dnn nworkers=$nworkers \
      ./scripts/dist.sh --max-iters 200
    # single-group
    MGX_ADAPTIVE_MERGE=0 threshold=536870912 dnn=$dnn nworkers=$nworkers \
      ./scripts/dist.sh --max-iters 200
  done
done
