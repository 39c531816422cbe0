#!/bin/bash
# Distributed launch (replaces the reference's mpirun dist_mpi.sh:12-16):
# torchrun env-var rendezvous, one rank per GPU over RCCL/xGMI.
#
#   dnn=resnet50 nworkers=8 ./scripts/dist.sh
#
# Extra env knobs: threshold (merge threshold elems; unset => MG-WFBP
# solver), MGX_* settings (see mgwfbp_amd/settings.py).
dnn="${dnn:-resnet20}"
source "$(dirname "$0")/../exp_configs/${dnn}.conf"
nworkers="${nworkers:-8}"
threshold="${threshold:-0}"
extra=""
if [ -n "$threshold" ] && [ "$threshold" != "0" ]; then
  export MGX_ADAPTIVE_MERGE=0
  extra="--threshold $threshold"
fi
exec python -m torch.distributed.run \
  --nnodes=1 --nproc-per-node "$nworkers" \
  --master-addr 127.0.0.1 --master-port "${master_port:-29500}" \
  -m mgwfbp_amd.dist_trainer \
  --dnn "$dnn" --dataset "$dataset" --data-dir "$data_dir" \
  --lr "$lr" --batch-size "$batch_size" --nsteps-update "$nstepsupdate" \
  --max-epochs "$max_epochs" --nworkers "$nworkers" $extra "$@"
