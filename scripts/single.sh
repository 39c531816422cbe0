#!/bin/bash
# Single-GPU training (reference single.sh): the 1-GPU scaling point.
#   dnn=resnet20 ./scripts/single.sh
dnn="${dnn:-resnet20}"
source "$(dirname "$0")/../exp_configs/${dnn}.conf"
exec python -c "
from mgwfbp_amd.dl_trainer import train_with_single
train_with_single('$dnn', '$dataset', '$data_dir', 1, $lr, $batch_size,
                  $nstepsupdate, $max_epochs)
"
