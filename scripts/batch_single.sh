#!/bin/bash
# Sweep single-GPU configs (reference batch_single.sh).
for dnn in resnet20 resnet56 vgg16; do
  dnn=$dnn ./scripts/single.sh
done
