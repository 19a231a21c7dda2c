#!/usr/bin/env bash
# BD-BNN CIFAR-10 recipe (kurtosis regularization + EDE), 1 GPU.
# Data: put cifar-10-batches-py under $1 (default ./data), or add
# --synthetic-data to smoke-test the pipeline without the dataset.
set -ex
DATA=${1:-./data}
python train.py "$DATA" --dataset cifar10 -a resnet20 \
    -b 256 --epochs 400 -lr 0.1 --momentum 0.9 -wd 1e-4 \
    --w-kurtosis --weight-name all --kurtosis-mode avg \
    --w-kurtosis-target 1.8 --ede --amp --seed 1 "$@"
