#!/usr/bin/env bash
# BD-BNN ImageNet recipe, step 1 (binary activations, ReActNet-style),
# 8 GPUs over RCCL/xGMI.  Produces the checkpoint step 2 resumes from.
set -ex
DATA=${1:-/data/imagenet}
torchrun --standalone --nproc-per-node 8 train.py "$DATA" \
    --dataset imagenet -a resnet18_react -b 2048 --epochs 90 -lr 1e-3 \
    --amp --log_path log/step1 "$@"
