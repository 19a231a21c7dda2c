#!/usr/bin/env bash
# BD-BNN ImageNet recipe, step 2: 1W/1A student distilled from the fp32
# teacher (logit KD + weight-space KD + kurtosis with the per-layer
# diffkurt targets) — the paper's headline configuration.
set -ex
DATA=${1:-/data/imagenet}
STEP1=${2:-log/step1/1.8/latest/checkpoint.pth.tar}
# the fp32 teacher MUST come from a real checkpoint — no pretrained
# weights ship in this offline image (distilling from random init is a
# bug, not a recipe)
TEACHER=${3:-log/teacher/checkpoint.pth.tar}
torchrun --standalone --nproc-per-node 8 train.py "$DATA" \
    --dataset imagenet -a resnet18 -b 2048 --epochs 90 -lr 1e-3 \
    --imagenet_setting_step_2_ts -a_teacher resnet18 \
    --resume_teacher "$TEACHER" \
    --alpha 0.9 --beta 200 \
    --w-kurtosis --weight-name all --diffkurt \
    --resume "$STEP1" --reset_resume \
    --amp --log_path log/step2 "$@"
