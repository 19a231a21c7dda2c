#!/usr/bin/env bash
# Full benchmark sweep on one MI355X (the driver runs bench.py with its
# own flags; this reproduces the numbers in benchmarks/RESULTS.md).
set -ex
python bench.py --steps 12 --warmup 3 --batch-size 512
python bench.py --steps 12 --warmup 3 --batch-size 2048
python bench.py --ts --steps 10 --warmup 3 --batch-size 1024
python bench.py --arch resnet34 --steps 8 --warmup 3 --batch-size 256
python bench.py --infer --steps 40 --warmup 5 --batch-size 256
python benchmarks/kernel_bench.py
