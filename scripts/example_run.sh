#!/usr/bin/env bash
# The reference's known-good GCN recipe (example_run.sh:2 / test.sh:8),
# on this framework. Uses the Reddit-shaped synthetic dataset (no network
# for the real reddit-dgl files); pass --file <prefix> --layers 602-256-41
# to train on a real .lux dataset instead.
set -e
python train.py \
  --dataset reddit-synthetic \
  --model gcn --hidden 256 --num-layers 2 \
  --lr 0.01 --weight-decay 0.0001 --decay-rate 0.97 --decay-steps 100 \
  --dropout 0.5 --seed 1 --epochs 3000 --eval-every 5 "$@"
