#!/usr/bin/env bash
# Yelp-shaped GAT (BASELINE.json config 4: SDDMM / edge-softmax path).
python main.py \
  --dataset yelp \
  --dropout 0.1 \
  --lr 0.01 \
  --n-partitions 8 \
  --n-epochs 2400 \
  --model gat \
  --heads 4 \
  --sampling-rate 0.1 \
  --n-layers 3 \
  --n-hidden 128 \
  --log-every 10 \
  --inductive
