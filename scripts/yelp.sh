#!/usr/bin/env bash
# Reproduction script (reference: scripts/yelp.sh) — Yelp-shaped
# GraphSAGE (multilabel BCE), 3 partitions, inductive, MLP tail.
python main.py \
  --dataset yelp \
  --dropout 0.1 \
  --lr 0.01 \
  --n-partitions 3 \
  --n-epochs 2400 \
  --model graphsage \
  --sampling-rate 0.1 \
  --n-layers 3 \
  --n-hidden 512 \
  --n-linear 1 \
  --log-every 10 \
  --inductive \
  --use-pp
