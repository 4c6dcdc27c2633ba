#!/usr/bin/env bash
# Reproduction script (reference: scripts/yelp.sh) — Yelp-shaped
# GraphSAGE (multilabel BCE), 3 partitions, inductive, 4 layers with a
# 2-layer MLP tail, reference hyperparameters.
python main.py \
  --dataset yelp \
  --dropout 0.1 \
  --weight-decay 0 \
  --lr 0.001 \
  --n-partitions 3 \
  --n-epochs 3000 \
  --model graphsage \
  --sampling-rate 0.1 \
  --n-layers 4 \
  --n-linear 2 \
  --n-hidden 512 \
  --log-every 10 \
  --inductive \
  --use-pp
