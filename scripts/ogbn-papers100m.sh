#!/usr/bin/env bash
# ogbn-papers100M-shaped run (BASELINE.json config 5): random partition,
# 8 partitions, 288 GB HBM sizing. --data-scale shrinks the synthetic
# graph for smoke runs (1.0 = full 111M nodes / 1.6B edges).
SCALE=${SCALE:-1.0}
python main.py \
  --dataset ogbn-papers100M \
  --data-scale $SCALE \
  --dropout 0.1 \
  --lr 0.01 \
  --n-partitions 8 \
  --n-epochs 100 \
  --model graphsage \
  --sampling-rate 0.01 \
  --partition-method random \
  --n-layers 3 \
  --n-hidden 128 \
  --log-every 10 \
  --no-eval \
  --use-pp
