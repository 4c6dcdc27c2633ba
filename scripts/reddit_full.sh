#!/usr/bin/env bash
# Full sweep (reference: scripts/reddit_full.sh): partitions x sampling
# rates, outputs tee'd to results/ (reproduces the paper tables).
mkdir -p results
for P in 2 4 8; do
  for RATE in 0.1 0.01 0.0; do
    python main.py \
      --dataset reddit --model graphsage --inductive --use-pp \
      --n-partitions $P --sampling-rate $RATE \
      --n-layers 4 --n-hidden 256 --n-epochs 3000 --log-every 10 \
      | tee -a results/reddit_full.txt
  done
done
