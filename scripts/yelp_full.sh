#!/usr/bin/env bash
# Full sweep (reference: scripts/yelp_full.sh): partitions x sampling
# rates, per-config output tee'd to results/ (pkill cleanup dropped —
# see ogbn-products_full.sh).
mkdir -p results
for N_PARTITIONS in 3 6 10; do
  for SAMPLING_RATE in 0.10 0.01 0.00; do
    echo "== ${N_PARTITIONS} partitions, ${SAMPLING_RATE} sampling rate =="
    python main.py \
      --dataset yelp \
      --dropout 0.1 \
      --lr 0.001 \
      --n-partitions ${N_PARTITIONS} \
      --n-epochs 3000 \
      --model graphsage \
      --sampling-rate ${SAMPLING_RATE} \
      --n-layers 4 \
      --n-linear 2 \
      --n-hidden 512 \
      --log-every 10 \
      --inductive \
      --use-pp \
      |& tee results/yelp_n${N_PARTITIONS}_p${SAMPLING_RATE}_full.txt
  done
done
