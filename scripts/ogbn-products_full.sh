#!/usr/bin/env bash
# Full sweep (reference: scripts/ogbn-products_full.sh): partitions x
# sampling rates, per-config output tee'd to results/. The reference's
# pkill-based process cleanup is intentionally dropped (pattern kills are
# unsafe); our launcher tears its workers down cleanly.
mkdir -p results
for N_PARTITIONS in 5 8 10; do
  for SAMPLING_RATE in 0.10 0.01 0.00; do
    echo "== ${N_PARTITIONS} partitions, ${SAMPLING_RATE} sampling rate =="
    python main.py \
      --dataset ogbn-products \
      --dropout 0.3 \
      --lr 0.003 \
      --n-partitions ${N_PARTITIONS} \
      --n-epochs 500 \
      --model graphsage \
      --sampling-rate ${SAMPLING_RATE} \
      --n-layers 3 \
      --n-hidden 128 \
      --log-every 10 \
      --use-pp \
      |& tee results/ogbn-products_n${N_PARTITIONS}_p${SAMPLING_RATE}_full.txt
  done
done
