#!/usr/bin/env bash
# Multi-node launch (reference: scripts/reddit_multi_node.sh — which as
# shipped passes undefined flags; fixed here).
# Run on EVERY node, with NODE_RANK=0,1,... and MASTER_ADDR set:
#   MASTER_ADDR=10.0.0.1 NODE_RANK=0 bash scripts/reddit_multi_node.sh
#   MASTER_ADDR=10.0.0.1 NODE_RANK=1 bash scripts/reddit_multi_node.sh
# Node 0 must hold (or share) the partition store; other nodes use
# --skip-partition with a copied store (reference README.md:112-117).
python main.py \
  --dataset reddit \
  --n-partitions 16 \
  --parts-per-node 8 \
  --node-rank ${NODE_RANK:?set NODE_RANK} \
  --master-addr ${MASTER_ADDR:?set MASTER_ADDR} \
  --model graphsage --inductive --use-pp \
  --sampling-rate 0.1 --n-layers 4 --n-hidden 256 \
  --n-epochs 3000 --log-every 10 \
  $([ "${NODE_RANK}" != "0" ] && echo --skip-partition)
