#!/bin/bash
# Multi-node Reddit (reference: scripts/reddit_multi_node.sh).
# Run on EVERY node with its own --node-rank; node 0 partitions first.
# Requires --fix-seed so all nodes agree on the model init.
: "${MASTER_ADDR:?set MASTER_ADDR to node 0's address}"
: "${NODE_RANK:?set NODE_RANK (0-based)}"
python main.py \
  --dataset reddit \
  --dropout 0.5 \
  --lr 0.01 \
  --n-partitions 16 \
  --parts-per-node 8 \
  --node-rank "$NODE_RANK" \
  --master-addr "$MASTER_ADDR" \
  --n-epochs 3000 \
  --model graphsage \
  --n-layers 4 \
  --n-hidden 256 \
  --log-every 10 \
  --inductive \
  --fix-seed \
  --use-pp \
  --enable-pipeline \
  --backend nccl
