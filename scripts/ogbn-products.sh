#!/bin/bash
# Canonical ogbn-products config (reference: scripts/ogbn-products.sh).
python main.py \
  --dataset ogbn-products \
  --dropout 0.3 \
  --lr 0.003 \
  --n-partitions 5 \
  --n-epochs 500 \
  --model graphsage \
  --n-layers 3 \
  --n-hidden 128 \
  --log-every 5 \
  --use-pp \
  --enable-pipeline \
  --backend nccl
