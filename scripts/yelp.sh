#!/bin/bash
# Canonical Yelp config (reference: scripts/yelp.sh — multi-label BCE,
# 2 conv + 2 linear layers, hidden 512).
python main.py \
  --dataset yelp \
  --dropout 0.1 \
  --lr 0.001 \
  --n-partitions 3 \
  --n-epochs 3000 \
  --model graphsage \
  --n-layers 4 \
  --n-linear 2 \
  --n-hidden 512 \
  --log-every 10 \
  --inductive \
  --use-pp \
  --enable-pipeline \
  --backend nccl
