#!/bin/bash
# Canonical Reddit config (reference: scripts/reddit.sh — 2 partitions,
# 4-layer GraphSAGE, hidden 256, inductive, pipelined, use-pp).
# Backend nccl = RCCL over xGMI, one process per MI355X GPU.
python main.py \
  --dataset reddit \
  --dropout 0.5 \
  --lr 0.01 \
  --n-partitions 2 \
  --n-epochs 3000 \
  --model graphsage \
  --n-layers 4 \
  --n-hidden 256 \
  --log-every 10 \
  --inductive \
  --use-pp \
  --enable-pipeline \
  --backend nccl
