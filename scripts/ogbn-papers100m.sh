#!/bin/bash
# ogbn-papers100M, 8 partitions on one 8-GPU MI355X node (288 GB HBM3E per
# GPU: ~14M nodes + ~400M edges + halo per partition fits comfortably).
python main.py \
  --dataset ogbn-papers100m \
  --dropout 0.3 \
  --lr 0.003 \
  --n-partitions 8 \
  --n-epochs 100 \
  --model graphsage \
  --n-layers 3 \
  --n-hidden 256 \
  --log-every 10 \
  --no-eval \
  --use-pp \
  --enable-pipeline \
  --backend nccl
