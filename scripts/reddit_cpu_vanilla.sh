#!/bin/bash
# BASELINE config #1: Reddit 2-partition vanilla (no pipeline) on CPU/gloo —
# plumbing check, runs without a GPU (synthetic Reddit shape when no
# dataset files are present).
python main.py \
  --dataset reddit \
  --n-partitions 2 \
  --n-epochs 10 \
  --model graphsage \
  --n-layers 4 \
  --n-hidden 256 \
  --log-every 5 \
  --no-eval \
  --backend gloo
