#!/bin/bash
# Llama-3-70B on one node: TP=8 + sequence parallelism (288 GB HBM3E/GPU
# holds the shards without pipeline stages at short context).
set -euo pipefail
HSA_ENABLE_IPC_MODE_LEGACY=0 torchrun --nproc-per-node 8 --master-addr 127.0.0.1 \
  pretrain_gpt.py \
  --num-layers 80 --hidden-size 8192 --num-attention-heads 64 \
  --num-query-groups 8 --ffn-hidden-size 28672 --vocab-size 128256 \
  --tensor-model-parallel-size 8 --sequence-parallel \
  --seq-length 4096 --micro-batch-size 1 --global-batch-size 16 --bf16 \
  --use-distributed-optimizer --recompute-granularity selective \
  --mock-data --train-iters 100 "$@"
