#!/bin/bash
# Hybrid Mamba2/attention model (9:1 SSD:attention layers).
set -euo pipefail
HSA_ENABLE_IPC_MODE_LEGACY=0 torchrun --nproc-per-node 8 --master-addr 127.0.0.1 \
  pretrain_mamba.py \
  --num-layers 48 --hidden-size 4096 --num-attention-heads 32 \
  --num-query-groups 8 --ffn-hidden-size 14336 --vocab-size 128256 \
  --hybrid-attention-ratio 0.1 --hybrid-mlp-ratio 0.3 \
  --seq-length 4096 --micro-batch-size 4 --global-batch-size 128 --bf16 \
  --use-distributed-optimizer --mock-data --train-iters 1000 "$@"
