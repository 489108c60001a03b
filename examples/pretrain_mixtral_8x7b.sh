#!/bin/bash
# Mixtral-8x7B-shape MoE pretraining: EP=8 over single-hop xGMI all-to-all.
set -euo pipefail
HSA_ENABLE_IPC_MODE_LEGACY=0 torchrun --nproc-per-node 8 --master-addr 127.0.0.1 \
  pretrain_gpt.py \
  --num-layers 32 --hidden-size 4096 --num-attention-heads 32 \
  --num-query-groups 8 --ffn-hidden-size 14336 --vocab-size 32000 \
  --num-experts 8 --moe-router-topk 2 --moe-aux-loss-coeff 0.01 \
  --expert-model-parallel-size 8 \
  --seq-length 4096 --micro-batch-size 2 --global-batch-size 64 --bf16 \
  --use-distributed-optimizer --mock-data --train-iters 1000 "$@"
