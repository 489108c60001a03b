#!/bin/bash
# Llama-3-8B pretraining on one 8-GPU MI355X node (DP=8, ZeRO-1).
# Mock data; point --data-path at preprocessed MMIDIDX corpora for real runs.
set -euo pipefail
HSA_ENABLE_IPC_MODE_LEGACY=0 torchrun --nproc-per-node 8 --master-addr 127.0.0.1 \
  pretrain_gpt.py \
  --num-layers 32 --hidden-size 4096 --num-attention-heads 32 \
  --num-query-groups 8 --ffn-hidden-size 14336 --vocab-size 128256 \
  --seq-length 4096 --max-position-embeddings 8192 --rotary-base 500000 \
  --micro-batch-size 4 --global-batch-size 128 --bf16 \
  --use-distributed-optimizer --overlap-param-gather \
  --lr 3e-4 --min-lr 3e-5 --lr-decay-style cosine --lr-warmup-iters 200 \
  --mock-data --train-iters 1000 --log-interval 10 "$@"
