#!/bin/bash
# LLaVA-style multimodal pretraining (CLIP ViT encoder + projector + LLM),
# synthetic images in mock mode.
set -euo pipefail
HSA_ENABLE_IPC_MODE_LEGACY=0 torchrun --nproc-per-node 8 --master-addr 127.0.0.1 \
  pretrain_vlm.py \
  --num-layers 32 --hidden-size 4096 --num-attention-heads 32 \
  --num-query-groups 8 --ffn-hidden-size 14336 --vocab-size 128256 \
  --seq-length 2048 --micro-batch-size 2 --global-batch-size 64 --bf16 \
  --use-distributed-optimizer --mock-data --train-iters 1000 "$@"
