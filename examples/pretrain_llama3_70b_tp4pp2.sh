#!/bin/bash
# Llama-3-70B on one node: TP=4 x PP=2 interleaved 1F1B (VPP=2), SP on.
set -e
torchrun --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 pretrain_gpt.py \
  --num-layers 80 --hidden-size 8192 --num-attention-heads 64 --num-query-groups 8 \
  --ffn-hidden-size 28672 --vocab-size 128256 --seq-length 4096 \
  --max-position-embeddings 8192 --rotary-base 500000 --swiglu \
  --untie-embeddings-and-output-weights \
  --tensor-model-parallel-size 4 --pipeline-model-parallel-size 2 \
  --virtual-pipeline-model-parallel-size 2 --sequence-parallel \
  --micro-batch-size 1 --global-batch-size 32 --train-iters 1000 \
  --bf16 --use-distributed-optimizer --recompute-granularity selective \
  --lr 1.5e-4 --clip-grad 1.0 --mock-data --log-interval 5
