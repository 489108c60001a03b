#!/bin/bash
# Finetune from a pretrained checkpoint: weights only, fresh optimizer and
# schedule (--finetune), document-boundary attention reset + EOD loss mask.
set -e
torchrun --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 pretrain_gpt.py \
  --load ckpts/llama3-8b --finetune \
  --num-layers 32 --hidden-size 4096 --num-attention-heads 32 --num-query-groups 8 \
  --ffn-hidden-size 14336 --vocab-size 128256 --seq-length 4096 \
  --max-position-embeddings 8192 --rotary-base 500000 --swiglu \
  --untie-embeddings-and-output-weights \
  --data-path 1.0 /data/sft_corpus --reset-attention-mask --eod-mask-loss \
  --micro-batch-size 1 --global-batch-size 32 --train-iters 2000 \
  --bf16 --use-distributed-optimizer --lr 2e-5 --lr-decay-style cosine \
  --lr-warmup-iters 50 --clip-grad 1.0 \
  --log-interval 10 --save ckpts/llama3-8b-sft --save-interval 500
