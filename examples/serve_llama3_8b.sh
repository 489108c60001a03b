#!/bin/bash
# Continuous-batching REST serving (paged KV + hipGraph decode steps).
set -euo pipefail
python tools/run_text_generation_server.py \
  --load "$1" \
  --num-layers 32 --hidden-size 4096 --num-attention-heads 32 \
  --num-query-groups 8 --ffn-hidden-size 14336 --vocab-size 128256 \
  --bf16 --port 5000
