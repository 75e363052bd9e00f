#!/bin/bash
# Offline batch generation through the static engine (reference
# tools/text_generation_cli usage).
python tools/text_generation_cli.py \
    --num-layers 12 --hidden-size 768 --num-attention-heads 12 \
    --seq-length 512 --max-position-embeddings 1024 \
    --micro-batch-size 1 --bf16 --vocab-size 51200 \
    ${LOAD:+--load $LOAD} \
    --prompts "the quick brown fox" --tokens-to-generate 64 "$@"
