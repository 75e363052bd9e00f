#!/bin/bash
# GPT-3 1.3B PP=4 — the BASELINE headline config; add --use-dpp for the
# MegaDPP dynamic pipeline schedule (reference train_gpt3_175b_distributed.sh).
torchrun --nproc-per-node 4 --master-addr 127.0.0.1 pretrain_gpt.py \
    --num-layers 24 --hidden-size 2048 --num-attention-heads 16 \
    --seq-length 2048 --max-position-embeddings 2048 \
    --micro-batch-size 2 --global-batch-size 16 \
    --pipeline-model-parallel-size 4 \
    --bf16 --mock-data --train-iters 50 --lr 1e-4 \
    --log-interval 5 --eval-iters 0 "$@"
