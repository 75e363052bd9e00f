#!/bin/bash
# Mixtral-style MoE pretraining: 8 experts, top-2 router with aux loss,
# expert parallelism over the DP group (mirrors the reference
# examples/mixtral usage).
GPUS_PER_NODE=${GPUS_PER_NODE:-1}
torchrun --nproc-per-node $GPUS_PER_NODE --master-addr 127.0.0.1 \
    pretrain_gpt.py \
    --num-layers 16 --hidden-size 2048 --num-attention-heads 32 \
    --group-query-attention --num-query-groups 8 \
    --ffn-hidden-size 8192 --swiglu --normalization RMSNorm \
    --position-embedding-type rope --disable-bias-linear \
    --untie-embeddings-and-output-weights \
    --num-experts 8 --moe-router-topk 2 \
    --moe-router-load-balancing-type aux_loss --moe-aux-loss-coeff 1e-2 \
    --expert-model-parallel-size ${EP:-1} \
    --seq-length 2048 --max-position-embeddings 2048 \
    --micro-batch-size 1 --global-batch-size 16 \
    --bf16 --mock-data --train-iters 50 --lr 1e-4 \
    --log-interval 5 --eval-iters 0 "$@"
