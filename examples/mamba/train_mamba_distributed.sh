#!/bin/bash
# Mamba pretraining (reference examples/mamba): pure-SSM stack, or pass
# --hybrid-pattern 'MMM*MMM*MMM*' for a hybrid attention/mamba stack.
GPUS_PER_NODE=${GPUS_PER_NODE:-1}
torchrun --nproc-per-node $GPUS_PER_NODE --master-addr 127.0.0.1 \
    pretrain_mamba.py \
    --num-layers 12 --hidden-size 768 --num-attention-heads 12 \
    --normalization RMSNorm \
    --seq-length 1024 --max-position-embeddings 1024 \
    --micro-batch-size 4 --global-batch-size 16 \
    --bf16 --mock-data --train-iters 50 --lr 1e-4 \
    --log-interval 5 --eval-iters 0 "$@"
