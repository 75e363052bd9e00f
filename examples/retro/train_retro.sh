#!/bin/bash
# Retro pretraining (reference examples/retro).  Build a retrieval
# project first (python -c "from tools.retro.preprocess import ..."),
# or omit --retro-project-dir to train on synthetic neighbors.
GPUS_PER_NODE=${GPUS_PER_NODE:-1}
torchrun --nproc-per-node $GPUS_PER_NODE --master-addr 127.0.0.1 \
    pretrain_retro.py \
    --num-layers 12 --hidden-size 768 --num-attention-heads 12 \
    --seq-length 512 --max-position-embeddings 1024 \
    --retro-chunk-length 64 --retro-num-neighbors 2 \
    --micro-batch-size 2 --global-batch-size 8 \
    --bf16 --train-iters 50 --lr 1e-4 \
    --log-interval 5 --eval-iters 0 "$@"
