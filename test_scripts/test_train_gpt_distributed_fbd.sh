#!/bin/bash
# MegaFBD smoke: 4 ranks = (fwd,bwd) x PP2 (reference test_train_gpt_distributed_fbd.sh)
torchrun --nproc-per-node 4 --master-addr 127.0.0.1 pretrain_gpt.py \
    --num-layers 8 --hidden-size 512 --num-attention-heads 8 \
    --seq-length 1024 --max-position-embeddings 1024 \
    --micro-batch-size 2 --global-batch-size 16 \
    --pipeline-model-parallel-size 2 \
    --bf16 --mock-data --train-iters 10 --lr 1e-4 --log-interval 1 \
    --eval-iters 0 --forward-backward-disaggregating "$@"
