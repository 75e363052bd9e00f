#!/bin/bash
# BERT-large-class masked-LM pretraining (mirrors the reference
# examples/bert/train_bert_340m_distributed.sh usage; mock data since the
# image has no network).
GPUS_PER_NODE=${GPUS_PER_NODE:-1}
torchrun --nproc-per-node $GPUS_PER_NODE --master-addr 127.0.0.1 \
    pretrain_bert.py \
    --num-layers 24 --hidden-size 1024 --num-attention-heads 16 \
    --seq-length 512 --max-position-embeddings 512 \
    --micro-batch-size 4 --global-batch-size 32 \
    --bf16 --mock-data --train-iters 50 --lr 1e-4 \
    --lr-decay-style linear --lr-warmup-fraction 0.01 \
    --log-interval 5 --eval-iters 0 "$@"
