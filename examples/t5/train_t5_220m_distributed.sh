#!/bin/bash
# T5-base-class encoder-decoder pretraining (mirrors the reference
# examples/t5/train_t5_220m_distributed.sh usage).
GPUS_PER_NODE=${GPUS_PER_NODE:-1}
torchrun --nproc-per-node $GPUS_PER_NODE --master-addr 127.0.0.1 \
    pretrain_t5.py \
    --num-layers 12 --hidden-size 768 --num-attention-heads 12 \
    --kv-channels 64 --ffn-hidden-size 3072 \
    --seq-length 512 --decoder-seq-length 128 \
    --max-position-embeddings 512 \
    --micro-batch-size 4 --global-batch-size 32 \
    --bf16 --mock-data --train-iters 50 --lr 1e-4 \
    --lr-decay-style linear --lr-warmup-fraction 0.01 \
    --log-interval 5 --eval-iters 0 "$@"
