#!/bin/bash
# ViT classification (reference pretrain_vision_classify).  Also see
# pretrain_vision_inpaint.py and pretrain_vision_dino.py.
GPUS_PER_NODE=${GPUS_PER_NODE:-1}
torchrun --nproc-per-node $GPUS_PER_NODE --master-addr 127.0.0.1 \
    pretrain_vision_classify.py \
    --num-layers 12 --hidden-size 768 --num-attention-heads 12 \
    --seq-length 1 --max-position-embeddings 1024 \
    --img-h 224 --img-w 224 --patch-dim 16 --num-classes 1000 \
    --micro-batch-size 32 --global-batch-size 128 \
    --bf16 --train-iters 50 --lr 1e-3 \
    --log-interval 5 --eval-iters 0 "$@"
