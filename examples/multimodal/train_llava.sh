#!/bin/bash
# LLaVA multimodal pretraining (reference examples/multimodal): CLIP ViT
# tower + projector + GPT decoder; <image> tokens expand into patch
# embeddings.  Synthetic image/text data — point MockVLMDataset at a
# real pipeline for corpus training.
GPUS_PER_NODE=${GPUS_PER_NODE:-1}
torchrun --nproc-per-node $GPUS_PER_NODE --master-addr 127.0.0.1 \
    pretrain_vlm.py \
    --num-layers 12 --hidden-size 768 --num-attention-heads 12 \
    --seq-length 512 --max-position-embeddings 1024 \
    --img-h 336 --img-w 336 --patch-dim 14 \
    --vision-num-layers 6 --vision-hidden-size 512 \
    --vision-num-attention-heads 8 \
    --micro-batch-size 2 --global-batch-size 8 \
    --bf16 --train-iters 50 --lr 1e-4 \
    --log-interval 5 --eval-iters 0 "$@"
