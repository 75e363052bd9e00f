#!/bin/bash
# Llama-3-8B-class architecture (GQA 8 groups, SwiGLU, RMSNorm, rope,
# untied embeddings) with the distributed optimizer.
GPUS_PER_NODE=${GPUS_PER_NODE:-8}
torchrun --nproc-per-node $GPUS_PER_NODE --master-addr 127.0.0.1 \
    pretrain_gpt.py \
    --num-layers 32 --hidden-size 4096 --num-attention-heads 32 \
    --group-query-attention --num-query-groups 8 \
    --ffn-hidden-size 14336 --swiglu --normalization RMSNorm \
    --position-embedding-type rope --disable-bias-linear \
    --untie-embeddings-and-output-weights \
    --tensor-model-parallel-size ${TP:-1} \
    --pipeline-model-parallel-size ${PP:-1} \
    --use-distributed-optimizer --overlap-grad-reduce \
    --seq-length 2048 --max-position-embeddings 8192 \
    --micro-batch-size 1 --global-batch-size 32 \
    --bf16 --mock-data --train-iters 50 --lr 3e-4 \
    --log-interval 5 --eval-iters 0 "$@"
