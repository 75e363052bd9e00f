#!/bin/bash
# MegaDPP smoke: TP1 PP2 VPP2 (reference test_train_gpt_single_dpp.sh shape)
torchrun --nproc-per-node 2 --master-addr 127.0.0.1 pretrain_gpt.py \
    --num-layers 8 --hidden-size 512 --num-attention-heads 8 \
    --seq-length 1024 --max-position-embeddings 1024 \
    --micro-batch-size 2 --global-batch-size 16 \
    --pipeline-model-parallel-size 2 --num-layers-per-virtual-pipeline-stage 2 \
    --bf16 --mock-data --train-iters 10 --lr 1e-4 --log-interval 1 \
    --eval-iters 0 --use-dpp --dpp-policy depth_first "$@"
