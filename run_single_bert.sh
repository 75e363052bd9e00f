#!/bin/bash
# Single-node BERT launch (reference run_single_bert.sh shape).
GPUS=${GPUS:-$(python -c "import torch;print(torch.cuda.device_count() or 1)")}
torchrun --nproc-per-node "$GPUS" --master-addr 127.0.0.1 pretrain_bert.py \
    --num-layers 24 --hidden-size 1024 --num-attention-heads 16 \
    --seq-length 512 --max-position-embeddings 512 \
    --micro-batch-size 4 --global-batch-size $((8 * GPUS)) \
    --bf16 --mock-data --train-iters 100 --lr 1e-4 \
    --log-interval 10 --eval-iters 0 "$@"
