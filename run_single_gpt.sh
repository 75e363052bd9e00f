#!/bin/bash
# Single-node GPT launch (reference run_single_gpt.sh shape): all GPUs on
# this node, BASELINE-style GPT-3 1.3B PP=4 when 4+ GPUs are present.
GPUS=${GPUS:-$(python -c "import torch;print(torch.cuda.device_count() or 1)")}
PP=$(( GPUS >= 4 ? 4 : 1 ))
torchrun --nproc-per-node "$GPUS" --master-addr 127.0.0.1 pretrain_gpt.py \
    --num-layers 24 --hidden-size 2048 --num-attention-heads 16 \
    --seq-length 2048 --max-position-embeddings 2048 \
    --micro-batch-size 2 --global-batch-size $((4 * GPUS)) \
    --pipeline-model-parallel-size "$PP" \
    --bf16 --mock-data --train-iters 100 --lr 1e-4 \
    --log-interval 10 --eval-iters 0 "$@"
