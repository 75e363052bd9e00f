#!/bin/bash
# GPT 345M-class with the MegaScan tracing demo flags (mirrors the
# reference examples/gpt3/train_gpt3_345m_distributed.sh usage).
GPUS_PER_NODE=${GPUS_PER_NODE:-1}
torchrun --nproc-per-node $GPUS_PER_NODE --master-addr 127.0.0.1 \
    pretrain_gpt.py \
    --num-layers 24 --hidden-size 1024 --num-attention-heads 16 \
    --seq-length 2048 --max-position-embeddings 2048 \
    --micro-batch-size 2 --global-batch-size 16 \
    --bf16 --mock-data --train-iters 50 --lr 1.5e-4 --min-lr 1e-5 \
    --lr-decay-style cosine --lr-warmup-iters 10 \
    --log-interval 5 --eval-iters 0 \
    --trace --trace-dir trace_output --trace-interval 5 \
    --continuous-trace-iterations 2 --trace-granularity full "$@"
echo "aggregate with: python scripts/aggregate.py --trace-dir trace_output --detect"
