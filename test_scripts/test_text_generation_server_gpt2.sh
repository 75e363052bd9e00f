#!/bin/bash
# MegaScope inference server smoke (reference test_text_generation_server_gpt2.sh)
python tools/run_text_generation_server.py \
    --num-layers 4 --hidden-size 256 --num-attention-heads 4 \
    --seq-length 512 --max-position-embeddings 1024 \
    --micro-batch-size 1 --vocab-size 4096 \
    --inference-ws-port 5000 "$@"
# then open transformer-visualize/index.html and connect to ws://host:5000
