#!/bin/bash
# Serve + query with the REST CLI client (reference
# tools/text_generation_cli.py is a client for the server's /api).
PORT=${PORT:-5001}
python tools/run_text_generation_server.py \
    --num-layers 12 --hidden-size 768 --num-attention-heads 12 \
    --seq-length 512 --max-position-embeddings 1024 \
    --micro-batch-size 1 --bf16 --vocab-size 51200 \
    ${LOAD:+--load $LOAD} --rest --port "$PORT" &
SERVER=$!
sleep 30
python tools/text_generation_cli.py "127.0.0.1:$PORT" "the quick brown fox"
kill $SERVER
