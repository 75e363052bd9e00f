#!/bin/bash
# Serve a (random-init or checkpointed) GPT over REST and query it
# (reference examples/inference usage).  Decode runs through the hipGraph
# capture path (2.3x eager decode on MI355X).
PORT=${PORT:-5000}
python tools/run_text_generation_server.py \
    --num-layers 24 --hidden-size 1024 --num-attention-heads 16 \
    --seq-length 1024 --max-position-embeddings 2048 \
    --micro-batch-size 1 --bf16 --vocab-size 51200 \
    ${LOAD:+--load $LOAD} --rest --port "$PORT" &
SERVER=$!
sleep 30
curl -s -X PUT -H "Content-Type: application/json" \
    -d '{"prompts": ["hello world"], "tokens_to_generate": 32, "top_k": 1}' \
    "http://127.0.0.1:$PORT/api"
kill $SERVER
