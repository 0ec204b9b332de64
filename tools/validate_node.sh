#!/bin/bash
# Full-stack validation on a GPU box: node launcher -> C++ dispatcher ->
# UDS -> GPU engine worker serving Llama-3-8B over the Ollama/OpenAI wire,
# plus multi-model routing (8B + tiny) and a small stress run.
set -x
cd /root/repo
PORT=18111
python -m ollamamq_amd.launch --gpus 1 --model llama3-8b --max-ctx 2048 \
    --max-batch 16 --no-tui --port $PORT --sock-dir /tmp \
    -c /tmp/absent.yaml &
LAUNCH=$!
trap "kill $LAUNCH 2>/dev/null" EXIT

for i in $(seq 1 120); do
  curl -sf http://127.0.0.1:$PORT/health >/dev/null && break
  sleep 1
done
curl -s http://127.0.0.1:$PORT/health; echo

# wait until the worker probe shows 8B loaded
for i in $(seq 1 120); do
  curl -s http://127.0.0.1:$PORT/admin/models | grep -q '"llama3-8b"' && break
  sleep 1
done
curl -s http://127.0.0.1:$PORT/admin/models | head -c 400; echo

# single chat request through the whole stack
time curl -s -X POST http://127.0.0.1:$PORT/api/chat \
     -H 'X-User-ID: valid1' \
     -d '{"model":"llama3-8b","messages":[{"role":"user","content":"hello world"}],"options":{"num_predict":16}}' | tail -c 300; echo

# OpenAI surface
curl -s -X POST http://127.0.0.1:$PORT/v1/chat/completions \
     -H 'X-User-ID: valid2' \
     -d '{"model":"llama3-8b","max_tokens":8,"messages":[{"role":"user","content":"hi"}]}' | head -c 300; echo

# load a second model (mixed fleet, config 5 style) and route to it
curl -s -X POST http://127.0.0.1:$PORT/admin/models/load \
     -d '{"model":"tiny","backend":0,"num_ctx":512}'; echo
for i in $(seq 1 60); do
  curl -s http://127.0.0.1:$PORT/admin/models | grep -q '"tiny"' && break
  sleep 1
done
curl -s -X POST http://127.0.0.1:$PORT/api/generate \
     -H 'X-User-ID: valid3' \
     -d '{"model":"tiny","prompt":"abc","stream":false,"options":{"num_predict":4}}' | head -c 200; echo

# embeddings through the stack
curl -s -X POST http://127.0.0.1:$PORT/api/embed \
     -H 'X-User-ID: valid4' \
     -d '{"model":"llama3-8b","input":"embed me"}' | head -c 120; echo

# stress: 12 users, both models, cancels included
timeout 240 python tools/stress.py --base http://127.0.0.1:$PORT \
    --users 12 --models llama3-8b,tiny --max-tokens 12 --par 8

# metrics surface
curl -s http://127.0.0.1:$PORT/admin/stats | head -c 500; echo
