#!/bin/bash
# Terminal demo (stands in for the reference's VHS .tape assets): brings
# up a full node — C++ dispatcher + engine worker — on CPU with the tiny
# preset, then walks the wire surface an Ollama/OpenAI client would use.
# Runs anywhere (no GPU needed): bash packaging/demo.sh
set -e
cd "$(dirname "$0")/.."
PORT=${PORT:-18333}

echo "== building (first run only) =="
python -m ollamamq_amd.build >/dev/null

echo "== starting node (CPU, tiny preset) =="
python -m ollamamq_amd.launch --gpus 1 --model tiny --max-ctx 256 \
    --max-batch 4 --no-tui --port $PORT --sock-dir /tmp \
    -c /tmp/absent.yaml &
NODE=$!
trap "kill $NODE 2>/dev/null" EXIT
for i in $(seq 1 60); do
  curl -sf 127.0.0.1:$PORT/health >/dev/null 2>&1 && break; sleep 0.5
done

echo; echo "== /api/chat (Ollama JSON-lines streaming) =="
curl -s 127.0.0.1:$PORT/api/chat -H 'X-User-ID: alice' \
     -d '{"model":"tiny","messages":[{"role":"user","content":"hi"}],
          "options":{"num_predict":6}}' | head -4

echo; echo "== /v1/chat/completions (OpenAI SSE) =="
curl -s 127.0.0.1:$PORT/v1/chat/completions -H 'X-User-ID: bob' \
     -d '{"model":"tiny","stream":true,"max_tokens":4,
          "messages":[{"role":"user","content":"hello"}]}' | head -4

echo; echo "== /api/embed =="
curl -s 127.0.0.1:$PORT/api/embed -H 'X-User-ID: carol' \
     -d '{"model":"tiny","input":"vector me"}' | head -c 120; echo " ..."

echo; echo "== admin inventory + stats + metrics =="
curl -s 127.0.0.1:$PORT/admin/models | head -c 300; echo " ..."
curl -s 127.0.0.1:$PORT/admin/stats | head -c 200; echo " ..."
curl -s 127.0.0.1:$PORT/metrics | head -6

echo; echo "demo OK"
