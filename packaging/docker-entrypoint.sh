#!/bin/sh
# Env-var surface mirrors the reference's entrypoint contract
# (BACKEND_URLS / PORT / HOST / TIMEOUT) plus the GPU-worker knobs.
set -e
ARGS="--no-tui --gpus ${OMQ_GPUS:-1} --model ${OMQ_MODEL:-llama3-8b} \
      --port ${OMQ_PORT:-11435} --host ${OMQ_HOST:-0.0.0.0}"
[ -n "$OMQ_TP" ] && ARGS="$ARGS --tp $OMQ_TP"
[ -n "$BACKEND_URLS" ] && ARGS="$ARGS --extra-backends $BACKEND_URLS"
[ -n "$OLLAMA_URLS" ] && ARGS="$ARGS --extra-backends $OLLAMA_URLS"
exec python -m ollamamq_amd.launch $ARGS
