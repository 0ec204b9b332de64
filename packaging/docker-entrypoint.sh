#!/bin/sh
# Env-var surface mirrors the reference's entrypoint contract
# (BACKEND_URLS / OLLAMA_URLS / PORT / HOST / TIMEOUT — those names work
# unchanged) plus OMQ_* GPU-worker knobs which take precedence.
set -e
ARGS="--no-tui --gpus ${OMQ_GPUS:-1} --model ${OMQ_MODEL:-llama3-8b} \
      --port ${OMQ_PORT:-${PORT:-11435}} --host ${OMQ_HOST:-${HOST:-0.0.0.0}}"
[ -n "$OMQ_TP" ] && ARGS="$ARGS --tp $OMQ_TP"
[ -n "$TIMEOUT" ] && ARGS="$ARGS --timeout $TIMEOUT"
[ -n "$BACKEND_URLS" ] && ARGS="$ARGS --extra-backends $BACKEND_URLS"
[ -n "$OLLAMA_URLS" ] && ARGS="$ARGS --extra-backends $OLLAMA_URLS"
exec python -m ollamamq_amd.launch $ARGS
