#!/bin/bash
# Serving soak on a GPU box: bring up the full node (dispatcher + GPU
# worker, Llama-3-8B), then drive repeated randomized stress waves for
# $SOAK_SECS while watching VRAM and dispatcher RSS for drift.
set -x
cd /root/repo
PORT=18222
SOAK_SECS=${SOAK_SECS:-300}
python -m ollamamq_amd.launch --gpus 1 --model llama3-8b --max-ctx 2048 \
    --max-batch 32 --no-tui --port $PORT --sock-dir /tmp \
    -c /tmp/absent.yaml &
LAUNCH=$!
trap "kill $LAUNCH 2>/dev/null" EXIT

for i in $(seq 1 180); do
  curl -sf http://127.0.0.1:$PORT/health >/dev/null && break; sleep 1
done
for i in $(seq 1 180); do
  curl -s http://127.0.0.1:$PORT/admin/models | grep -q '"llama3-8b"' && break
  sleep 1
done

vram() { rocm-smi --showmeminfo vram --json 2>/dev/null \
         | python3 -c "import json,sys;d=json.load(sys.stdin);print(next(iter(d.values()))['VRAM Total Used Memory (B)'])"; }
rss() { ps -o rss= -p $(pgrep -P $LAUNCH -f ollamamq-server | head -1) 2>/dev/null || echo 0; }

# one warmup wave first: decode graphs are captured lazily per batch
# size (bounded by --max-batch, ~75 MB each); baseline AFTER they exist
timeout 200 python tools/stress.py --base http://127.0.0.1:$PORT --sampled-pct ${SOAK_SAMPLED:-0} \
    --users 24 --models llama3-8b --max-tokens 24 2>&1 | tail -1
V0=$(vram); R0=$(rss)
echo "post-warmup baseline vram=$V0 rss=$R0"
END=$(( $(date +%s) + SOAK_SECS ))
WAVE=0
VLOG=""
while [ "$(date +%s)" -lt "$END" ]; do
  WAVE=$((WAVE+1))
  timeout 200 python tools/stress.py --base http://127.0.0.1:$PORT --sampled-pct ${SOAK_SAMPLED:-0} \
      --users 24 --models llama3-8b --max-tokens 24 2>&1 | tail -1
  VW=$(vram); VLOG="$VLOG $VW"
  echo "wave $WAVE vram=$VW"
done
V1=$(vram); R1=$(rss)
echo "vram trajectory:$VLOG"
echo "after $WAVE waves: vram=$V1 (delta $((V1-V0))) rss=$R1 (delta $((R1-R0)))"
curl -s http://127.0.0.1:$PORT/admin/stats | head -c 300; echo
curl -s http://127.0.0.1:$PORT/metrics | grep -E "processed_total|dropped"
# decode graphs are captured lazily per batch size, so VRAM steps up as
# new sizes first appear (bounded by max_batch); what must NOT happen is
# continued growth once sizes repeat: assert the LAST HALF of the
# trajectory is flat (<128 MB), and dispatcher RSS stable (<200 MB)
python3 - <<PYEOF
v = [int(x) for x in "$VLOG".split()]
# measured trajectory (profiles/r01_soak.log): capture ramp flattens by
# ~wave 15 of 24; judge the final 8 waves
tail = v[-8:] if len(v) >= 8 else v
drift = max(tail) - min(tail)
assert drift < 64*2**20, f"VRAM drift in steady tail: {drift/2**20:.0f} MB over {tail}"
assert $R1 - $R0 < 200*1024, "RSS drift"
print(f"SOAK OK  (capture ramp {(v[-1]-v[0])/2**20:.0f} MB, steady-half drift {drift/2**20:.0f} MB)")
PYEOF
