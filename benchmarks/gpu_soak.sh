#!/bin/bash
# Sustained serving soak on a GPU box: real llama-3-8b engine behind
# the gateway, llmlb-bench driving /v1/chat/completions at high
# concurrency.  Emits the loadgen CSV + gateway and backend health.
set -e
cd "$(dirname "$0")/.."
DUR=${DUR:-45} CONNS=${CONNS:-192} GEN=${GEN:-16} PROMPT_WORDS=${PROMPT_WORDS:-8}
GW=${GW_PORT:-19610} BE=${BE_PORT:-19611}
export LLMLB_NO_AUDIT=   # audit ON (realistic)
python -m llmlb_amd.backend.launcher --model llama-3-8b --device cuda:0 \
  --port $BE --kv-blocks 16384 >/tmp/soak_be.log 2>&1 &
BE_PID=$!
LLMLB_PORT=$GW LLMLB_DATABASE_URL=/tmp/soak.sqlite3 \
  LLMLB_ADMIN_TOKEN=admintok LLMLB_HEALTH_CHECK_INTERVAL=5 \
  gateway/bin/llmlb-gw --workers 8 >/tmp/soak_gw.log 2>&1 &
GW_PID=$!
trap 'kill $BE_PID $GW_PID 2>/dev/null' EXIT
for i in $(seq 120); do
  curl -sf 127.0.0.1:$BE/api/health >/dev/null 2>&1 && break; sleep 1
done
sleep 0.5
curl -sf -X POST 127.0.0.1:$GW/api/endpoints \
  -H 'Authorization: Bearer admintok' \
  -d "{\"name\":\"soak\",\"base_url\":\"http://127.0.0.1:$BE\",\"device_index\":0}" >/dev/null
sleep 3   # let sync discover the model
PROMPT=$(python3 -c "import sys; print(' '.join(f'w{i}' for i in range(int(sys.argv[1]))))" $PROMPT_WORDS)
BODY="{\"model\":\"llama-3-8b\",\"max_tokens\":$GEN,\"stream\":true,\"messages\":[{\"role\":\"user\",\"content\":\"$PROMPT\"}]}"
gateway/bin/llmlb-bench --host 127.0.0.1 --port $GW \
  --path /v1/chat/completions --method POST --body "$BODY" \
  --threads 8 --connections $CONNS --duration $DUR --label gpu_soak
echo "--- backend health:"
curl -sf 127.0.0.1:$BE/api/health
echo
echo "--- gateway summary:"
curl -sf 127.0.0.1:$GW/api/dashboard/summary -H 'Authorization: Bearer admintok' | head -c 400
echo
