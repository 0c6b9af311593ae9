#!/bin/bash
# Config-5 soak (BASELINE.json): heterogeneous fleet under sustained
# QPS with live failure injection — a real 8B engine plus a mock
# replica of the same model; mid-run the mock is killed (health eject)
# and later restarted (re-detect + resync recovery) while llmlb-bench
# drives traffic the whole time.
set -e
cd "$(dirname "$0")/.."
DUR=${DUR:-60} CONNS=${CONNS:-32} GEN=${GEN:-16} MODEL=${MODEL:-llama-3-8b}
GW=${GW_PORT:-19620} BE=${BE_PORT:-19621} MOCK=${MOCK_PORT:-19622}
python -m llmlb_amd.backend.launcher --model $MODEL --device ${DEVICE:-cuda:0} \
  --port $BE --kv-blocks 16384 --served-name soak-model \
  >/tmp/c5_be.log 2>&1 &
BE_PID=$!
gateway/bin/llmlb-mock --port $MOCK --workers 1 --models soak-model \
  >/tmp/c5_mock.log 2>&1 &
MOCK_PID=$!
LLMLB_PORT=$GW LLMLB_DATABASE_URL=/tmp/c5.sqlite3 LLMLB_ADMIN_TOKEN=t \
  LLMLB_HEALTH_CHECK_INTERVAL=2 gateway/bin/llmlb-gw --workers 8 \
  >/tmp/c5_gw.log 2>&1 &
GW_PID=$!
trap 'kill $BE_PID $MOCK_PID $GW_PID 2>/dev/null || true' EXIT
for i in $(seq 180); do
  curl -sf 127.0.0.1:$BE/api/health >/dev/null 2>&1 && break; sleep 1
done
curl -sm10 -X POST 127.0.0.1:$GW/api/endpoints -H "Authorization: Bearer t" \
  -d "{\"name\":\"engine\",\"base_url\":\"http://127.0.0.1:$BE\",\"device_index\":0}" >/dev/null
curl -sm10 -X POST 127.0.0.1:$GW/api/endpoints -H "Authorization: Bearer t" \
  -d "{\"name\":\"flaky\",\"base_url\":\"http://127.0.0.1:$MOCK\"}" >/dev/null
sleep 1
( sleep $((DUR/3)); kill $MOCK_PID 2>/dev/null; echo "[inject] mock killed at t=$((DUR/3))" >&2
  sleep $((DUR/3));
  gateway/bin/llmlb-mock --port $MOCK --workers 1 --models soak-model \
    >/tmp/c5_mock2.log 2>&1 &
  echo $! > /tmp/c5_mock2.pid
  echo "[inject] mock restarted at t=$((2*DUR/3))" >&2 ) &
INJ_PID=$!
BODY="{\"model\":\"soak-model\",\"max_tokens\":$GEN,\"stream\":true,\"messages\":[{\"role\":\"user\",\"content\":\"soak prompt for config five\"}]}"
gateway/bin/llmlb-bench --host 127.0.0.1 --port $GW \
  --path /v1/chat/completions --method POST --body "$BODY" \
  --header "Content-Type: application/json" \
  --threads 4 --connections $CONNS --duration $DUR --label config5
wait $INJ_PID 2>/dev/null || true
kill $(cat /tmp/c5_mock2.pid 2>/dev/null) 2>/dev/null || true
echo "--- endpoints at end:"
curl -sm10 127.0.0.1:$GW/api/endpoints -H "Authorization: Bearer t" \
  | python3 -c "import json,sys; [print(e['name'], e['status'], 'req', e['total_requests'], 'fail', e['total_failures']) for e in json.load(sys.stdin)]"
echo "--- eject/recover events:"
curl -sm10 "127.0.0.1:$GW/api/events" -H "Authorization: Bearer t" \
  | python3 -c "import json,sys; [print(e['type'], e.get('detail')) for e in json.load(sys.stdin) if 'Status' in e['type']]" | tail -8
