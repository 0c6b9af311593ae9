#!/bin/bash
# Config-1 benchmark: gateway + mock backend + loadgen on loopback.
# Reference headline to beat: 170,600 req/s, p50 0.249 ms
# (wrk -t10 -c50 -d30s against /v1/chat/completions).
#
# Env knobs: DUR CONNS THREADS GWWORKERS MOCKWORKERS PORT MOCKPORT
set -e
cd "$(dirname "$0")/.."
BIN=gateway/bin
DUR=${DUR:-10}
CONNS=${CONNS:-50}
THREADS=${THREADS:-10}
NPROC=$(nproc)
# on small shared boxes split cores between the three processes
if [ -z "$GWWORKERS" ]; then
  if [ "$NPROC" -le 8 ]; then GWWORKERS=3; else GWWORKERS=$((NPROC / 2)); fi
fi
if [ -z "$MOCKWORKERS" ]; then
  if [ "$NPROC" -le 8 ]; then MOCKWORKERS=2; else MOCKWORKERS=$((NPROC / 4)); fi
fi
if [ "$NPROC" -le 8 ]; then THREADS=${LGTHREADS:-2}; fi
PORT=${PORT:-18080}
MOCKPORT=${MOCKPORT:-19100}
DB=$(mktemp /tmp/llmlb-bench-XXXX.sqlite3)
cleanup() { kill $GW $MOCK 2>/dev/null || true; rm -f "$DB"*; }
trap cleanup EXIT
$BIN/llmlb-mock --port $MOCKPORT --workers $MOCKWORKERS >/dev/null 2>&1 &
MOCK=$!
LLMLB_PORT=$PORT LLMLB_DATABASE_URL=$DB LLMLB_ADMIN_TOKEN=bench \
  LLMLB_HEALTH_CHECK_INTERVAL=60 $BIN/llmlb-gw --workers $GWWORKERS \
  >/dev/null 2>&1 &
GW=$!
sleep 0.5
# register the mock endpoint and verify the model is routable
for i in $(seq 1 20); do
  curl -sm10 -X POST localhost:$PORT/api/endpoints \
    -H "Authorization: Bearer bench" \
    -d "{\"name\":\"mock\",\"base_url\":\"http://127.0.0.1:$MOCKPORT\"}" \
    >/dev/null 2>&1 || true
  if curl -sm3 localhost:$PORT/v1/models | grep -q mock-model; then
    break
  fi
  sleep 0.5
done
if ! curl -sm3 localhost:$PORT/v1/models | grep -q mock-model; then
  echo "ERROR: mock endpoint failed to register" >&2
  exit 1
fi
$BIN/llmlb-bench --host 127.0.0.1 --port $PORT --path /v1/chat/completions \
  --threads $THREADS --connections $CONNS --duration $DUR --label local "$@"
