#!/bin/bash
# Config-4 demo: a TP replica group under torchrun, registered with the
# gateway, serving /v1/messages (Anthropic ingress) + chat.
#   MODEL=llama-small TP=2 ./benchmarks/tp_demo.sh
set -e
cd "$(dirname "$0")/.."
MODEL=${MODEL:-llama-small}
TP=${TP:-2}
PORT=${PORT:-9400}
GWPORT=${GWPORT:-18088}
DB=$(mktemp /tmp/llmlb-tp-XXXX.sqlite3)
cleanup() { kill $TPPID $GW 2>/dev/null || true; rm -f "$DB"*; }
trap cleanup EXIT
python -m torch.distributed.run --standalone --nnodes=1 \
  --nproc-per-node $TP --local-addr 127.0.0.1 \
  -m llmlb_amd.backend.tp_runner --model $MODEL --port $PORT \
  --kv-blocks ${KVBLOCKS:-2048} &
TPPID=$!
LLMLB_PORT=$GWPORT LLMLB_DATABASE_URL=$DB LLMLB_ADMIN_TOKEN=t \
  LLMLB_HEALTH_CHECK_INTERVAL=5 gateway/bin/llmlb-gw --workers 4 \
  >/dev/null 2>&1 &
GW=$!
# wait for the TP backend then register
for i in $(seq 1 120); do
  curl -sm2 localhost:$PORT/v1/models >/dev/null 2>&1 && break
  sleep 1
done
curl -sm10 -X POST localhost:$GWPORT/api/endpoints \
  -H "Authorization: Bearer t" \
  -d "{\"name\":\"tp\",\"base_url\":\"http://127.0.0.1:$PORT\"}"
echo
echo "== chat through gateway -> TP=$TP group:"
time curl -sm60 -X POST localhost:$GWPORT/v1/chat/completions \
  -d "{\"model\":\"$MODEL\",\"max_tokens\":16,\"messages\":[{\"role\":\"user\",\"content\":\"hello tensor parallel world\"}]}"
echo
echo "== anthropic /v1/messages through gateway -> TP group:"
curl -sm60 -X POST localhost:$GWPORT/v1/messages \
  -H "anthropic-version: 2023-06-01" \
  -d "{\"model\":\"$MODEL\",\"max_tokens\":16,\"messages\":[{\"role\":\"user\",\"content\":\"hi\"}]}"
echo
