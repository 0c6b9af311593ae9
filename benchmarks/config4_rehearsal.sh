#!/bin/bash
# Config-4 rehearsal (BASELINE.json: 2x TP replicas behind the gateway,
# Anthropic /v1/messages ingress with streaming) in ONE command.
#
# On an 8-GPU node:  MODEL=llama-3-70b TP=4 ./benchmarks/config4_rehearsal.sh
#   -> two TP=4 quads pinned to xGMI-adjacent GPUs (topology.py picks
#      the quads; tp_runner pins HIP_VISIBLE_DEVICES per group)
# On 1 GPU / CPU CI:  MODEL=llama-small TP=2 GPUS=shared ./benchmarks/config4_rehearsal.sh
#   -> both TP groups share the device (gloo/RCCL rehearsal of the
#      exact orchestration the quad deployment uses)
set -e
cd "$(dirname "$0")/.."
MODEL=${MODEL:-llama-small}
TP=${TP:-2}
REPLICAS=${REPLICAS:-2}
PORT0=${PORT0:-9400}
GWPORT=${GWPORT:-18090}
DB=$(mktemp /tmp/llmlb-c4-XXXX.sqlite3)
PIDS=()
cleanup() { kill "${PIDS[@]}" 2>/dev/null || true; rm -f "$DB"*; }
trap cleanup EXIT

for r in $(seq 0 $((REPLICAS-1))); do
  PORT=$((PORT0 + r))
  if [ "${GPUS:-auto}" = "shared" ]; then DEVSEL=""; else
    # quad selection: topology.py prints the xGMI-adjacent group for
    # this replica index (falls back to consecutive ids off-node)
    DEVSEL=$(python -c "
import torch
from llmlb_amd.parallel.topology import place_tp_groups
n = torch.cuda.device_count()
groups = place_tp_groups(n, $TP, n_groups=$REPLICAS)
print(','.join(str(d) for d in groups[$r]))" 2>/dev/null || true)
  fi
  HIP_VISIBLE_DEVICES=${DEVSEL:-} python -m torch.distributed.run \
    --standalone --nnodes=1 --nproc-per-node $TP --local-addr 127.0.0.1 \
    -m llmlb_amd.backend.tp_runner --model $MODEL --port $PORT \
    --kv-blocks ${KVBLOCKS:-1024} >/tmp/llmlb_c4_tp$r.log 2>&1 &
  PIDS+=($!)
done

LLMLB_PORT=$GWPORT LLMLB_DATABASE_URL=$DB LLMLB_ADMIN_TOKEN=t \
  LLMLB_HEALTH_CHECK_INTERVAL=5 gateway/bin/llmlb-gw --workers 4 \
  >/tmp/llmlb_c4_gw.log 2>&1 &
PIDS+=($!)

for r in $(seq 0 $((REPLICAS-1))); do
  PORT=$((PORT0 + r))
  for i in $(seq 1 180); do
    curl -sm2 localhost:$PORT/v1/models >/dev/null 2>&1 && break
    sleep 1
  done
  curl -sm10 -X POST localhost:$GWPORT/api/endpoints \
    -H "Authorization: Bearer t" \
    -d "{\"name\":\"tp-$r\",\"base_url\":\"http://127.0.0.1:$PORT\"}" \
    >/dev/null
done

echo "== Anthropic /v1/messages through the gateway (streaming) =="
curl -sm60 -N -X POST localhost:$GWPORT/v1/messages \
  -H "anthropic-version: 2023-06-01" -H "content-type: application/json" \
  -d "{\"model\":\"$MODEL\",\"max_tokens\":16,\"stream\":true,
       \"messages\":[{\"role\":\"user\",\"content\":\"hello\"}]}" \
  | head -12
echo "== chat (both replicas take traffic) =="
for i in 1 2 3 4; do
  curl -sm60 -X POST localhost:$GWPORT/v1/chat/completions \
    -H "content-type: application/json" \
    -d "{\"model\":\"$MODEL\",\"max_tokens\":8,
         \"messages\":[{\"role\":\"user\",\"content\":\"hi $i\"}]}" \
    | python3 -c "import json,sys; d=json.load(sys.stdin); print(d['id'], d['choices'][0]['finish_reason'])"
done
curl -sm10 localhost:$GWPORT/api/endpoints -H "Authorization: Bearer t" \
  | python3 -c "import json,sys; [print(e['name'], e['status'], e['total_requests']) for e in json.load(sys.stdin)]"
echo "config-4 rehearsal OK"
