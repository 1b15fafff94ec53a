#!/usr/bin/env bash
# Launch a full single-node mesh: DHT bootstrap + one DP worker per GPU +
# consumer gateway (BASELINE config 3: 8 replica workers DHT-advertised on
# 8x MI355X, gateway load-balancing).
#
# Usage: scripts/launch_node.sh [MODEL] [NGPUS] [SCHEME]
set -euo pipefail
MODEL="${1:-llama3-8b}"
NGPUS="${2:-$(python -c 'from crowdllama_amd.ops import get_core; print(get_core().device_count())')}"
SCHEME="${3:-q4_k_m}"
cd "$(dirname "$0")/.."

PIDS=()
cleanup() { kill "${PIDS[@]}" 2>/dev/null || true; }
trap cleanup EXIT INT TERM

echo "[node] starting DHT bootstrap on :9000"
python -m crowdllama_amd.cli dht --port 9000 --key /tmp/cla-dht.key &
PIDS+=($!)
sleep 1

for ((i = 0; i < NGPUS; i++)); do
    echo "[node] starting worker $i on GPU $i"
    python -m crowdllama_amd.cli start --worker-mode \
        --models "$MODEL" --scheme "$SCHEME" --device "$i" \
        --bootstrap 127.0.0.1:9000 --key "/tmp/cla-worker$i.key" &
    PIDS+=($!)
done

echo "[node] starting gateway on :9001"
python -m crowdllama_amd.cli start --bootstrap 127.0.0.1:9000 \
    --port 9001 --key /tmp/cla-consumer.key &
PIDS+=($!)

echo "[node] mesh up: gateway http://localhost:9001 (POST /api/chat)"
wait
