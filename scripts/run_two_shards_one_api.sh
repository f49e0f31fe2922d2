#!/usr/bin/env bash
# Localhost 2-shard + API dev cluster (reference: scripts/run_two_shards_one_api.sh).
set -euo pipefail
cd "$(dirname "$0")/.."
HOSTFILE=$(mktemp)
cat > "$HOSTFILE" <<HOSTS
shard0 127.0.0.1 8081 50052 0
shard1 127.0.0.1 8181 50152 1
HOSTS
trap 'kill 0' EXIT
python -m dnet_amd.cli.shard --name shard0 --http-port 8081 --wire-port 50052 &
python -m dnet_amd.cli.shard --name shard1 --http-port 8181 --wire-port 50152 &
sleep 5
python -m dnet_amd.cli.api --hostfile "$HOSTFILE" --port 8080 --wire-port 50051 \
  --callback-addr 127.0.0.1:50051 &
echo "API on http://127.0.0.1:8080 — e.g.:"
echo '  curl -X POST localhost:8080/v1/prepare_topology -H "content-type: application/json" -d "{\"model\":\"tiny-random\"}"'
echo '  curl -X POST localhost:8080/v1/load_model -H "content-type: application/json" -d "{\"model\":\"tiny-random\"}"'
echo '  curl -X POST localhost:8080/v1/chat/completions -H "content-type: application/json" -d "{\"model\":\"tiny-random\",\"messages\":[{\"role\":\"user\",\"content\":\"hi\"}]}"'
wait
