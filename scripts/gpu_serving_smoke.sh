#!/usr/bin/env bash
# Single-shard GPU serving smoke: load qwen-32b int8 synthetic via the API,
# run one chat with profile metrics. Run from repo root on a GPU box.
set -uo pipefail
mkdir -p gpurun_out
cat > gpurun_out/hosts1 <<HOSTS
shard0 127.0.0.1 18081 15052 0
HOSTS
python -m dnet_amd.cli.shard --name shard0 --host 127.0.0.1 --http-port 18081 --wire-port 15052 > gpurun_out/smoke_shard.log 2>&1 &
SHARD_PID=$!
python -m dnet_amd.cli.api --hostfile gpurun_out/hosts1 --host 127.0.0.1 --port 18080 --wire-port 15051 --callback-addr 127.0.0.1:15051 > gpurun_out/smoke_api.log 2>&1 &
API_PID=$!
trap 'kill -9 $SHARD_PID $API_PID 2>/dev/null' EXIT
for i in $(seq 1 60); do
  curl -s -m 2 http://127.0.0.1:18080/health > /dev/null && break
  sleep 1
done
curl -s -m 30 -X POST http://127.0.0.1:18080/v1/prepare_topology \
  -H 'content-type: application/json' \
  -d '{"model":"qwen-2.5-32b-int8-synthetic","quant":"int8-g128"}' | head -c 400; echo
curl -s -m 300 -X POST http://127.0.0.1:18080/v1/load_model \
  -H 'content-type: application/json' \
  -d '{"model":"qwen-2.5-32b-int8-synthetic","quant":"int8-g128","max_seq":1024}' | head -c 200; echo
time curl -s -m 120 -X POST http://127.0.0.1:18080/v1/chat/completions \
  -H 'content-type: application/json' \
  -d '{"model":"qwen-2.5-32b-int8-synthetic","max_tokens":32,"profile":true,"messages":[{"role":"user","content":"benchmark me"}]}'
echo
tail -n 3 gpurun_out/smoke_shard.log
