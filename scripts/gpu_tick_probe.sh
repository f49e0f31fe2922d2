#!/usr/bin/env bash
# Focused probe: single-model serving with TICK gpu_ms instrumentation.
set -uo pipefail
mkdir -p gpurun_out
cat > gpurun_out/hosts1 <<HOSTS
shard0 127.0.0.1 18081 15052 0
HOSTS
DNET_OBS_PROFILE=true python -m dnet_amd.cli.shard --name shard0 --host 127.0.0.1 --http-port 18081 --wire-port 15052 > gpurun_out/tick_shard.log 2>&1 &
SHARD_PID=$!
python -m dnet_amd.cli.api --hostfile gpurun_out/hosts1 --host 127.0.0.1 --port 18080 --wire-port 15051 --callback-addr 127.0.0.1:15051 > gpurun_out/tick_api.log 2>&1 &
API_PID=$!
trap 'kill -9 $SHARD_PID $API_PID 2>/dev/null' EXIT
for i in $(seq 1 60); do
  curl -s -m 2 http://127.0.0.1:18080/health > /dev/null && break
  sleep 1
done
curl -s -m 600 -X POST http://127.0.0.1:18080/v1/load_model -H 'content-type: application/json' \
  -d '{"model":"qwen-2.5-32b-int8-synthetic","max_batch":8,"quant":"int8-g128"}' | head -c 80; echo
echo "--- 1 stream x 300 tokens ---"
python scripts/serving_concurrent_bench.py --streams 1 --max-tokens 300
echo "--- 8 streams x 64 ---"
python scripts/serving_concurrent_bench.py --streams 8 --max-tokens 64
echo "--- 1 stream again (warm) ---"
python scripts/serving_concurrent_bench.py --streams 1 --max-tokens 300
curl -s -m 120 -X POST http://127.0.0.1:18080/v1/unload_model | head -c 40; echo
grep -E "PROFILE..TICK" gpurun_out/tick_shard.log | tail -20
