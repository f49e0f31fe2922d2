#!/usr/bin/env bash
# Serving marathon on one GPU box: cycle every model family through ONE
# api+shard pair (load -> chats incl. concurrent + long prompt -> unload),
# proving hot-swap across families and steady-state serving throughput.
set -uo pipefail
mkdir -p gpurun_out
cat > gpurun_out/hosts1 <<HOSTS
shard0 127.0.0.1 18081 15052 0
HOSTS
DNET_OBS_PROFILE=true python -m dnet_amd.cli.shard --name shard0 --host 127.0.0.1 --http-port 18081 --wire-port 15052 > gpurun_out/mar_shard.log 2>&1 &
SHARD_PID=$!
python -m dnet_amd.cli.api --hostfile gpurun_out/hosts1 --host 127.0.0.1 --port 18080 --wire-port 15051 --callback-addr 127.0.0.1:15051 > gpurun_out/mar_api.log 2>&1 &
API_PID=$!
trap 'kill -9 $SHARD_PID $API_PID 2>/dev/null' EXIT
for i in $(seq 1 60); do
  curl -s -m 2 http://127.0.0.1:18080/health > /dev/null && break
  sleep 1
done

chat() {  # chat <model> <max_tokens> <prompt>
  curl -s -m 240 -X POST http://127.0.0.1:18080/v1/chat/completions \
    -H 'content-type: application/json' \
    -d "{\"model\":\"$1\",\"max_tokens\":$2,\"profile\":true,\"messages\":[{\"role\":\"user\",\"content\":\"$3\"}]}" \
    | python3 -c 'import json,sys; d=json.load(sys.stdin); m=d.get("metrics",{}); print("  tokens:", d["usage"]["completion_tokens"], "ttfb_ms:", round(m.get("ttfb_ms",0),1), "tps:", round(m.get("tps_decoding",0),1))'
}

cycle() {  # cycle <model> [extra-load-json]
  local model=$1 extra=${2:-}
  echo "=== $model ==="
  local t0=$(date +%s)
  curl -s -m 400 -X POST http://127.0.0.1:18080/v1/load_model \
    -H 'content-type: application/json' \
    -d "{\"model\":\"$model\",\"max_seq\":1024$extra}" | head -c 80; echo
  echo "  load: $(( $(date +%s) - t0 ))s"
  chat "$model" 32 "warm request one"
  chat "$model" 32 "warm request two"
  # concurrent pair
  ( chat "$model" 24 "concurrent a" & chat "$model" 24 "concurrent b" & wait )
  # long prompt (600 chars)
  chat "$model" 16 "$(printf 'ctx%.0s' $(seq 1 200))"
  curl -s -m 120 -X POST http://127.0.0.1:18080/v1/unload_model | head -c 40; echo
}

cycle qwen-2.5-32b-int8-synthetic ',"quant":"int8-g128"'
cycle gpt-oss-20b-synthetic
cycle deepseek-v2-lite-synthetic
cycle mixtral-8x7b-synthetic ',"quant":"int8-g128"'
echo "=== concurrent streams (qwen-32b int8, slots) ==="
curl -s -m 600 -X POST http://127.0.0.1:18080/v1/load_model -H 'content-type: application/json' \
  -d '{"model":"qwen-2.5-32b-int8-synthetic","max_batch":8}' > /dev/null
python scripts/serving_concurrent_bench.py --streams 1 --max-tokens 48
python scripts/serving_concurrent_bench.py --streams 4 --max-tokens 48
python scripts/serving_concurrent_bench.py --streams 8 --max-tokens 48
echo "=== marathon done ==="
grep -c "PROFILE..DECODE" gpurun_out/mar_shard.log | xargs echo "decode runs:"
grep -ciE "traceback|error" gpurun_out/mar_shard.log | xargs echo "shard errors:"
