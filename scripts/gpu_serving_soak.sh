#!/usr/bin/env bash
# Serving soak on one GPU box: load qwen-32b int8 synthetic, run several
# chat requests (streaming + non-streaming), then unload and hot-swap to
# gpt-oss-20b (MoE + sinks + sliding window) through the same API process.
# Run from repo root on a GPU box; logs + metrics under gpurun_out/.
set -uo pipefail
mkdir -p gpurun_out
cat > gpurun_out/hosts1 <<HOSTS
shard0 127.0.0.1 18081 15052 0
HOSTS
DNET_OBS_PROFILE=true python -m dnet_amd.cli.shard --name shard0 --host 127.0.0.1 --http-port 18081 --wire-port 15052 > gpurun_out/soak_shard.log 2>&1 &
SHARD_PID=$!
python -m dnet_amd.cli.api --hostfile gpurun_out/hosts1 --host 127.0.0.1 --port 18080 --wire-port 15051 --callback-addr 127.0.0.1:15051 > gpurun_out/soak_api.log 2>&1 &
API_PID=$!
trap 'kill -9 $SHARD_PID $API_PID 2>/dev/null' EXIT
for i in $(seq 1 60); do
  curl -s -m 2 http://127.0.0.1:18080/health > /dev/null && break
  sleep 1
done

chat() {  # chat <model> <max_tokens> <prompt>
  curl -s -m 180 -X POST http://127.0.0.1:18080/v1/chat/completions \
    -H 'content-type: application/json' \
    -d "{\"model\":\"$1\",\"max_tokens\":$2,\"profile\":true,\"messages\":[{\"role\":\"user\",\"content\":\"$3\"}]}"
}

echo "=== load qwen-2.5-32b int8 ==="
curl -s -m 300 -X POST http://127.0.0.1:18080/v1/load_model \
  -H 'content-type: application/json' \
  -d '{"model":"qwen-2.5-32b-int8-synthetic","quant":"int8-g128","max_seq":1024}' | head -c 200; echo
for i in 1 2 3; do
  echo "--- request $i ---"
  chat qwen-2.5-32b-int8-synthetic 48 "soak request $i: tell me something" \
    | python3 -c 'import json,sys; d=json.load(sys.stdin); m=d.get("metrics",{}); print("tokens:", d["usage"]["completion_tokens"], "ttfb_ms:", round(m.get("ttfb_ms",0),1), "tps_decoding:", round(m.get("tps_decoding",0),1))'
done
echo "--- streaming request ---"
curl -s -N -m 180 -X POST http://127.0.0.1:18080/v1/chat/completions \
  -H 'content-type: application/json' \
  -d '{"model":"qwen-2.5-32b-int8-synthetic","stream":true,"max_tokens":24,"messages":[{"role":"user","content":"stream me"}]}' \
  | grep -c '^data: ' | xargs echo "sse chunks:"
echo "=== unload ==="
curl -s -m 120 -X POST http://127.0.0.1:18080/v1/unload_model | head -c 120; echo
echo "=== hot-swap to gpt-oss-20b (MoE/sinks/sliding) ==="
curl -s -m 300 -X POST http://127.0.0.1:18080/v1/load_model \
  -H 'content-type: application/json' \
  -d '{"model":"gpt-oss-20b-synthetic","max_seq":1024}' | head -c 200; echo
for i in 1 2; do
  echo "--- gpt-oss request $i ---"
  chat gpt-oss-20b-synthetic 32 "hello moe $i" \
    | python3 -c 'import json,sys; d=json.load(sys.stdin); m=d.get("metrics",{}); print("tokens:", d["usage"]["completion_tokens"], "ttfb_ms:", round(m.get("ttfb_ms",0),1), "tps_decoding:", round(m.get("tps_decoding",0),1))'
done
echo "=== soak done ==="
grep "\[PROFILE\]\[DECODE\]" gpurun_out/soak_shard.log | tail -5
