#!/usr/bin/env bash
# Continuous-batching demo: 4 concurrent streams against one shard loaded
# with max_batch=4 (slot scheduler) vs max_batch=1 (serialized).
set -uo pipefail
mkdir -p gpurun_out
cat > gpurun_out/hosts1 <<HOSTS
shard0 127.0.0.1 18081 15052 0
HOSTS
DNET_OBS_PROFILE=true python -m dnet_amd.cli.shard --name shard0 --host 127.0.0.1 --http-port 18081 --wire-port 15052 > gpurun_out/slots_shard.log 2>&1 &
SHARD_PID=$!
python -m dnet_amd.cli.api --hostfile gpurun_out/hosts1 --host 127.0.0.1 --port 18080 --wire-port 15051 --callback-addr 127.0.0.1:15051 > gpurun_out/slots_api.log 2>&1 &
API_PID=$!
trap 'kill -9 $SHARD_PID $API_PID 2>/dev/null' EXIT
for i in $(seq 1 60); do
  curl -s -m 2 http://127.0.0.1:18080/health > /dev/null && break
  sleep 1
done

fire4() {
  local t0=$(date +%s.%N)
  # subshell so `wait` covers only the curls (a bare wait in the main
  # shell would also wait on the server daemons — forever)
  (
    for i in 1 2 3 4; do
      curl -s -m 90 -X POST http://127.0.0.1:18080/v1/chat/completions \
        -H 'content-type: application/json' \
        -d "{\"model\":\"gpt-oss-20b-synthetic\",\"max_tokens\":64,\"profile\":true,\"messages\":[{\"role\":\"user\",\"content\":\"stream $i\"}]}" \
        | python3 -c 'import json,sys; d=json.load(sys.stdin); m=d.get("metrics",{}); print("  tokens:", d["usage"]["completion_tokens"], "tps:", round(m.get("tps_decoding",0),1), "total_ms:", round(m.get("total_ms",0),1))' &
    done
    wait
  )
  python3 -c "import time; print('  wall: %.2fs' % ($(date +%s.%N) - $t0))"
}

echo "=== max_batch=4 (slot scheduler) ==="
curl -s -m 300 -X POST http://127.0.0.1:18080/v1/load_model \
  -H 'content-type: application/json' \
  -d '{"model":"gpt-oss-20b-synthetic","max_seq":1024,"max_batch":4}' | head -c 60; echo
fire4   # warm (includes graph capture)
fire4
curl -s -m 120 -X POST http://127.0.0.1:18080/v1/unload_model > /dev/null
echo "=== max_batch=1 (serialized) ==="
curl -s -m 300 -X POST http://127.0.0.1:18080/v1/load_model \
  -H 'content-type: application/json' \
  -d '{"model":"gpt-oss-20b-synthetic","max_seq":1024,"max_batch":1}' | head -c 60; echo
fire4
fire4
grep "PROFILE..SLOT" gpurun_out/slots_shard.log | tail -4
