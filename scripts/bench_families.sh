#!/usr/bin/env bash
# Per-family end-to-end decode benches — run at round end, EVERY round:
# a headline-only bench let a 3.5x MoE regression sit undetected for
# most of round 2 (profiles/r02_moe_regression.md).
set -uo pipefail
run() { # run <label> <args...>
  local label=$1; shift
  printf "%-28s " "$label"
  timeout 300 python bench.py "$@" 2>/dev/null | tail -1 | \
    python3 -c "import json,sys; d=json.loads(sys.stdin.read()); print(d['value'], 'tok/s', d['ms_per_step'], 'ms/step')" \
    || echo FAILED
}
run qwen-32b-int8-b64   --steps 12 --warmup 4
run qwen-32b-int4-b64   --steps 8 --warmup 3 --quant int4
run qwen-32b-int8-2kctx --steps 6 --warmup 2 --mb-size 32 --prompt-len 2048 --smax 2560
run gpt-oss-bf16-b32    --model gpt-oss-20b --quant bf16 --mb-size 32 --steps 6 --warmup 2
run gpt-oss-mxfp4-b32   --model gpt-oss-20b --quant mxfp4 --mb-size 32 --steps 6 --warmup 2
run mixtral-int8-b32    --model mixtral-8x7b --quant int8 --mb-size 32 --steps 6 --warmup 2
run deepseek-bf16-b16   --model deepseek-v2-lite --quant bf16 --mb-size 16 --steps 6 --warmup 2
run llama8b-bf16-b64    --model llama-3-8b --quant bf16 --steps 8 --warmup 3
run qwen3-8b-int8-b64   --model qwen3-8b --quant int8 --steps 8 --warmup 3
