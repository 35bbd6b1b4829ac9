#!/bin/bash
# Full-HTTP-stack saturation check: loadgen at an over-capacity rate
# against one llama-3-8b engine through the real gateway (SSE streaming,
# auth off, accounting on). Compares the achieved req/s with the bare
# engine's open-loop capacity to show where the HTTP layer stands.
set -x
mkdir -p gpurun_out
cat > providers.json <<'JSON'
[ { "local-llama-8b": { "baseUrl": "local://llama-3-8b?device=0", "apikey": "" } } ]
JSON
cat > models_fallback_rules.json <<'JSON'
[ { "gateway_model_name": "llmgateway/llama-8b",
    "fallback_models": [ { "provider": "local-llama-8b", "model": "llama-3-8b" } ] } ]
JSON
python main.py > gpurun_out/fs_server.log 2>&1 &
SRV=$!
sleep 8
timeout 240 python tools/loadgen.py --qps 2 --duration 5 --max-tokens 8 --model llmgateway/llama-8b > gpurun_out/fs_warm.json 2>&1
timeout 120 python tools/loadgen.py --qps 320 --duration 25 --max-tokens 64 --model llmgateway/llama-8b > gpurun_out/fs_sat.json 2>&1
kill $SRV 2>/dev/null
echo "=== saturation run ==="; cat gpurun_out/fs_sat.json
