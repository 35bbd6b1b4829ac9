#!/bin/bash
# End-to-end gateway validation on one MI355X (BASELINE configs[2] and [3]):
# 2-model fallback under injected failures (llama-3-8b -> mistral-7b) and
# round-robin rotation across two 8B replica engines, both through the real
# HTTP stack with the loadgen. Run via gpurun from the repo root.
set -x
mkdir -p gpurun_out

cat > providers.json <<'EOF'
[
    { "local-llama-8b": { "baseUrl": "local://llama-3-8b?device=0", "apikey": "" } },
    { "local-mistral":  { "baseUrl": "local://mistral-7b?device=0", "apikey": "" } },
    { "local-llama-8b-r2": { "baseUrl": "local://", "apikey": "",
                             "engine": { "model": "llama-3-8b", "device": 0, "max_batch_size": 128 } } }
]
EOF
cat > models_fallback_rules.json <<'EOF'
[
    { "gateway_model_name": "llmgateway/resilient",
      "fallback_models": [
          { "provider": "local-llama-8b", "model": "llama-3-8b", "retry_count": 1, "retry_delay": 1 },
          { "provider": "local-mistral",  "model": "mistral-7b" } ] },
    { "gateway_model_name": "llmgateway/rotating",
      "rotate_models": true,
      "fallback_models": [
          { "provider": "local-llama-8b",    "model": "llama-3-8b" },
          { "provider": "local-llama-8b-r2", "model": "llama-3-8b" } ] }
]
EOF

python main.py > gpurun_out/server.log 2>&1 &
SRV=$!
sleep 8
# warm both engines of the resilient chain (primary serves; then force the fallback warm)
timeout 180 python tools/loadgen.py --qps 1 --duration 3 --max-tokens 4 --model llmgateway/resilient > gpurun_out/fb_warm1.json 2>&1
curl -s -X POST http://127.0.0.1:9100/v1/admin/engines/local-llama-8b/failures \
     -H 'Content-Type: application/json' -d '{"fail_rate": 1.0}' > gpurun_out/inject.json
timeout 180 python tools/loadgen.py --qps 1 --duration 3 --max-tokens 4 --model llmgateway/resilient > gpurun_out/fb_warm2.json 2>&1
# full fallback load: every request fails on the primary and must stream from mistral
timeout 90 python tools/loadgen.py --qps 40 --duration 10 --max-tokens 32 --model llmgateway/resilient > gpurun_out/fb_run.json 2>&1
curl -s -X POST http://127.0.0.1:9100/v1/admin/engines/local-llama-8b/failures \
     -H 'Content-Type: application/json' -d '{"fail_rate": 0.0}' > /dev/null
# rotation: round-robin between the two replica engines
timeout 180 python tools/loadgen.py --qps 1 --duration 3 --max-tokens 4 --model llmgateway/rotating > gpurun_out/rot_warm.json 2>&1
timeout 90 python tools/loadgen.py --qps 40 --duration 10 --max-tokens 32 --model llmgateway/rotating > gpurun_out/rot_run.json 2>&1
curl -s http://127.0.0.1:9100/metrics | grep -E "gateway_chat_requests_total|fallback_attempts|prefix_cached_tokens_total" > gpurun_out/metrics_tail.txt
curl -s http://127.0.0.1:9100/v1/api/engine-stats > gpurun_out/engstats.json
kill $SRV 2>/dev/null
echo "=== fallback run ==="; cat gpurun_out/fb_run.json
echo "=== rotation run ==="; cat gpurun_out/rot_run.json
echo "=== metrics ==="; cat gpurun_out/metrics_tail.txt
echo "=== engines ==="; python -c "import json;rows=json.load(open('gpurun_out/engstats.json'))['engines'];[print({k: r[k] for k in ('engine','requests','finished','failed','running')}) for r in rows]"
