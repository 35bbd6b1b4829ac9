#!/usr/bin/env bash
# Entry checks mirroring the reference docker/entrypoint.sh: require the API
# key and both config files before starting; fail fast otherwise.
set -euo pipefail

if [[ -z "${GATEWAY_API_KEY:-}" ]]; then
    echo "FATAL: GATEWAY_API_KEY is not set" >&2
    exit 1
fi
for f in providers.json models_fallback_rules.json; do
    if [[ ! -f "/app/$f" ]]; then
        echo "FATAL: /app/$f not found (mount it as a volume)" >&2
        exit 1
    fi
done

# MI355X multi-process GPU work needs dmabuf IPC
export HSA_ENABLE_IPC_MODE_LEGACY=0

exec python /app/main.py
