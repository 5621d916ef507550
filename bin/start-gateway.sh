#!/usr/bin/env bash
# Gateway launcher — the native analogue of the reference's
# bin/start-gateway.sh (parse + validate the config, then exec the
# server).  Validation lives in the typed loader
# (resilient_llm_amd/config.py) instead of inline regexes.
#
# Usage: ./bin/start-gateway.sh [config/config.yaml] [port]
set -euo pipefail
cd "$(dirname "$0")/.."

CONFIG="${1:-config/config.yaml}"
PORT="${2:-}"

if ! python - "$CONFIG" <<'PY'
import sys
from resilient_llm_amd.config import load_config
try:
    cfg = load_config(sys.argv[1])
except Exception as e:
    print(f"config validation failed: {e}", file=sys.stderr)
    sys.exit(1)
print(f"config ok: {len(cfg.deployments)} deployments, "
      f"port {cfg.cluster.port}, aliases {cfg.aliases}")
PY
then
  echo "falling back to the stub config" >&2
  CONFIG=config/config.stub.yaml
fi

exec python -m resilient_llm_amd.gateway.server --config "$CONFIG" \
  ${PORT:+--port "$PORT"}
