#!/usr/bin/env bash
# Start the SwarmDB-AMD API server (reference start.sh analog,
# README.md:70-76 of the reference).
#   ./deploy/serve.sh development   # reload, debug logs
#   ./deploy/serve.sh production    # single device-owner worker
set -euo pipefail
cd "$(dirname "$0")/.."

MODE="${1:-development}"
export API_ENV="$MODE"
PORT="${PORT:-8000}"

python build_ext.py

if [ "$MODE" = "production" ]; then
  exec uvicorn swarmdb_amd.api.app:get_app --factory \
    --host 0.0.0.0 --port "$PORT" --workers 1 --log-level info
else
  exec uvicorn swarmdb_amd.api.app:get_app --factory \
    --host 127.0.0.1 --port "$PORT" --reload --log-level debug
fi
