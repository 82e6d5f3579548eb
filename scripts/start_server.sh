#!/usr/bin/env bash
# Start the dts_amd server (parity: reference scripts/start_server.sh).
set -euo pipefail
cd "$(dirname "$0")/.."
MODEL="${DTS_MODEL:-llama-3-8b}"
PORT="${DTS_PORT:-8000}"
python -m dts_amd.ops.build
exec python -m dts_amd.server --model "$MODEL" --port "$PORT"
