#!/usr/bin/env bash
# Elastic remote rollout instance on a spot/preemptible MI355X node
# (the reference's launch_sglang.sh lifecycle): starts the
# HTTP engine server and self-registers with the scheduler endpoint.
set -ex
MODEL=${MODEL:-qwen2.5-1.5b}
MANAGER=${MANAGER:-http://head-node:5000}
python -m polyrl_amd.server.engine_server \
  --model "$MODEL" --port "${PORT:-30001}" --kv-gb "${KV_GB:-64}" \
  --manager "$MANAGER" "$@"
