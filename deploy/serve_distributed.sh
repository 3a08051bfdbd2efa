#!/usr/bin/env bash
# Sharded REST service: one rank (and one API server) per GPU, RCCL
# over xGMI between shards, agent-scoped requests gateway-routed to
# owner ranks. Rank r serves on $SWARMDB_BASE_PORT + r (default 8000+r).
#
#   ./deploy/serve_distributed.sh 8          # 8 GPUs -> ports 8000..8007
set -euo pipefail
NPROC="${1:-8}"
export HSA_ENABLE_IPC_MODE_LEGACY="${HSA_ENABLE_IPC_MODE_LEGACY:-0}"
exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "$NPROC" \
  --master-addr 127.0.0.1 -m swarmdb_amd.api.serve_distributed
