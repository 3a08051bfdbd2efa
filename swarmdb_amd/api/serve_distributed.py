"""Per-rank API server over a DistributedSwarmsDB shard.

Launch one process per GPU under torch.distributed (RCCL over xGMI on
GPUs, gloo on CPU); each rank serves the full REST surface on
``base port + rank`` and 307-redirects agent-scoped requests to the
owner rank (the distributed gateway). The tick loop runs on a
background thread so the control/data exchange keeps flowing while
uvicorn serves.

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 -m swarmdb_amd.api.serve_distributed

Env:
    SWARMDB_BASE_PORT      first rank's port (default 8000; rank r
                           serves on base+r)
    SWARMDB_PEER_HOST      host peers advertise in redirects
                           (default 127.0.0.1)
    SWARMDB_PEER_URLS      comma-separated explicit peer URLs
                           (overrides host/port derivation)
    SWARMDB_TICK_INTERVAL  tick period seconds (default 0.002)

plus the standard config tiers (swarmdb_amd/core/config.py).
"""

from __future__ import annotations

import os


def build(world_size_override: int | None = None):
    """Construct (service, app) for this rank. Split from main() so
    tests can drive the glue without a socket server."""
    import torch
    import torch.distributed as dist

    from ..core.config import QueueConfig
    from ..parallel.service import DistributedSwarmsDB
    from .app import ApiSettings, create_app

    if not dist.is_initialized():
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(backend=backend)
    rank = dist.get_rank()
    world = world_size_override or dist.get_world_size()

    base_port = int(os.environ.get("SWARMDB_BASE_PORT", "8000"))
    host = os.environ.get("SWARMDB_PEER_HOST", "127.0.0.1")
    peers_env = os.environ.get("SWARMDB_PEER_URLS")
    if peers_env:
        peers = [u.strip() for u in peers_env.split(",")]
    else:
        peers = [f"http://{host}:{base_port + r}" for r in range(world)]

    cfg = QueueConfig.from_env()
    cfg.world_size = world
    cfg.rank = rank
    if torch.cuda.is_available():
        cfg.device_index = rank % max(1, torch.cuda.device_count())
    svc = DistributedSwarmsDB(config=cfg)
    svc.start_ticker(float(os.environ.get("SWARMDB_TICK_INTERVAL", "0.002")))
    app = create_app(db=svc, settings=ApiSettings(), peer_urls=peers)
    return svc, app, base_port + rank


def main() -> None:
    import uvicorn

    svc, app, port = build()
    try:
        uvicorn.run(app, host="0.0.0.0", port=port, log_level="info")
    finally:
        svc.close()


if __name__ == "__main__":
    main()
