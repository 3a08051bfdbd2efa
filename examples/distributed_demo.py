#!/usr/bin/env python3
"""Distributed service demo — agent-sharded SwarmsDB across ranks.

Run (CPU, gloo):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 examples/distributed_demo.py

On a multi-GPU node the same script runs one rank per GPU (nccl = RCCL
over xGMI picks up automatically when devices are visible).
"""

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402

from swarmdb_amd import MessagePriority, QueueConfig  # noqa: E402
from swarmdb_amd.parallel.service import DistributedSwarmsDB  # noqa: E402


def main() -> None:
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend=backend)
    rank = dist.get_rank()

    cfg = QueueConfig(auto_save=False, max_agents=128,
                      use_gpu=torch.cuda.is_available())
    svc = DistributedSwarmsDB(config=cfg)

    agents = [f"agent{i}" for i in range(6)]
    for a in agents:
        svc.register_agent(a)
    svc.tick()  # control-plane sync: registry now identical on all ranks

    if rank == 0:
        local = [a for a in agents if svc.is_local(a)]
        print(f"[rank {rank}] owns {local}")

    # every rank sends from its local agents to everyone
    for s in agents:
        if svc.is_local(s):
            for r in agents:
                if r != s:
                    svc.send_message(s, f"{s} -> {r}", receiver_id=r,
                                     priority=MessagePriority.NORMAL)
    svc.tick()  # data-plane all-to-all: messages land on owner ranks

    for r in agents:
        if svc.is_local(r):
            msgs = svc.receive_messages(r, timeout=0)
            print(f"[rank {rank}] {r} received {len(msgs)} messages")

    stats = svc.get_stats()
    if rank == 0:
        print(f"[rank {rank}] node stats: {stats['messages_by_status']}")

    # live re-sharding: move agent0 to the other rank, pending traffic
    # re-homed through the exchange
    world = dist.get_world_size()
    if world > 1:
        target = (svc.owner_rank("agent0") + 1) % world
        if rank == 0:
            svc.migrate_agent("agent0", target)
        svc.tick()  # op applies + old owner drains
        svc.tick()  # re-homed records deliver
        if svc.is_local("agent0"):
            print(f"[rank {rank}] agent0 migrated here "
                  f"(owner={svc.owner_rank('agent0')})")
    svc.config.auto_save = False
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
