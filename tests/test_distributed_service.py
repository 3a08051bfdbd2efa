"""DistributedSwarmsDB tests: 2 processes over gloo (CPU).

Covers the tick-synchronized control plane (consistent registry across
ranks), cross-rank p2p delivery, broadcast, visibility, and owner-local
reads — the service-level counterpart of BASELINE config 4.
"""

import os
import subprocess
import sys
import textwrap
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent

WORKER = textwrap.dedent(
    """
    import json, sys
    sys.path.insert(0, %r)
    import numpy as np
    import torch.distributed as dist

    from swarmdb_amd import MessagePriority, QueueConfig
    from swarmdb_amd.parallel.service import DistributedSwarmsDB

    dist.init_process_group(backend="gloo")
    rank = dist.get_rank()
    world = dist.get_world_size()

    cfg = QueueConfig(use_gpu=False, auto_save=False, max_agents=128)
    svc = DistributedSwarmsDB(config=cfg)

    agents = [f"agent{i}" for i in range(8)]
    # both ranks register overlapping agent sets; indices must converge
    for a in agents[rank::2] + ["shared"]:
        svc.register_agent(a)
    svc.tick()
    svc.tick()

    # registry replicated and identical
    assert svc.registered_agents == set(agents) | {"shared"}, (
        rank, svc.registered_agents)
    idx_table = {a: svc._agent_idx[a] for a in sorted(svc._agent_idx)}
    gathered = [None] * world
    dist.all_gather_object(gathered, idx_table)
    assert gathered[0] == gathered[1], "index tables diverged"

    # each agent is active only on its owner rank
    for a in agents:
        active = bool(svc.engine.active_agents()[svc._agent_idx[a]])
        assert active == svc.is_local(a), (a, rank)

    # cross-rank p2p: every agent messages every other agent
    ids = {}
    for s in agents:
        if not svc.is_local(s):
            continue
        for r in agents:
            if r != s:
                ids[(s, r)] = svc.send_message(
                    s, f"{s}->{r}", receiver_id=r,
                    priority=MessagePriority.HIGH)
    svc.tick()

    got = 0
    for r in agents:
        if not svc.is_local(r):
            continue
        msgs = svc.receive_messages(r, timeout=0)
        got += len(msgs)
        senders = {m.sender_id for m in msgs}
        assert senders == set(a for a in agents if a != r), (r, senders)
        for m in msgs:
            assert m.content == f"{m.sender_id}->{r}"
            assert m.priority == MessagePriority.HIGH
            assert m.id == ids.get((m.sender_id, r), m.id)
    t = __import__("torch").tensor([got])
    dist.all_reduce(t)
    assert int(t[0]) == 8 * 7, int(t[0])

    # broadcast from rank 0's first local agent reaches everyone else
    bsender = next(a for a in agents if svc.owner_rank(a) == 0)
    if rank == 0:
        bid = svc.broadcast_message(bsender, {"note": "all hands"})
    svc.tick()
    bgot = 0
    for r in agents + ["shared"]:
        if not svc.is_local(r) or r == bsender:
            continue
        msgs = svc.receive_messages(r, timeout=0)
        for m in msgs:
            assert m.receiver_id is None
            assert m.content == {"note": "all hands"}
            bgot += 1
    if svc.is_local(bsender):
        assert svc.receive_messages(bsender, timeout=0) == []
    t = __import__("torch").tensor([bgot])
    dist.all_reduce(t)
    assert int(t[0]) == 8, int(t[0])  # 9 registered minus the sender

    # groups replicate
    svc.add_agent_group("team", agents[:4])
    svc.tick()
    assert svc.get_agent_groups() == {"team": agents[:4]}

    # stats visible on each rank for its shard
    stats = svc.get_stats()
    assert stats["active_agents"] == 9

    if rank == 0:
        print(json.dumps({"ok": True}))
    svc.config.auto_save = False
    dist.destroy_process_group()
    """
) % str(REPO)


def test_distributed_service_world2(tmp_path):
    script = tmp_path / "svc_worker.py"
    script.write_text(WORKER)
    env = dict(os.environ)
    env.setdefault("GLOO_SOCKET_IFNAME", "lo")
    proc = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node=2",
            "--master-addr=127.0.0.1", "--master-port=29519",
            str(script),
        ],
        capture_output=True, text=True, timeout=240, env=env, cwd=str(REPO),
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert '"ok": true' in proc.stdout.lower()
