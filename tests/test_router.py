"""Cross-GPU router tests on CPU: gloo backend, world_size 2 subprocesses.

Covers the all-to-all exchange path (BASELINE config 4's transport) that
the driver's 8-GPU scaling bench exercises with RCCL.
"""

import json
import os
import subprocess
import sys
import textwrap
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent

WORKER = textwrap.dedent(
    """
    import json, os, sys
    sys.path.insert(0, %r)
    import numpy as np
    import torch
    import torch.distributed as dist

    from swarmdb_amd.core.config import QueueConfig
    from swarmdb_amd.parallel.router import CrossGpuRouter
    from swarmdb_amd.runtime.cpu_engine import CpuEngine
    from swarmdb_amd.runtime.engine import (
        BROADCAST, NO_BITMAP, REC_DTYPE, VIS_ALL,
    )

    dist.init_process_group(backend="gloo")
    rank = dist.get_rank()
    world = dist.get_world_size()
    rng = np.random.default_rng(100 + rank)

    agents_global = 16
    cfg = QueueConfig(use_gpu=False, max_agents=64, auto_save=False)
    eng = CpuEngine(cfg)
    local_agents = np.arange(rank, agents_global, world, dtype=np.uint32)
    for a in local_agents:
        eng.register_agent(int(a))

    router = CrossGpuRouter(torch.device("cpu"))

    n = 40
    plen = 32
    recs = np.zeros(n, dtype=REC_DTYPE)
    recs["sender"] = rng.choice(local_agents, n)
    recv = rng.integers(0, agents_global, n).astype(np.uint32)
    recv[: 4] = BROADCAST  # 4 broadcasts per rank
    recs["receiver"] = recv
    recs["type"] = 0
    recs["priority"] = 1
    recs["vis_mode"] = VIS_ALL
    recs["bitmap"] = NO_BITMAP
    recs["payload_len"] = plen
    recs["content_len"] = plen
    recs["payload_off"] = np.arange(n, dtype=np.uint64) * plen
    # payload encodes (rank, msg index) for integrity checking
    payload = b"".join(
        (b"r%%dm%%04d" %% (rank, i)).ljust(plen, b".") for i in range(n)
    )

    in_recs, in_pay = router.route(recs, payload)

    # every inbound p2p message must belong to this rank
    p2p = in_recs[in_recs["receiver"] != BROADCAST]
    assert (p2p["receiver"] %% world == rank).all(), "misrouted p2p message"
    # broadcasts from every rank arrive exactly once
    nb = int((in_recs["receiver"] == BROADCAST).sum())
    assert nb == 4 * world, f"expected {4*world} broadcasts, got {nb}"

    # payload integrity through the exchange
    src = np.frombuffer(in_pay, dtype=np.uint8)
    for i in range(len(in_recs)):
        o = int(in_recs["payload_off"][i]); l = int(in_recs["payload_len"][i])
        blob = src[o:o+l].tobytes()
        assert blob[:1] == b"r" and blob.endswith(b"."), blob

    # enqueue locally and drain: conservation check
    eng.enqueue_batch(in_recs, in_pay)
    counts, seqs = eng.receive_many(local_agents, 1000)
    delivered = int(counts.sum())

    t = torch.tensor([delivered, len(p2p)], dtype=torch.int64)
    dist.all_reduce(t)
    total_delivered, total_p2p = int(t[0]), int(t[1])
    # conservation: every p2p delivered once; every broadcast delivered to
    # (local actives) minus nobody (visible to all) on each rank
    expect_bcast = 4 * world * agents_global  # replicated to every agent
    assert total_delivered == total_p2p + expect_bcast, (
        total_delivered, total_p2p, expect_bcast)

    if rank == 0:
        print(json.dumps({"ok": True, "delivered": total_delivered}))
    dist.destroy_process_group()
    """
) % str(REPO)


def test_router_world2(tmp_path):
    script = tmp_path / "worker.py"
    script.write_text(WORKER)
    env = dict(os.environ)
    env.setdefault("GLOO_SOCKET_IFNAME", "lo")
    proc = subprocess.run(
        [
            sys.executable,
            "-m",
            "torch.distributed.run",
            "--nnodes=1",
            "--nproc-per-node=2",
            "--master-addr=127.0.0.1",
            "--master-port=29511",
            str(script),
        ],
        capture_output=True,
        text=True,
        timeout=180,
        env=env,
        cwd=str(REPO),
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert '"ok": true' in proc.stdout.lower()
