#!/usr/bin/env python3
"""The five BASELINE.json benchmark configs.

  1 cpu-loopback   single-agent register + send/receive loopback,
                   in-process CPU queue (plumbing)
  2 p2p-gpu        2 agents point-to-point, GPU ring on 1 MI355X, 1 KB msgs
  3 group-fanout   256 agents / 8 groups, group-broadcast fan-out on
                   1 MI355X, priority-sort dequeue
  4 alltoall       4096 agents sharded across N GPUs, RCCL all-to-all
                   (launch under torch.distributed.run; this is bench.py
                   with --agents 4096/N)
  5 loadbalancer   8 mock backends, least-loaded dispatch + JSON history
                   spill

Usage:
  python -m benchmarks.run --config {1,2,3,5} [--seconds S]
  python -m torch.distributed.run --nproc-per-node N benchmarks/run.py --config 4

Each config prints one JSON line:
  {"config": n, "name": ..., "messages_per_s": ..., "p50_latency_ms": ...}
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from swarmdb_amd import MessagePriority, QueueConfig, SwarmsDB  # noqa: E402
from swarmdb_amd.runtime.engine import (  # noqa: E402
    NO_BITMAP,
    REC_DTYPE,
    VIS_ALL,
)


def _emit(config: int, name: str, msgs: int, elapsed: float, lat_ms: float,
          extra=None):
    out = {
        "config": config,
        "name": name,
        "messages_per_s": round(msgs / elapsed, 1),
        "p50_latency_ms": round(lat_ms, 4),
        "messages": msgs,
        "seconds": round(elapsed, 3),
    }
    if extra:
        out.update(extra)
    print(json.dumps(out))


def _gpu_available() -> bool:
    try:
        import torch

        return torch.cuda.is_available()
    except Exception:
        return False


def config1_cpu_loopback(seconds: float) -> None:
    """Single agent, in-process CPU queue: the reference-shaped per-message
    API path (register + send_message + receive_messages loop)."""
    cfg = QueueConfig(use_gpu=False, auto_save=False, max_agents=64)
    db = SwarmsDB(config=cfg)
    db.register_agent("loop")
    payload = "x" * 1024
    lat = []
    n = 0
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < seconds:
        s = time.perf_counter()
        db.send_message("loop", payload, receiver_id="loop")
        got = db.receive_messages("loop", timeout=0)
        lat.append(time.perf_counter() - s)
        assert len(got) == 1
        n += 1
    elapsed = time.perf_counter() - t0
    db.close()
    _emit(1, "cpu-loopback", n, elapsed, float(np.median(lat) * 1000))


def _make_batch(rng, n, senders, receivers, payload_bytes):
    stride = (payload_bytes + 15) // 16 * 16
    recs = np.zeros(n, dtype=REC_DTYPE)
    recs["sender"] = rng.choice(senders, n)
    recs["receiver"] = rng.choice(receivers, n)
    recs["priority"] = rng.integers(0, 4, n)
    recs["timestamp"] = time.time()
    recs["vis_mode"] = VIS_ALL
    recs["bitmap"] = NO_BITMAP
    recs["payload_len"] = payload_bytes
    recs["content_len"] = payload_bytes
    recs["payload_off"] = np.arange(n, dtype=np.uint64) * stride
    payload = rng.integers(32, 127, n * stride, dtype=np.uint8).tobytes()
    return recs, payload


def config2_p2p_gpu(seconds: float) -> None:
    """2 agents point-to-point over the GPU ring, 1 KB chat messages."""
    from swarmdb_amd.runtime.gpu_engine import GpuEngine

    cfg = QueueConfig(use_gpu=True, auto_save=False, max_agents=64,
                      num_slots=1 << 20, staging_batch=16384)
    eng = GpuEngine(cfg)
    eng.register_agent(0)
    eng.register_agent(1)
    rng = np.random.default_rng(0)
    # a single inbox drains at most one dequeue window (4096) per call,
    # so the tick size matches it — larger batches would silently back up
    batch = 4096
    recs, payload = _make_batch(rng, batch, np.array([0]), np.array([1]), 1024)
    agents = np.array([1], dtype=np.uint32)
    lat = []
    n = 0
    # warmup
    for _ in range(3):
        eng.enqueue_batch(recs, payload)
        eng.receive_many(agents, batch)
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < seconds:
        s = time.perf_counter()
        eng.enqueue_batch(recs, payload)
        counts, seqs = eng.receive_many(agents, batch)
        eng.deliver_payloads(seqs, 1024)
        lat.append(time.perf_counter() - s)
        assert int(counts.sum()) == batch, "p2p tick failed to drain"
        n += int(counts.sum())
    elapsed = time.perf_counter() - t0
    eng.close()

    # single-message regime: the express doorbell lane (persistent
    # kernel, pinned mailboxes) is the latency plane for this config
    from swarmdb_amd import _swarmq

    db = _swarmq.DoorbellQueue(slot_bytes=1024, sub_cap=256, n_agents=2,
                               ring_cap=64, device=0)
    db.start(30.0)
    try:
        pay = b"x" * 1024
        for _ in range(20):
            db.send(receiver=1, sender=0, payload=pay)
            db.recv_spin(1, timeout_us=2e6)
        ex = []
        for _ in range(300):
            s = time.perf_counter()
            db.send(receiver=1, sender=0, payload=pay)
            assert db.recv_spin(1, timeout_us=2e6) is not None
            ex.append(time.perf_counter() - s)
        express_p50_us = float(np.median(ex) * 1e6)
        express_p99_us = float(np.percentile(ex, 99) * 1e6)
    finally:
        db.stop()
        db.release()
    _emit(2, "p2p-gpu-1kb", n, elapsed, float(np.median(lat) * 1000),
          {"express_p50_us": round(express_p50_us, 1),
           "express_p99_us": round(express_p99_us, 1)})


def config3_group_fanout(seconds: float) -> None:
    """256 agents / 8 groups of 32; group messages via the single-slot
    fan-out kernel; priority-ordered dequeue (the priority-sort kernel)."""
    from swarmdb_amd.runtime.engine import VIS_GROUP
    from swarmdb_amd.runtime.gpu_engine import GpuEngine

    n_agents, n_groups = 256, 8
    cfg = QueueConfig(use_gpu=True, auto_save=False, max_agents=256,
                      num_slots=1 << 20, staging_batch=16384,
                      inbox_capacity=1 << 16)
    eng = GpuEngine(cfg)
    for a in range(n_agents):
        eng.register_agent(a)
    rng = np.random.default_rng(0)
    group_bitmaps = []
    for gi in range(n_groups):
        bits = np.zeros(cfg.max_agents, dtype=bool)
        bits[gi * 32 : (gi + 1) * 32] = True
        group_bitmaps.append(eng.alloc_bitmap(bits))
    batch = 4096  # group messages per tick; each fans out to 32 members
    stride = 1024
    recs = np.zeros(batch, dtype=REC_DTYPE)
    gidx = rng.integers(0, n_groups, batch)
    recs["sender"] = (gidx * 32).astype(np.uint32)  # a member of the group
    recs["receiver"] = 0xFFFFFFFF  # BROADCAST routing, group-filtered
    recs["priority"] = rng.integers(0, 4, batch)
    recs["vis_mode"] = VIS_GROUP
    recs["bitmap"] = np.array(group_bitmaps, dtype=np.uint32)[gidx]
    recs["payload_len"] = stride
    recs["content_len"] = stride
    recs["payload_off"] = np.arange(batch, dtype=np.uint64) * stride
    payload = rng.integers(32, 127, batch * stride, dtype=np.uint8).tobytes()
    agents = np.arange(n_agents, dtype=np.uint32)
    K = batch * 32 // n_agents * 2
    lat = []
    delivered = 0
    for _ in range(3):
        eng.enqueue_batch(recs, payload)
        eng.receive_many(agents, K, priority_order=True)
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < seconds:
        s = time.perf_counter()
        eng.enqueue_batch(recs, payload)
        counts, seqs = eng.receive_many(agents, K, priority_order=True)
        eng.deliver_payloads(seqs, stride)
        lat.append(time.perf_counter() - s)
        delivered += int(counts.sum())
    elapsed = time.perf_counter() - t0
    eng.close()
    _emit(3, "group-fanout-priority", delivered, elapsed,
          float(np.median(lat) * 1000),
          {"groups": n_groups, "agents": n_agents,
           "fanout_per_group_msg": 32})


def config6_search(seconds: float) -> None:
    """Bonus: device content search + filtered query over a large
    HBM-resident log (the reference does Python linear scans,
    swarmdb/ main.py:671-781)."""
    from swarmdb_amd.runtime.gpu_engine import GpuEngine

    n_msgs = 1 << 20  # 1M x 1KB resident messages
    cfg = QueueConfig(use_gpu=True, auto_save=False, max_agents=1024,
                      num_slots=n_msgs, slot_bytes=1024 + 64,
                      staging_batch=1 << 16, inbox_capacity=1 << 12)
    eng = GpuEngine(cfg)
    rng = np.random.default_rng(0)
    agents = np.arange(64, dtype=np.uint32)
    for a in agents:
        eng.register_agent(int(a))
    # fill the log; plant the needle in ~1/4096 messages
    batch = 1 << 16
    stride = 1024
    for b in range(n_msgs // batch):
        recs, payload = _make_batch(rng, batch, agents, agents, stride)
        pay = bytearray(payload)
        for i in range(0, batch, 4096):
            off = int(recs["payload_off"][i]) + 100
            pay[off : off + 12] = b"NEEDLE-%05d" % (b % 10)
        eng.enqueue_batch(recs, bytes(pay))
    total_bytes = n_msgs * stride

    searches = 0
    t0 = time.perf_counter()
    lat = []
    while time.perf_counter() - t0 < seconds:
        s = time.perf_counter()
        hits = eng.search(b"NEEDLE-00003", case_sensitive=True, limit=16384)
        lat.append(time.perf_counter() - s)
        assert len(hits) > 0
        searches += 1
    elapsed = time.perf_counter() - t0
    qlat = []
    for _ in range(20):
        s = time.perf_counter()
        eng.query(sender=3, type_code=0, limit=10000)
        qlat.append(time.perf_counter() - s)
    eng.close()
    print(json.dumps({
        "config": 6,
        "name": "device-content-search",
        "resident_messages": n_msgs,
        "resident_payload_gb": round(total_bytes / 1e9, 2),
        "searches_per_s": round(searches / elapsed, 2),
        "search_p50_ms": round(float(np.median(lat)) * 1000, 3),
        "scan_gb_per_s": round(total_bytes * searches / elapsed / 1e9, 1),
        "query_p50_ms": round(float(np.median(qlat)) * 1000, 3),
    }))


def config9_distributed_service(seconds: float) -> None:
    """Distributed service tick throughput (DistributedSwarmsDB).
    Launch under torch.distributed.run; gloo on CPU, RCCL on GPUs:

      python -m torch.distributed.run --nproc-per-node 2 \
        --master-addr 127.0.0.1 benchmarks/run.py --config 9
    """
    import torch
    import torch.distributed as dist

    from swarmdb_amd.parallel.service import DistributedSwarmsDB

    if not dist.is_initialized():
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(backend=backend)
    rank = dist.get_rank()
    world = dist.get_world_size()
    cfg = QueueConfig(auto_save=False, max_agents=1024,
                      use_gpu=torch.cuda.is_available())
    svc = DistributedSwarmsDB(config=cfg)
    agents = [f"agent{i}" for i in range(256)]
    for a in agents:
        svc.register_agent(a)
    svc.tick()
    local = [a for a in agents if svc.is_local(a)]
    rng = np.random.default_rng(rank)

    sent = 0
    delivered = 0
    ticks = 0
    lat = []
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < seconds:
        s = time.perf_counter()
        for _ in range(64):  # per-message API sends per tick per rank
            snd = local[int(rng.integers(0, len(local)))]
            rcv = agents[int(rng.integers(0, len(agents)))]
            if rcv != snd:
                svc.send_message(snd, "tick payload " + "x" * 200,
                                 receiver_id=rcv)
                sent += 1
        svc.tick()
        for a in local:
            delivered += len(svc.receive_messages(a, timeout=0))
        lat.append(time.perf_counter() - s)
        ticks += 1
    elapsed = time.perf_counter() - t0
    t = torch.tensor(
        [float(sent), float(delivered)],
        device="cuda" if dist.get_backend() == "nccl" else "cpu",
    )
    dist.all_reduce(t)
    if rank == 0:
        print(json.dumps({
            "config": 9,
            "name": "distributed-service",
            "world": world,
            "backend": dist.get_backend(),
            "messages_per_s": round(float(t[1]) / elapsed, 1),
            "tick_p50_ms": round(float(np.median(lat)) * 1000, 2),
            "ticks": ticks,
            "sent": int(t[0]),
            "delivered": int(t[1]),
        }))
    svc.config.auto_save = False
    dist.destroy_process_group()


def config8_hbm_scale(seconds: float) -> None:
    """Bonus: HBM-scale residency — fill a 64 GB / 32M-message log in
    device memory (the 288 GB HBM3E is the store, not a cache) and run
    full-content search + filtered queries against it."""
    from swarmdb_amd.runtime.gpu_engine import GpuEngine

    n_msgs = 1 << 25          # 32M messages
    slot = 2048               # 64 GiB of payload slots
    plen = 1984
    cfg = QueueConfig(use_gpu=True, auto_save=False, max_agents=64,
                      num_slots=n_msgs, slot_bytes=slot,
                      staging_batch=1 << 16, inbox_capacity=1 << 10)
    eng = GpuEngine(cfg)
    rng = np.random.default_rng(0)
    agents = np.arange(64, dtype=np.uint32)
    for a in agents:
        eng.register_agent(int(a))

    batch = 1 << 16
    recs, payload = _make_batch(rng, batch, agents, agents, plen)
    pay = bytearray(payload)
    stride = (plen + 15) // 16 * 16
    for i in range(0, batch, 1024):   # needle in 1/1024 messages
        off = i * stride + 500
        pay[off : off + 10] = b"DEEPNEEDLE"
    payload = bytes(pay)

    pr = eng.q.alloc_pinned(recs.nbytes)
    np.frombuffer(pr, dtype=np.uint8)[:] = np.frombuffer(
        np.ascontiguousarray(recs).tobytes(), np.uint8)
    pp = eng.q.alloc_pinned(len(payload))
    np.frombuffer(pp, dtype=np.uint8)[:] = np.frombuffer(payload, np.uint8)

    t0 = time.perf_counter()
    nfill = n_msgs // batch
    for it in range(nfill):
        s = it % 2
        eng.q.prefetch_from(s, pr.__array_interface__["data"][0],
                            pp.__array_interface__["data"][0], batch,
                            len(payload))
        eng.q.enqueue_staged(s)
    eng.q.sync()
    fill_s = time.perf_counter() - t0
    resident_gb = n_msgs * slot / 2**30

    # (a) FULL scan: absent needle forces a pass over every content byte
    full_lat = []
    for _ in range(5):
        s = time.perf_counter()
        hits = eng.search(b"NO-SUCH-NEEDLE-ANYWHERE", case_sensitive=True,
                          limit=4096)
        full_lat.append(time.perf_counter() - s)
        assert len(hits) == 0
    # (b) early-exit: stop once `limit` planted needles are found
    lat = []
    nsearch = 0
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < seconds:
        s = time.perf_counter()
        hits = eng.search(b"DEEPNEEDLE", case_sensitive=True, limit=4096)
        lat.append(time.perf_counter() - s)
        assert len(hits) > 0
        nsearch += 1
    qlat = []
    for _ in range(10):
        s = time.perf_counter()
        eng.query(sender=3, limit=10000)
        qlat.append(time.perf_counter() - s)
    eng.close()
    full_p50 = float(np.median(full_lat))
    print(json.dumps({
        "config": 8,
        "name": "hbm-scale-64gb-log",
        "resident_messages": n_msgs,
        "resident_slots_gb": round(resident_gb, 1),
        "fill_s": round(fill_s, 2),
        "fill_msgs_per_s": round(n_msgs / fill_s, 0),
        "full_scan_p50_ms": round(full_p50 * 1000, 2),
        "full_scan_tb_per_s": round(resident_gb / 1024 / full_p50, 2),
        "first4096_hits_p50_ms": round(float(np.median(lat)) * 1000, 2),
        "query_p50_ms": round(float(np.median(qlat)) * 1000, 2),
        "searches": nsearch,
    }))


def config7_checkpoint(seconds: float) -> None:
    """Bonus: checkpoint/restore throughput — JSON history spill of a
    GPU-resident store (device->pinned gather on the copy stream + host
    serialization) and the batched replay load."""
    import shutil
    import tempfile

    from swarmdb_amd.runtime.gpu_engine import GpuEngine

    import os

    tmp = tempfile.mkdtemp(prefix="swarmdb_ckpt_")
    # 128k default; SWARMQ_CKPT_MSGS=1048576 for the 1M-message run
    n_msgs = int(os.environ.get("SWARMQ_CKPT_MSGS", str(1 << 17)))
    cfg = QueueConfig(use_gpu=_gpu_available(), auto_save=False,
                      max_agents=256, num_slots=n_msgs,
                      slot_bytes=512, staging_batch=1 << 14,
                      save_dir=tmp)
    db = SwarmsDB(config=cfg)
    rng = np.random.default_rng(0)
    agents = np.arange(64)
    for a in agents:
        db.agent_index(f"agent{a}")
    idxs = np.array([db.agent_index(f"agent{a}") for a in agents])
    batch = 1 << 14
    for _ in range(n_msgs // batch):
        recs, payload = _make_batch(rng, batch, idxs, idxs, 256)
        db.send_batch(recs, payload)

    t0 = time.perf_counter()
    path = db.save_message_history()
    save_s = time.perf_counter() - t0
    size_mb = Path(path).stat().st_size / 1e6

    # binary checkpoint plane (device-gather speed, zero per-message
    # Python) + delta append + reload
    t0 = time.perf_counter()
    bpath = db.save_checkpoint()
    bsave_s = time.perf_counter() - t0
    bsize_mb = Path(bpath).stat().st_size / 1e6
    extra = 1 << 14
    recs, payload = _make_batch(rng, extra, idxs, idxs, 256)
    db.send_batch(recs, payload)
    t0 = time.perf_counter()
    _, n_delta = db.save_checkpoint_delta()
    bdelta_s = time.perf_counter() - t0
    assert n_delta == extra

    cfg2 = QueueConfig(use_gpu=_gpu_available(), auto_save=False,
                       max_agents=256, num_slots=n_msgs, slot_bytes=512,
                       staging_batch=1 << 14, save_dir=tmp)
    db2 = SwarmsDB(config=cfg2)
    t0 = time.perf_counter()
    loaded = db2.load_message_history(path)
    load_s = time.perf_counter() - t0
    assert loaded == n_msgs, (loaded, n_msgs)

    cfg3 = QueueConfig(use_gpu=_gpu_available(), auto_save=False,
                       max_agents=256, num_slots=2 * n_msgs, slot_bytes=512,
                       staging_batch=1 << 14, save_dir=tmp)
    db3 = SwarmsDB(config=cfg3)
    t0 = time.perf_counter()
    bloaded = db3.load_checkpoint(bpath)
    bload_s = time.perf_counter() - t0
    assert bloaded == n_msgs + extra, (bloaded, n_msgs + extra)
    db3.config.auto_save = False
    db3.close()

    db.config.auto_save = False
    db2.config.auto_save = False
    db.close()
    db2.close()
    shutil.rmtree(tmp, ignore_errors=True)
    print(json.dumps({
        "config": 7,
        "name": "checkpoint-restore",
        "messages": n_msgs,
        "file_mb": round(size_mb, 1),
        "save_s": round(save_s, 2),
        "save_msgs_per_s": round(n_msgs / save_s, 0),
        "load_s": round(load_s, 2),
        "load_msgs_per_s": round(n_msgs / load_s, 0),
        "binary_ckpt_mb": round(bsize_mb, 1),
        "binary_save_msgs_per_s": round(n_msgs / bsave_s, 0),
        "binary_delta_msgs_per_s": round(extra / bdelta_s, 0),
        "binary_load_msgs_per_s": round((n_msgs + extra) / bload_s, 0),
        "engine": "gpu" if _gpu_available() else "cpu",
    }))


def config5_loadbalancer(seconds: float) -> None:
    """8 mock backends (1 per GPU on a full node); least-loaded dispatch
    via the wavefront min-reduce kernel at >=100k req/s, with concurrent
    JSON history spill of live queue traffic."""
    use_gpu = _gpu_available()
    cfg = QueueConfig(use_gpu=use_gpu, auto_save=False, max_agents=1024,
                      num_slots=1 << 18, staging_batch=16384)
    db = SwarmsDB(config=cfg)
    for i in range(8):
        db.register_llm_backend(f"backend{i}")
    db.set_llm_load_balancing(True)
    # background queue traffic so the spill has something to write
    rng = np.random.default_rng(0)
    senders = np.arange(0, 64)
    for a in senders:
        db.engine.register_agent(int(a))
    recs, payload = _make_batch(rng, 4096, senders, senders, 256)
    db.engine.enqueue_batch(recs, payload)

    dispatch_batch = 8192
    lat = []
    n = 0
    spills = 0
    t0 = time.perf_counter()
    next_spill = t0 + 1.0
    while time.perf_counter() - t0 < seconds:
        s = time.perf_counter()
        choices = db.engine.dispatch_batch(dispatch_batch, 8)
        lat.append(time.perf_counter() - s)
        n += len(choices)
        # completions keep loads bounded
        loads = db.engine.backend_loads()[:8]
        for b in range(8):
            if loads[b] > 0:
                db.engine.backend_add_load(b, -int(loads[b] * 3 // 4))
        if time.perf_counter() >= next_spill:
            db.save_message_history()  # JSON history spill
            spills += 1
            next_spill += 1.0
    elapsed = time.perf_counter() - t0
    loads = db.engine.backend_loads()[:8]
    db.config.auto_save = False
    db.close()
    _emit(5, "llm-loadbalancer", n, elapsed,
          float(np.median(lat) * 1000 / dispatch_batch),
          {"unit": "requests", "backends": 8, "spills": spills,
           "final_load_spread": int(loads.max() - loads.min()),
           "engine": "gpu" if use_gpu else "cpu"})


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", type=int, required=True,
                    choices=[1, 2, 3, 4, 5, 6, 7, 8, 9])
    ap.add_argument("--seconds", type=float, default=5.0)
    args = ap.parse_args()
    if args.config == 1:
        config1_cpu_loopback(args.seconds)
    elif args.config == 2:
        config2_p2p_gpu(args.seconds)
    elif args.config == 3:
        config3_group_fanout(args.seconds)
    elif args.config == 4:
        # config 4 IS the flagship bench with 4096 global agents
        world = int(os.environ.get("WORLD_SIZE", "1"))
        sys.argv = [
            "bench.py",
            "--agents", str(max(1, 4096 // world)),
            "--steps", "30",
            "--warmup", "5",
        ]
        import bench

        return bench.main()
    elif args.config == 5:
        config5_loadbalancer(args.seconds)
    elif args.config == 6:
        config6_search(args.seconds)
    elif args.config == 7:
        config7_checkpoint(args.seconds)
    elif args.config == 8:
        config8_hbm_scale(args.seconds)
    elif args.config == 9:
        config9_distributed_service(args.seconds)
    return 0


if __name__ == "__main__":
    sys.exit(main())
