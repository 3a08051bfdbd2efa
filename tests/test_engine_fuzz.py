"""Randomized fuzz: CpuEngine vs an independent naive oracle.

The CPU engine is the reference the GPU kernels are diffed against
(test_gpu_engine.test_parity_cpu_vs_gpu), so it gets its own independent
check here: a deliberately-simple pure-Python model of the delivery
semantics (per-agent FIFO inbox in seq order, visibility filter,
priority ordering, statuses, counters).
"""

import numpy as np
import pytest

from swarmdb_amd.core.config import QueueConfig
from swarmdb_amd.runtime.cpu_engine import CpuEngine
from swarmdb_amd.runtime.engine import (
    BROADCAST,
    NO_BITMAP,
    REC_DTYPE,
    ST_DELETED,
    ST_DELIVERED,
    ST_READ,
    VIS_ALL,
    VIS_BITMAP,
    VIS_GROUP,
)


class Oracle:
    """Naive reference model. Deliberately different data structures:
    plain dicts and lists, no numpy."""

    def __init__(self, max_agents):
        self.max_agents = max_agents
        self.msgs = {}          # seq -> dict
        self.inbox = {}         # agent -> [seq] (append order)
        self.pending = {}       # agent -> set of undelivered seqs
        self.active = set()
        self.bitmaps = []
        self.count = 0

    def register(self, a):
        self.active.add(a)
        self.inbox.setdefault(a, [])
        self.pending.setdefault(a, set())

    def alloc_bitmap(self, bits):
        self.bitmaps.append(set(np.flatnonzero(bits).tolist()))
        return len(self.bitmaps) - 1

    def enqueue(self, recs):
        out = []
        for r in recs:
            seq = self.count
            self.count += 1
            m = {k: r[k].item() if hasattr(r[k], "item") else r[k]
                 for k in REC_DTYPE.names}
            m["status"] = ST_DELIVERED
            self.msgs[seq] = m
            recv = int(r["receiver"])
            if recv == BROADCAST:
                if int(r["vis_mode"]) == VIS_GROUP:
                    members = self.bitmaps[int(r["bitmap"])] & self.active
                    for a in members:
                        self.inbox[a].append(seq)
                        self.pending[a].add(seq)
                else:
                    for a in sorted(self.active):
                        self.inbox[a].append(seq)
                        self.pending[a].add(seq)
            else:
                self.register(recv)
                self.inbox[recv].append(seq)
                self.pending[recv].add(seq)
            out.append(seq)
        return out

    def visible(self, agent, seq):
        m = self.msgs[seq]
        if m["status"] == ST_DELETED:
            return False
        if m["vis_mode"] in (VIS_BITMAP, VIS_GROUP) and m["bitmap"] != NO_BITMAP:
            return agent in self.bitmaps[int(m["bitmap"])]
        return True

    def receive(self, agent, k, priority):
        self.register(agent)
        cand = sorted(s for s in self.pending[agent] if self.visible(agent, s))
        dropped = [s for s in self.pending[agent] if not self.visible(agent, s)]
        for s in dropped:
            self.pending[agent].discard(s)
        if priority:
            cand.sort(key=lambda s: (-self.msgs[s]["priority"], s))
        out = cand[:k]
        for s in out:
            self.pending[agent].discard(s)
            if self.msgs[s]["status"] == ST_DELIVERED:
                self.msgs[s]["status"] = ST_READ
        return out


@pytest.mark.parametrize("seed", [0, 1, 2, 7])
def test_cpu_engine_vs_oracle(seed):
    rng = np.random.default_rng(seed)
    n_agents = 24
    cfg = QueueConfig(use_gpu=False, max_agents=64, auto_save=False)
    eng = CpuEngine(cfg)
    orc = Oracle(cfg.max_agents)
    for a in range(n_agents):
        eng.register_agent(a)
        orc.register(a)

    for round_i in range(12):
        n = int(rng.integers(5, 60))
        recs = np.zeros(n, dtype=REC_DTYPE)
        recs["sender"] = rng.integers(0, n_agents, n)
        recv = rng.integers(0, n_agents, n).astype(np.uint32)
        bmask = rng.random(n) < 0.15
        recv[bmask] = BROADCAST
        recs["receiver"] = recv
        recs["type"] = rng.integers(0, 7, n)
        recs["priority"] = rng.integers(0, 4, n)
        recs["vis_mode"] = VIS_ALL
        recs["bitmap"] = NO_BITMAP
        plen = 32
        recs["payload_len"] = plen
        recs["content_len"] = plen
        recs["payload_off"] = np.arange(n, dtype=np.uint64) * plen
        # some broadcasts restricted / group-routed
        for j in np.flatnonzero(bmask):
            mode = rng.integers(0, 3)
            if mode == 0:
                continue
            bits = np.zeros(cfg.max_agents, dtype=bool)
            bits[rng.integers(0, n_agents, 8)] = True
            be = eng.alloc_bitmap(bits)
            bo = orc.alloc_bitmap(bits)
            assert be == bo
            recs["vis_mode"][j] = VIS_BITMAP if mode == 1 else VIS_GROUP
            recs["bitmap"][j] = be
        payload = bytes(n * plen)

        se = eng.enqueue_batch(recs, payload)
        so = orc.enqueue(recs)
        assert se.tolist() == so

        # random polls
        polls = rng.permutation(n_agents)[: int(rng.integers(4, n_agents))]
        k = int(rng.integers(1, 25))
        prio = bool(rng.integers(0, 2))
        for a in polls:
            ge = eng.receive(int(a), k, prio).tolist()
            go = orc.receive(int(a), k, prio)
            assert ge == go, (round_i, a, k, prio, ge, go)

        # occasional deletes
        if round_i % 3 == 2 and eng.total_messages() > 4:
            s = int(rng.integers(0, eng.total_messages()))
            eng.delete(s)
            orc.msgs[s]["status"] = ST_DELETED

    # final drain: everything matches
    for a in range(n_agents):
        assert eng.receive(a, 10_000, False).tolist() == orc.receive(
            a, 10_000, False
        )
    # status agreement
    for s in range(eng.total_messages()):
        assert eng.get_status(s) == orc.msgs[s]["status"], s


def test_checkpoint_roundtrip_fuzz(tmp_path):
    """Randomized checkpoint roundtrip: a random mixed workload
    (p2p/broadcast/restricted/groups/statuses/deletes) saved and
    replayed into a fresh facade must reproduce every live message's
    content, status and visibility-filtered delivery sets."""
    from swarmdb_amd import QueueConfig, SwarmsDB

    rng = np.random.default_rng(99)
    cfg = QueueConfig(use_gpu=False, save_dir=str(tmp_path),
                      auto_save=False, max_agents=64, slot_bytes=512)
    db = SwarmsDB(config=cfg)
    agents = [f"a{i}" for i in range(10)]
    for a in agents:
        db.register_agent(a)
    ids = []
    for i in range(120):
        kind = rng.integers(0, 4)
        sender = agents[int(rng.integers(0, len(agents)))]
        content = f"m{i}-" + "x" * int(rng.integers(0, 200))
        if kind == 0:
            ids.append(db.send_message(
                sender, content,
                receiver_id=agents[int(rng.integers(0, len(agents)))]))
        elif kind == 1:
            ids.append(db.broadcast_message(sender, content))
        elif kind == 2:
            vis = list(rng.choice(agents, 3, replace=False))
            ids.append(db.send_message(sender, content, receiver_id=None,
                                       visible_to=vis))
        else:
            prio = int(rng.integers(0, 4))
            ids.append(db.send_message(
                sender, content, priority=prio,
                receiver_id=agents[int(rng.integers(0, len(agents)))]))
    for mid in rng.choice(ids, 10, replace=False):
        db.mark_message_as_processed(str(mid))
    for mid in rng.choice(ids, 5, replace=False):
        db.delete_message(str(mid))
    base = db.save_checkpoint()
    for i in range(30):
        db.send_message(agents[0], f"d{i}", receiver_id=agents[1])
    db.save_checkpoint_delta()

    db2 = SwarmsDB(config=QueueConfig(
        use_gpu=False, save_dir=str(tmp_path), auto_save=False,
        max_agents=64, slot_bytes=512))
    db2.load_checkpoint(base)
    # delivery parity: each agent's receive set matches the original
    for a in agents:
        got1 = sorted(m.content for m in db.receive_messages(a, 1000,
                                                             timeout=0))
        got2 = sorted(m.content for m in db2.receive_messages(a, 1000,
                                                              timeout=0))
        assert got1 == got2, (a, len(got1), len(got2))
    s1, s2 = db.get_stats(), db2.get_stats()
    # the original's running counters include the 5 tombstoned
    # messages (counted at enqueue); the checkpoint persists only live
    # records, so the replayed counter is exactly those 5 lower
    assert s2["messages_by_type"]["chat"] == (
        s1["messages_by_type"]["chat"] - 5
    )
    db.close()
    db2.close()
