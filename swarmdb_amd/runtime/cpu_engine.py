"""CPU delivery engine — numpy struct-of-arrays, thread-safe.

The in-process queue backend (BASELINE config 1) and the permanent no-GPU
test double. Implements the same slot/inbox/cursor model
as the GPU engine so parity tests can diff the two directly; the reference
analog is the Kafka produce/consume tier it replaces (swarmdb/
main.py:192-207, 334-345, 466-484, 553-588).

Unlike the reference — where every agent's consumer scans the whole topic
and filters client-side, O(agents x messages) aggregate —
delivery here is per-agent inbox rings with read cursors: each message is
touched O(recipients) times.
"""

from __future__ import annotations

import threading
import time
from typing import Dict, List, Optional, Tuple

import numpy as np

from ..core.config import QueueConfig
from .engine import (
    BROADCAST,
    NO_BITMAP,
    REC_DTYPE,
    ST_DELETED,
    ST_DELIVERED,
    ST_FAILED,
    ST_PENDING,
    ST_READ,
    VIS_BITMAP,
    VIS_GROUP,
    Engine,
)

_GROW = 2

# fetch output layout: REC fields + status + seq (precomputed — building
# it per call costs a dtype introspection per fetch)
_FETCH_DTYPE = np.dtype(
    REC_DTYPE.descr + [("status", np.uint8), ("seq", np.uint64)]
)


class _U64Ring:
    """Growable append-only u64 vector (inbox log)."""

    __slots__ = ("buf", "n")

    def __init__(self, cap: int = 64):
        self.buf = np.empty(cap, dtype=np.uint64)
        self.n = 0

    def append_many(self, vals: np.ndarray) -> None:
        need = self.n + len(vals)
        if need > len(self.buf):
            cap = len(self.buf)
            while cap < need:
                cap *= _GROW
            nb = np.empty(cap, dtype=np.uint64)
            nb[: self.n] = self.buf[: self.n]
            self.buf = nb
        self.buf[self.n : need] = vals
        self.n = need

    def view(self) -> np.ndarray:
        return self.buf[: self.n]


class CpuEngine(Engine):
    def __init__(self, config: Optional[QueueConfig] = None):
        self.cfg = config or QueueConfig()
        c = self.cfg
        self._lock = threading.RLock()

        cap = 4096
        self._cap = cap
        self._count = 0  # seq == dense index (no ring reuse on CPU)

        self._hdr = np.zeros(cap, dtype=REC_DTYPE)
        self._status = np.full(cap, ST_PENDING, dtype=np.uint8)
        self._read_ts = np.zeros(cap, dtype=np.float64)
        self._heap = bytearray()
        self._pay_off = np.zeros(cap, dtype=np.uint64)
        self._pay_len = np.zeros(cap, dtype=np.uint32)

        self._active = np.zeros(c.max_agents, dtype=bool)
        self._inbox: Dict[int, _U64Ring] = {}
        self._cursor = np.zeros(c.max_agents, dtype=np.int64)
        # visible-but-unconsumed entries from partial drains (the inbox
        # ring itself is append-only history; the GPU engine keeps the
        # same per-agent carry buffer device-side)
        self._carry: Dict[int, np.ndarray] = {}
        self._window = c.recv_window  # max entries examined per receive

        self._bitmaps: List[np.ndarray] = []

        self._by_type = np.zeros(7, dtype=np.int64)
        self._by_status = np.zeros(6, dtype=np.int64)
        self._sent = np.zeros(c.max_agents, dtype=np.int64)
        self._received = np.zeros(c.max_agents, dtype=np.int64)
        self._recv_ts: Dict[int, _U64Ring] = {}  # read timestamps (ms) per agent

        self._backend_load = np.zeros(c.num_backends, dtype=np.int64)

    # --- capacity ---

    def _ensure(self, n: int) -> None:
        need = self._count + n
        if need <= self._cap:
            return
        cap = self._cap
        while cap < need:
            cap *= _GROW
        for name in ("_hdr", "_status", "_read_ts", "_pay_off", "_pay_len"):
            old = getattr(self, name)
            nb = np.zeros(cap, dtype=old.dtype)
            nb[: self._count] = old[: self._count]
            setattr(self, name, nb)
        self._cap = cap

    # --- registry ---

    def register_agent(self, agent_idx: int) -> None:
        with self._lock:
            self._active[agent_idx] = True
            if agent_idx not in self._inbox:
                self._inbox[agent_idx] = _U64Ring()
                self._recv_ts[agent_idx] = _U64Ring()

    def deregister_agent(self, agent_idx: int) -> None:
        with self._lock:
            self._active[agent_idx] = False

    def active_agents(self) -> np.ndarray:
        return self._active.copy()

    # --- send plane ---

    def enqueue_batch(self, recs: np.ndarray, payloads: bytes) -> np.ndarray:
        n = len(recs)
        if n == 0:
            return np.empty(0, dtype=np.uint64)
        with self._lock:
            self._ensure(n)
            base = self._count
            sl = slice(base, base + n)
            # error lane (matches k_enqueue's validation — reference
            # _errors topic analog, swarmdb/ main.py:260-273, 501-519):
            # malformed records park as FAILED, no delivery
            bad = (
                (recs["type"] >= 7)
                | (recs["priority"] > 3)
                | (recs["payload_len"] > self.cfg.slot_bytes)
                | (
                    (recs["receiver"] != BROADCAST)
                    & (recs["receiver"] >= self.cfg.max_agents)
                )
                | (recs["sender"] >= self.cfg.max_agents)
            )
            good = ~bad
            self._hdr[sl] = recs
            heap_base = len(self._heap)
            self._heap += payloads
            self._pay_off[sl] = recs["payload_off"] + np.uint64(heap_base)
            self._pay_len[sl] = np.where(good, recs["payload_len"], 0)
            self._status[sl] = np.where(good, ST_DELIVERED, ST_FAILED)
            self._count = base + n
            seqs = np.arange(base, base + n, dtype=np.uint64)

            # counters
            np.add.at(self._by_type, recs["type"][good], 1)
            self._by_status[ST_DELIVERED] += int(good.sum())
            self._by_status[ST_FAILED] += int(bad.sum())
            np.add.at(self._sent, recs["sender"][good], 1)

            # inbox fan-out (valid records only)
            recv = recs["receiver"]
            bmask = (recv == BROADCAST) & good
            if bmask.any():
                bseqs = seqs[bmask]
                gmask = recs["vis_mode"][bmask] == VIS_GROUP
                plain = bseqs[~gmask]
                if len(plain):
                    for a in np.flatnonzero(self._active):
                        self._inbox[int(a)].append_many(plain)
                # group fan-out: member inboxes only
                for s, bm in zip(bseqs[gmask], recs["bitmap"][bmask][gmask]):
                    members = np.flatnonzero(
                        self._bitmaps[int(bm)] & self._active
                    )
                    one = np.array([s], dtype=np.uint64)
                    for a in members:
                        ai = int(a)
                        if ai not in self._inbox:
                            self._inbox[ai] = _U64Ring()
                            self._recv_ts[ai] = _U64Ring()
                        self._inbox[ai].append_many(one)
            pmask = (recv != BROADCAST) & good
            if pmask.any():
                prs = recv[pmask]
                pseqs = seqs[pmask]
                order = np.argsort(prs, kind="stable")
                prs_s, pseqs_s = prs[order], pseqs[order]
                bounds = np.flatnonzero(np.diff(prs_s)) + 1
                starts = np.concatenate(([0], bounds))
                ends = np.concatenate((bounds, [len(prs_s)]))
                for s, e in zip(starts, ends):
                    a = int(prs_s[s])
                    if a not in self._inbox:  # auto-registered by facade normally
                        self._inbox[a] = _U64Ring()
                        self._recv_ts[a] = _U64Ring()
                    self._inbox[a].append_many(pseqs_s[s:e])
            return seqs

    def alloc_bitmap(self, bits: np.ndarray) -> int:
        with self._lock:
            self._bitmaps.append(bits.astype(bool).copy())
            return len(self._bitmaps) - 1

    # --- receive plane ---

    def _visible_mask(self, agent_idx: int, seqs: np.ndarray) -> np.ndarray:
        """Vectorized delivery filter (reference swarmdb/ main.py:579-585):
        deliver iff (receiver == agent or broadcast) and (agent in
        visible_to or visible_to empty). Inbox membership already implies
        the receiver/broadcast condition."""
        st = self._status[seqs]
        ok = st != ST_DELETED
        vm = self._hdr["vis_mode"][seqs]
        restricted = (vm == VIS_BITMAP) | (vm == VIS_GROUP)
        if restricted.any():
            bidx = self._hdr["bitmap"][seqs[restricted]]
            allowed = np.fromiter(
                (self._bitmaps[int(b)][agent_idx] if b != NO_BITMAP else True
                 for b in bidx),
                dtype=bool,
                count=int(restricted.sum()),
            )
            sub = ok[restricted]
            ok[np.flatnonzero(restricted)] = sub & allowed
        return ok

    def receive(
        self, agent_idx: int, max_messages: int, priority_order: bool = False
    ) -> np.ndarray:
        with self._lock:
            ring = self._inbox.get(agent_idx)
            if ring is None:
                return np.empty(0, dtype=np.uint64)
            cur = int(self._cursor[agent_idx])
            carry = self._carry.get(agent_idx)
            ncarry = 0 if carry is None else len(carry)
            # examine at most `window` entries per call (the GPU dequeue
            # kernel's LDS window); the cursor advances past what was
            # examined, leftovers go to the carry buffer
            take_new = min(ring.n - cur, self._window - ncarry)
            fresh = ring.view()[cur : cur + take_new]
            self._cursor[agent_idx] = cur + take_new
            if ncarry:
                pend = np.sort(np.concatenate((carry, fresh)))
            else:
                # sort into seq order regardless of append order: batched
                # enqueue appends per-receiver groups (and the GPU engine
                # appends with atomics) — the dequeue kernel sorts in LDS
                pend = np.sort(fresh)
            if len(pend) == 0:
                self._carry.pop(agent_idx, None)
                return pend.copy()
            ok = self._visible_mask(agent_idx, pend)
            cand = pend[ok]
            if priority_order and len(cand) > 1:
                pri = self._hdr["priority"][cand].astype(np.int64)
                # stable sort by priority desc (seq order preserved within)
                cand = cand[np.argsort(-pri, kind="stable")]
            out = cand[:max_messages]
            rest = cand[max_messages:]
            if len(rest):
                self._carry[agent_idx] = np.sort(rest) if priority_order else rest.copy()
            else:
                self._carry.pop(agent_idx, None)
            if len(out):
                prev = self._status[out]
                was_delivered = prev == ST_DELIVERED
                np.add.at(self._by_status, prev[was_delivered], -1)
                self._by_status[ST_READ] += int(was_delivered.sum())
                self._status[out[was_delivered]] = ST_READ
                self._read_ts[out] = time.time()
                self._received[agent_idx] += len(out)
                now_ms = np.uint64(time.time() * 1000)
                self._recv_ts[agent_idx].append_many(
                    np.full(len(out), now_ms, dtype=np.uint64)
                )
            return out.astype(np.uint64)

    def peek_inbox(self, agent_idx: int) -> np.ndarray:
        with self._lock:
            ring = self._inbox.get(agent_idx)
            if ring is None:
                return np.empty(0, dtype=np.uint64)
            seqs = np.sort(ring.view())  # log (seq) order
            alive = self._status[seqs] != ST_DELETED
            return seqs[alive].copy()

    def unread_count(self, agent_idx: int) -> int:
        with self._lock:
            ring = self._inbox.get(agent_idx)
            if ring is None:
                return 0
            seqs = ring.view()
            return int((self._status[seqs] == ST_DELIVERED).sum())

    # --- message store ---

    def fetch(self, seqs: np.ndarray) -> Tuple[np.ndarray, List[bytes]]:
        with self._lock:
            hdr = self._hdr[seqs].copy()
            status = self._status[seqs].copy()
            mv = memoryview(self._heap)
            pays = [
                bytes(mv[int(o) : int(o) + int(l)])
                for o, l in zip(self._pay_off[seqs], self._pay_len[seqs])
            ]
            out = np.zeros(len(seqs), dtype=_FETCH_DTYPE)
            for name in REC_DTYPE.names:
                out[name] = hdr[name]
            out["status"] = status
            out["seq"] = seqs
            return out, pays

    def fetch_raw_chunks(self, seqs: np.ndarray):
        """Vectorized ragged gather out of the heap (no per-message
        bytes objects) — mirrors the GPU engine's checkpoint path."""
        with self._lock:
            seqs = np.asarray(seqs, dtype=np.int64)
            n = len(seqs)
            stride = int(self.cfg.slot_bytes)
            out = np.zeros(n, dtype=_FETCH_DTYPE)
            h = self._hdr[seqs]
            for name in REC_DTYPE.names:
                out[name] = h[name]
            out["status"] = self._status[seqs]
            out["seq"] = seqs.astype(np.uint64)
            flat = np.zeros((n, stride), dtype=np.uint8)
            heap = np.frombuffer(self._heap, dtype=np.uint8)
            offs = self._pay_off[seqs].astype(np.int64)
            lens = self._pay_len[seqs].astype(np.int64)
            if n:
                L = int(lens[0])
                d = int(offs[1] - offs[0]) if n > 1 else L
                uniform = (
                    (lens == L).all()
                    and d >= L
                    and (n < 2 or (np.diff(offs) == d).all())
                    and int(offs[0]) + (n - 1) * d + L <= len(heap)
                )
                if uniform and L:
                    # constant-stride rows: one strided view, no index
                    # arrays (the giant fancy-index version cost 18 us
                    # per row; this is a memcpy)
                    view = np.lib.stride_tricks.as_strided(
                        heap[int(offs[0]):], shape=(n, L), strides=(d, 1)
                    )
                    flat[:, :L] = view
                else:
                    for i in range(n):
                        o, l = int(offs[i]), int(lens[i])
                        if l:
                            flat[i, :l] = heap[o : o + l]
            return out, flat.reshape(-1), stride

    def read_bitmap(self, slot: int, epoch: int):
        # the CPU pool never recycles: handle == slot, epoch vacuous
        with self._lock:
            if 0 <= slot < len(self._bitmaps):
                return self._bitmaps[slot].copy()
            return None

    def set_status(self, seq: int, status: int) -> None:
        with self._lock:
            prev = int(self._status[seq])
            if prev != status:
                self._by_status[prev] -= 1
                self._by_status[status] += 1
                self._status[seq] = status

    def get_status(self, seq: int) -> int:
        return int(self._status[seq])

    def statuses(self, seqs: np.ndarray) -> np.ndarray:
        with self._lock:
            return self._status[np.asarray(seqs, dtype=np.int64)].copy()

    def set_statuses(self, seqs: np.ndarray, statuses: np.ndarray) -> None:
        with self._lock:
            idx = np.asarray(seqs, dtype=np.int64)
            old = self._status[idx]
            np.add.at(self._by_status, old, -1)
            np.add.at(self._by_status, statuses, 1)
            self._status[idx] = statuses

    def query(
        self,
        sender: Optional[int] = None,
        receiver: Optional[int] = None,
        type_code: Optional[int] = None,
        status: Optional[int] = None,
        after: Optional[float] = None,
        before: Optional[float] = None,
        limit: int = 100,
    ) -> np.ndarray:
        with self._lock:
            n = self._count
            mask = self._status[:n] != ST_DELETED
            h = self._hdr[:n]
            if sender is not None:
                mask &= h["sender"] == sender
            if receiver is not None:
                mask &= h["receiver"] == receiver
            if type_code is not None:
                mask &= h["type"] == type_code
            if status is not None:
                mask &= self._status[:n] == status
            if after is not None:
                mask &= h["timestamp"] > after  # exclusive (main.py:725-729)
            if before is not None:
                mask &= h["timestamp"] < before  # exclusive (main.py:731-735)
            idx = np.flatnonzero(mask)
            return idx[::-1][:limit].astype(np.uint64)  # newest-first

    def search(self, needle: bytes, case_sensitive: bool, limit: int) -> np.ndarray:
        with self._lock:
            heap = bytes(self._heap)
            if not case_sensitive:
                heap = heap.lower()
                needle = needle.lower()
            n = self._count
            offs = self._pay_off[:n].astype(np.int64)
            # search only the content window of each payload (reference
            # scans content only, swarmdb/ main.py:742-781)
            ends = offs + self._hdr["content_len"][:n]
            hits = np.zeros(n, dtype=bool)
            pos = heap.find(needle)
            while pos != -1:
                i = int(np.searchsorted(offs, pos, side="right")) - 1
                if 0 <= i < n and pos + len(needle) <= ends[i]:
                    hits[i] = True
                pos = heap.find(needle, pos + 1)
            hits &= self._status[:n] != ST_DELETED
            idx = np.flatnonzero(hits)
            return idx[::-1][:limit].astype(np.uint64)

    def delete(self, seq: int) -> bool:
        with self._lock:
            if seq >= self._count or self._status[seq] == ST_DELETED:
                return False
            self.set_status(int(seq), ST_DELETED)
            return True

    # --- counters / stats ---

    def total_messages(self) -> int:
        return self._count

    def stats_arrays(self) -> Dict[str, np.ndarray]:
        with self._lock:
            return {
                "by_type": self._by_type.copy(),
                "by_status": self._by_status.copy(),
                "sent": self._sent.copy(),
                "received": self._received.copy(),
                "dropped": 0,  # CPU inboxes are unbounded
            }

    def recv_rate_window(self, agent_idx: int, window_s: float) -> int:
        with self._lock:
            ring = self._recv_ts.get(agent_idx)
            if ring is None:
                return 0
            cutoff = np.uint64((time.time() - window_s) * 1000)
            return int((ring.view() >= cutoff).sum())

    # --- load balancer ---

    def backend_add_load(self, backend_idx: int, delta: int) -> None:
        with self._lock:
            self._backend_load[backend_idx] += delta

    def backend_loads(self) -> np.ndarray:
        return self._backend_load.copy()

    def least_loaded_backend(self, n_backends: int) -> int:
        with self._lock:
            return int(np.argmin(self._backend_load[:n_backends]))
