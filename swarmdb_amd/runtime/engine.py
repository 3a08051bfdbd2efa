"""Delivery-engine interface + shared record layout.

The engine is the MI355X-native replacement for the reference's Kafka tier
(librdkafka producer/consumer + broker, reference swarmdb/ main.py:192-207,
334-345, 476-484, 553-571). It owns:

- the message slot store (payload bytes + binary routing header),
- per-agent inbox rings with read cursors,
- delivery status words,
- visibility bitmaps for restricted broadcasts,
- per-agent / per-backend load counters.

Two engines implement it: :class:`~swarmdb_amd.runtime.cpu_engine.CpuEngine`
(numpy, runs anywhere — the permanent test double, BASELINE config 1) and
:class:`~swarmdb_amd.runtime.gpu_engine.GpuEngine` (HBM-resident rings with
HIP kernels via the ``_swarmq`` extension).

The batch record layout (``REC_DTYPE``) is the host-side staging format:
callers build arrays of records + one contiguous payload buffer, and the
engine enqueues the whole batch in one shot (one pinned-memcpy + one kernel
on GPU). This replaces the reference's per-message produce path
(swarmdb/ main.py:466-484) with a batched one.
"""

from __future__ import annotations

import abc
from typing import Dict, List, Optional, Tuple

import numpy as np

# ---- wire constants (must match csrc/swarmq_common.h) ----

BROADCAST = 0xFFFFFFFF  # receiver index meaning "broadcast"
NO_BITMAP = 0xFFFFFFFF  # bitmap index meaning "no visibility restriction"

# status codes (device status word). Order matters: the reference lifecycle
# pending -> delivered -> read -> processed, + failed (swarmdb/ main.py:44-51)
ST_PENDING = 0
ST_DELIVERED = 1
ST_READ = 2
ST_PROCESSED = 3
ST_FAILED = 4
ST_DELETED = 5  # tombstone (delete_message / flush_old_messages)

STATUS_NAMES = ["pending", "delivered", "read", "processed", "failed", "deleted"]

# message type codes (index into TYPE_NAMES = MessageType order)
TYPE_NAMES = [
    "chat",
    "command",
    "function_call",
    "function_result",
    "system",
    "error",
    "status",
]
TYPE_CODES = {n: i for i, n in enumerate(TYPE_NAMES)}

# visibility modes
VIS_ALL = 0      # no restriction (visible_to empty)
VIS_BITMAP = 1   # restricted: agent must be set in the bitmap (inbox entry
#                  still lands in every inbox, filtered at dequeue — the
#                  reference's broadcast-listing behavior)
VIS_GROUP = 2    # group fan-out: inbox entries land ONLY in member inboxes
#                  (one slot per group message; the fan-out kernel filters)

# record flags
FLAG_HAS_EXTRAS = 1   # payload tail carries an extras-JSON blob
FLAG_JSON_CONTENT = 2  # content bytes are JSON (dict/list), else raw utf-8 str
FLAG_DERIVED_ID = 4   # message id is derived from (rank, seq), no extras id
FLAG_OVERFLOW = 8     # payload exceeds the slot capacity and lives in the
#                       facade's host-side overflow store (device slot empty;
#                       not visible to the device search kernel)

# per-message batch record (host staging). 48 bytes, 8-byte aligned.
# Payload layout: [content bytes (content_len)][extras JSON (rest)] —
# search scans only the content window (reference searches the content
# field only, swarmdb/ main.py:742-781).
REC_DTYPE = np.dtype(
    {
        "names": [
            "sender", "receiver", "type", "priority", "vis_mode", "flags",
            "token_count", "timestamp", "payload_off", "payload_len", "bitmap",
            "content_len", "bitmap_epoch",
        ],
        "formats": [
            np.uint32, np.uint32, np.uint8, np.uint8, np.uint8, np.uint8,
            np.uint32, np.float64, np.uint64, np.uint32, np.uint32,
            np.uint32, np.uint32,
        ],
        "offsets": [0, 4, 8, 9, 10, 11, 12, 16, 24, 32, 36, 40, 44],
        "itemsize": 48,
    }
)


class Engine(abc.ABC):
    """Abstract delivery engine. All indices are dense agent indices
    assigned by :meth:`register_agent`; the facade maps agent-id strings
    to indices."""

    # --- registry ---

    @abc.abstractmethod
    def register_agent(self, agent_idx: int) -> None:
        """Activate an agent slot (idempotent)."""

    @abc.abstractmethod
    def deregister_agent(self, agent_idx: int) -> None:
        """Deactivate an agent slot. Inbox/messages survive (reference
        behavior, swarmdb/ main.py:351-372 — SURVEY.md §8.15)."""

    @abc.abstractmethod
    def active_agents(self) -> np.ndarray:
        """Bool array [max_agents] of active flags."""

    # --- send plane ---

    @abc.abstractmethod
    def enqueue_batch(self, recs: np.ndarray, payloads: bytes) -> np.ndarray:
        """Enqueue a batch. ``recs`` is a REC_DTYPE array whose
        payload_off/len index into ``payloads``. Returns the assigned
        sequence numbers (uint64 array). Status of each message is
        DELIVERED once this returns (the enqueue-ack — replaces the Kafka
        delivery callback, swarmdb/ main.py:374-391)."""

    @abc.abstractmethod
    def alloc_bitmap(self, bits: np.ndarray) -> int:
        """Store a visibility bitmap (bool array [max_agents]); returns an
        allocation HANDLE for REC_DTYPE.bitmap. On the GPU engine the
        pool is an epoch-tagged ring: the engine splits the handle into
        (pool slot, epoch) at enqueue time, and dequeue hides any message
        whose pool slot was recycled (exact visibility — never filtered
        against the wrong bitmap). The CPU engine's pool never recycles,
        so handle == index there."""

    # --- receive plane ---

    @abc.abstractmethod
    def receive(
        self, agent_idx: int, max_messages: int, priority_order: bool = False
    ) -> np.ndarray:
        """Drain up to max_messages deliverable entries from the agent's
        inbox cursor. Applies the visibility filter (reference
        swarmdb/ main.py:579-585) engine-side — per-agent cursors replace
        the reference's every-consumer-scans-everything model (SURVEY.md
        §8.7). Marks returned messages READ. Returns uint64 seq array."""

    def receive_many(
        self,
        agent_idxs: np.ndarray,
        max_per_agent: int,
        priority_order: bool = False,
    ) -> Tuple[np.ndarray, np.ndarray]:
        """Drain many agents at once. Returns (counts[len(agent_idxs)],
        concatenated seqs). Default: loop over :meth:`receive`; the GPU
        engine overrides this with a single dequeue kernel (one launch for
        the whole poll tick)."""
        counts = np.zeros(len(agent_idxs), dtype=np.int64)
        chunks = []
        for i, a in enumerate(agent_idxs):
            s = self.receive(int(a), max_per_agent, priority_order)
            counts[i] = len(s)
            if len(s):
                chunks.append(s)
        seqs = (
            np.concatenate(chunks) if chunks else np.empty(0, dtype=np.uint64)
        )
        return counts, seqs

    @abc.abstractmethod
    def peek_inbox(self, agent_idx: int) -> np.ndarray:
        """All seqs ever appended to the agent's inbox (oldest first),
        tombstones excluded — backs get_agent_messages."""

    @abc.abstractmethod
    def unread_count(self, agent_idx: int) -> int:
        """Inbox entries in DELIVERED state (reference
        swarmdb/ main.py:1026-1047)."""

    # --- message store ---

    @abc.abstractmethod
    def fetch(self, seqs: np.ndarray) -> Tuple[np.ndarray, List[bytes]]:
        """Return (headers REC-like structured array incl. status + seq,
        payload bytes list) for the given seqs."""

    def fetch_raw_chunks(
        self, seqs: np.ndarray
    ) -> Tuple[np.ndarray, np.ndarray, int]:
        """Bulk fetch WITHOUT per-message Python objects: returns
        (headers structured array incl. status+seq, flat uint8 payload
        array laid out [n, stride], stride). The binary checkpoint path
        uses this to move the log at device-gather speed."""
        hdrs, pays = self.fetch(seqs)
        stride = int(self.cfg.slot_bytes)  # type: ignore[attr-defined]
        flat = np.zeros((len(seqs), stride), dtype=np.uint8)
        for i, p in enumerate(pays):
            flat[i, : len(p)] = np.frombuffer(p, dtype=np.uint8)
        return hdrs, flat.reshape(-1), stride

    def read_bitmap(self, slot: int, epoch: int) -> Optional[np.ndarray]:
        """Bit array of a visibility-bitmap pool slot if the epoch still
        matches (i.e. the referencing record's visibility set is still
        live); None when the slot was recycled."""
        return None

    @abc.abstractmethod
    def set_status(self, seq: int, status: int) -> None: ...

    def set_statuses(self, seqs: np.ndarray, statuses: np.ndarray) -> None:
        """Batched status restore (one kernel on GPU)."""
        for s, st in zip(seqs, statuses):
            self.set_status(int(s), int(st))

    @abc.abstractmethod
    def get_status(self, seq: int) -> int: ...

    def statuses(self, seqs: np.ndarray) -> np.ndarray:
        """Batched status lookup (one kernel + one D2H on GPU)."""
        return np.fromiter(
            (self.get_status(int(s)) for s in seqs), dtype=np.uint8,
            count=len(seqs),
        )

    @abc.abstractmethod
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
        """Filtered scan over all messages, newest-first (reference
        swarmdb/ main.py:671-740). Returns seq array."""

    @abc.abstractmethod
    def search(self, needle: bytes, case_sensitive: bool, limit: int) -> np.ndarray:
        """Substring scan over payloads, newest-first (reference
        swarmdb/ main.py:742-781). Returns seq array."""

    @abc.abstractmethod
    def delete(self, seq: int) -> bool:
        """Tombstone a message (reference swarmdb/ main.py:1132-1157)."""

    # --- counters / stats ---

    @abc.abstractmethod
    def total_messages(self) -> int: ...

    def evict_base(self) -> int:
        """Lowest seq still retained (slot-ring retention horizon). 0 on
        engines that never evict (the CPU log is append-only)."""
        return 0

    @abc.abstractmethod
    def stats_arrays(self) -> Dict[str, np.ndarray]:
        """Running counters: by_type [7], by_status [6], sent [max_agents],
        received [max_agents] (replaces the reference's O(N) scans,
        swarmdb/ main.py:973-1024 — SURVEY.md §5.5)."""

    @abc.abstractmethod
    def recv_rate_window(self, agent_idx: int, window_s: float) -> int:
        """Messages received by the agent in the last window (backs
        get_agent_load's processing_rate, swarmdb/ main.py:1069-1093)."""

    # --- load balancer ---

    @abc.abstractmethod
    def backend_add_load(self, backend_idx: int, delta: int) -> None: ...

    @abc.abstractmethod
    def backend_loads(self) -> np.ndarray: ...

    @abc.abstractmethod
    def least_loaded_backend(self, n_backends: int) -> int:
        """argmin over per-backend load counters (CDNA4 reduction kernel on
        GPU — BASELINE config 5; the mechanism the reference lacks,
        SURVEY.md §2.2 'LLM load balancing')."""

    def dispatch_batch(self, requests: int, n_backends: int) -> np.ndarray:
        """Exact sequential least-loaded dispatch of `requests` requests:
        each pick increments the chosen backend's in-flight count. One
        wavefront-shuffle reduction kernel on GPU; returns the chosen
        backend index per request."""
        out = np.empty(requests, dtype=np.uint32)
        for i in range(requests):
            b = self.least_loaded_backend(n_backends)
            self.backend_add_load(b, 1)
            out[i] = b
        return out

    # --- lifecycle ---

    def close(self) -> None:  # noqa: B027
        pass
