"""SwarmsDB — the core runtime facade.

Re-exposes the full public method surface of the reference's ``SwarmsDB``
class (reference ``swarmdb/ main.py:130-1394``, inventoried in SURVEY.md
§2.2) over the MI355X-native delivery engine. The FastAPI layer programs
against this contract exactly as the reference API does.

Architecture (vs the reference):

- The Kafka producer/consumer tier is replaced by an :class:`Engine`
  (CPU numpy double, or GPU HBM rings + HIP kernels).
- Shared state lives in ONE engine instance (device-resident on GPU)
  instead of per-worker copies (fixes SURVEY.md §8.8).
- History saves run on a background spill thread, off the send path
  (fixes SURVEY.md §8.14; cadence semantics kept: every ``save_interval``
  seconds or ``max_messages_per_file`` messages, swarmdb/ main.py:492-497).
- Partition routing uses a stable FNV-1a hash (fixes SURVEY.md §8.6).
- A least-loaded LLM-backend selector actually exists (the reference only
  stores a flag, swarmdb/ main.py:1281-1291).
"""

from __future__ import annotations

import json
import logging
import threading
import time
from concurrent.futures import ThreadPoolExecutor
from datetime import datetime
from pathlib import Path
from typing import Any, Callable, Dict, List, Optional, Union

import numpy as np
import yaml

from ..core.config import QueueConfig
from ..core.message import Message, MessagePriority, MessageStatus, MessageType
from ..core.wire import (
    decode_content,
    decode_extras,
    derived_id,
    encode_content,
    encode_extras,
    parse_derived_id,
)
from ..utils.hashing import partition_for
from ..utils.tracing import tracer
from .engine import (
    BROADCAST,
    FLAG_HAS_EXTRAS,
    FLAG_JSON_CONTENT,
    FLAG_OVERFLOW,
    NO_BITMAP,
    REC_DTYPE,
    ST_DELETED,
    ST_FAILED,
    ST_PROCESSED,
    STATUS_NAMES,
    TYPE_CODES,
    TYPE_NAMES,
    VIS_ALL,
    VIS_BITMAP,
    VIS_GROUP,
    Engine,
)

logger = logging.getLogger("swarmdb_amd")

_STATUS_CODES = {n: i for i, n in enumerate(STATUS_NAMES)}


def _status_code(status: Union[MessageStatus, str]) -> int:
    return _STATUS_CODES[status.value if isinstance(status, MessageStatus) else str(status)]


def _type_code(t: Union[MessageType, str]) -> int:
    return TYPE_CODES[t.value if isinstance(t, MessageType) else str(t)]


class SwarmsDB:
    """Agent message queue + LLM-backend load balancer.

    Method-for-method parity with the reference class;
    citations in each docstring point at the reference implementation the
    behavior mirrors.
    """

    _PRUNE_EVERY = 4096  # compat-map sweep cadence (inserts)

    def __init__(
        self,
        config: Optional[QueueConfig] = None,
        token_counter: Optional[Callable[[str], int]] = None,
        engine: Optional[Engine] = None,
        save_dir: Optional[str] = None,
        auto_save: Optional[bool] = None,
        save_interval: Optional[float] = None,
        max_messages_per_file: Optional[int] = None,
    ):
        """Reference: swarmdb/ main.py:156-237 (minus Kafka wiring)."""
        self.config = config or QueueConfig()
        if save_dir is not None:
            self.config.save_dir = save_dir
        if auto_save is not None:
            self.config.auto_save = auto_save
        if save_interval is not None:
            self.config.save_interval = save_interval
        if max_messages_per_file is not None:
            self.config.max_messages_per_file = max_messages_per_file

        if self.config.log_file:
            # reference loguru sink semantics: 10 MB rotation, INFO
            # (swarmdb/ main.py:170-189), stdlib implementation
            from logging.handlers import RotatingFileHandler

            handler = RotatingFileHandler(
                self.config.log_file, maxBytes=10 * 1024 * 1024, backupCount=5
            )
            handler.setFormatter(logging.Formatter(
                "%(asctime)s | %(levelname)s | %(name)s | %(message)s"
            ))
            logger.addHandler(handler)
            logger.setLevel(logging.INFO)

        self.token_counter = token_counter
        if engine is None:
            engine = self._default_engine(self.config)
        self.engine = engine

        self._lock = threading.RLock()
        self._agent_ids: List[str] = []          # idx -> agent_id
        self._agent_idx: Dict[str, int] = {}     # agent_id -> idx
        self.registered_agents: set = set()
        self.agent_metadata: Dict[str, dict] = {}
        self.metadata: Dict[str, Any] = {}       # groups + llm assignments
        self._id_to_seq: Dict[str, int] = {}     # compat-path id map
        self._failed: Dict[str, Message] = {}    # failed sends for resend
        # host-side store for payloads larger than a device slot:
        # seq -> (content_len, payload bytes)
        # (SURVEY.md §7 'variable-size content in fixed slots')
        self._overflow: Dict[int, tuple] = {}
        self._prune_tick = 0
        self._pruned_below = 0
        self._express = None  # doorbell express lane (express_start)

        self._llm_backends: List[str] = []
        self._llm_backend_idx: Dict[str, int] = {}
        self.llm_load_balancing = False

        self.save_dir = Path(self.config.save_dir)
        if self.config.auto_save:
            self.save_dir.mkdir(parents=True, exist_ok=True)
        self.last_save_time = time.time()
        self._spill = ThreadPoolExecutor(max_workers=1, thread_name_prefix="spill")
        self._spill_pending = False
        self._closed = False
        logger.info(
            "SwarmsDB up: engine=%s partitions=%d max_agents=%d",
            type(self.engine).__name__,
            self.config.num_partitions,
            self.config.max_agents,
        )

    @staticmethod
    def _default_engine(config: QueueConfig) -> Engine:
        use_gpu = config.use_gpu
        if use_gpu is None:
            try:
                import torch

                use_gpu = torch.cuda.is_available()
            except Exception:
                use_gpu = False
        if use_gpu:
            from .gpu_engine import GpuEngine

            return GpuEngine(config)
        from .cpu_engine import CpuEngine

        return CpuEngine(config)

    # ------------------------------------------------------------------
    # agent registry (reference swarmdb/ main.py:314-372)
    # ------------------------------------------------------------------

    def _idx_of(self, agent_id: str, create: bool = False) -> Optional[int]:
        idx = self._agent_idx.get(agent_id)
        if idx is None and create:
            idx = len(self._agent_ids)
            if idx >= self.config.max_agents:
                raise RuntimeError(
                    f"agent capacity exceeded ({self.config.max_agents})"
                )
            self._agent_ids.append(agent_id)
            self._agent_idx[agent_id] = idx
        return idx

    def register_agent(self, agent_id: str) -> bool:
        """Idempotent registration (reference swarmdb/ main.py:314-349).
        Per-agent read cursors replace the reference's per-agent Kafka
        consumer groups."""
        with self._lock:
            if agent_id in self.registered_agents:
                return True
            idx = self._idx_of(agent_id, create=True)
            self.engine.register_agent(idx)
            self.registered_agents.add(agent_id)
            logger.info("registered agent %s (idx %d)", agent_id, idx)
            return True

    def deregister_agent(self, agent_id: str) -> bool:
        """Reference swarmdb/ main.py:351-372. Inbox and messages are NOT
        cleaned up (reference behavior)."""
        with self._lock:
            if agent_id not in self.registered_agents:
                return False
            self.registered_agents.discard(agent_id)
            idx = self._agent_idx[agent_id]
            self.engine.deregister_agent(idx)
            logger.info("deregistered agent %s", agent_id)
            return True

    # ------------------------------------------------------------------
    # send path (reference swarmdb/ main.py:295-519, 810-850, 1229-1279)
    # ------------------------------------------------------------------

    def _count_tokens(self, content: Union[str, dict, list]) -> int:
        """Reference swarmdb/ main.py:295-307."""
        if self.token_counter is None:
            return 0
        if isinstance(content, (dict, list)):
            content = json.dumps(content)
        return int(self.token_counter(content))

    def _get_partition(self, agent_id: str) -> int:
        """Stable hash -> partition (reference swarmdb/ main.py:309-312;
        made deterministic per SURVEY.md §8.6)."""
        return partition_for(agent_id, self.config.num_partitions)

    def _slot_capacity(self) -> int:
        return int(self.config.slot_bytes)

    def _bitmap_for(self, visible_to: List[str]) -> int:
        bits = np.zeros(self.config.max_agents, dtype=bool)
        for a in visible_to:
            i = self._idx_of(a)
            if i is not None:
                bits[i] = True
        return self.engine.alloc_bitmap(bits)

    def send_message(
        self,
        sender_id: str,
        content: Union[str, Dict[str, Any], List[Any]],
        receiver_id: Optional[str] = None,
        message_type: Union[MessageType, str] = MessageType.CHAT,
        priority: Union[MessagePriority, int] = MessagePriority.NORMAL,
        metadata: Optional[Dict[str, Any]] = None,
        visible_to: Optional[List[str]] = None,
    ) -> str:
        """The per-message send path (reference swarmdb/ main.py:393-519).

        Auto-registers sender/receiver; broadcast with empty visible_to
        makes the message visible to all registered agents (main.py:449-450).
        The Kafka produce+delivery-callback becomes a single engine enqueue
        whose completion IS the DELIVERED ack.
        """
        with self._lock:
            self.register_agent(sender_id)
            if receiver_id is not None:
                self.register_agent(receiver_id)

            mtype = MessageType(message_type)
            prio = (
                priority
                if isinstance(priority, MessagePriority)
                else MessagePriority(priority)
            )
            token_count = self._count_tokens(content)
            vis = list(visible_to) if visible_to else []
            if receiver_id is None and not vis:
                vis = sorted(self.registered_agents)

            msg = Message(
                sender_id=sender_id,
                receiver_id=receiver_id,
                content=content,
                type=mtype,
                priority=prio,
                metadata=metadata or {},
                token_count=token_count,
                visible_to=vis,
            )

            rec, payload, overflow = self._encode_message(msg)

            try:
                seqs = self.engine.enqueue_batch(rec, payload)
            except Exception as e:
                # failure lane (reference error-topic republish,
                # swarmdb/ main.py:501-519): keep the message for resend
                msg.status = MessageStatus.FAILED
                msg.metadata["error"] = str(e)
                self._failed[msg.id] = msg
                logger.error("send failed for %s: %s", msg.id, e)
                raise
            self._register_enqueued(msg, int(seqs[0]), overflow)
            self._maybe_autosave()
            return msg.id

    def _encode_message(self, msg: Message, bitmap_idx: Optional[int] = None):
        """Build the engine record for a validated Message. Returns
        (rec[1], payload bytes, overflow tuple or None). Caller holds
        the lock; sender/receiver must be registered. ``bitmap_idx``
        reuses a pre-allocated visibility bitmap (group cache)."""
        rec = np.zeros(1, dtype=REC_DTYPE)
        content_b, is_json = encode_content(msg.content)
        extras_b = encode_extras(msg.id, msg.metadata, msg.visible_to)
        payload = content_b + extras_b
        flags = FLAG_HAS_EXTRAS
        if is_json:
            flags |= FLAG_JSON_CONTENT
        overflow = None
        if len(payload) > self._slot_capacity():
            # oversized content: the device slot stays empty and the
            # payload lives host-side; routing/delivery is unchanged
            overflow = (len(content_b), payload)
            payload = b""
            flags |= FLAG_OVERFLOW
        else:
            rec["payload_len"] = len(payload)
            rec["content_len"] = len(content_b)
        rec["sender"] = self._agent_idx[msg.sender_id]
        rec["receiver"] = (
            BROADCAST
            if msg.receiver_id is None
            else self._agent_idx[msg.receiver_id]
        )
        rec["type"] = _type_code(msg.type)
        rec["priority"] = msg.priority.value
        rec["timestamp"] = msg.timestamp
        rec["token_count"] = msg.token_count or 0
        rec["payload_off"] = 0
        rec["flags"] = flags
        if msg.visible_to:
            rec["vis_mode"] = VIS_BITMAP
            rec["bitmap"] = (
                bitmap_idx if bitmap_idx is not None
                else self._bitmap_for(msg.visible_to)
            )
        else:
            rec["vis_mode"] = VIS_ALL
            rec["bitmap"] = NO_BITMAP
        return rec, payload, overflow

    def _register_enqueued(self, msg: Message, seq: int, overflow) -> None:
        self._id_to_seq[msg.id] = seq
        if overflow is not None:
            self._overflow[seq] = overflow
        self._maybe_prune_host_maps()

    def _maybe_prune_host_maps(self) -> None:
        """Evict ``_id_to_seq``/``_overflow`` entries whose seq fell below
        the device retention horizon (the slot ring's evict_base) — the
        compat-path host maps must not outlive the messages they index,
        or a long-running service leaks host RAM while the ring evicts.
        Amortized: a full sweep every ``_PRUNE_EVERY`` inserts."""
        self._prune_tick += 1
        if self._prune_tick < self._PRUNE_EVERY:
            return
        self._prune_tick = 0
        eb = self.engine.evict_base()
        if eb <= self._pruned_below:
            return
        self._id_to_seq = {
            k: v for k, v in self._id_to_seq.items() if v >= eb
        }
        if self._overflow:
            self._overflow = {
                s: blob for s, blob in self._overflow.items() if s >= eb
            }
        self._pruned_below = eb

    def send_messages_bulk(self, msgs: List[Message]) -> List[str]:
        """Enqueue many validated Messages as ONE engine batch (the
        micro-batcher's flush path — SURVEY.md §7 hard part 3: batched
        submission across concurrent API requests). Senders/receivers
        must be registered; returns message ids in order."""
        with self._lock:
            recs = []
            payloads = []
            overflows = []
            off = 0
            for m in msgs:
                rec, payload, overflow = self._encode_message(m)
                pad = (-len(payload)) % 16
                rec["payload_off"] = off
                recs.append(rec)
                payloads.append(payload + b"\x00" * pad)
                overflows.append(overflow)
                off += len(payload) + pad
            batch = np.concatenate(recs) if recs else np.empty(0, dtype=REC_DTYPE)
            seqs = self.engine.enqueue_batch(batch, b"".join(payloads))
            for m, s, ov in zip(msgs, seqs, overflows):
                self._register_enqueued(m, int(s), ov)
        self._maybe_autosave()
        return [m.id for m in msgs]

    def make_message(
        self,
        sender_id: str,
        content: Union[str, Dict[str, Any], List[Any]],
        receiver_id: Optional[str] = None,
        message_type: Union[MessageType, str] = MessageType.CHAT,
        priority: Union[MessagePriority, int] = MessagePriority.NORMAL,
        metadata: Optional[Dict[str, Any]] = None,
        visible_to: Optional[List[str]] = None,
    ) -> Message:
        """Validate + register, producing a Message ready for
        send_messages_bulk (same semantics as send_message's prologue)."""
        with self._lock:
            self.register_agent(sender_id)
            if receiver_id is not None:
                self.register_agent(receiver_id)
            vis = list(visible_to) if visible_to else []
            if receiver_id is None and not vis:
                vis = sorted(self.registered_agents)
            return Message(
                sender_id=sender_id,
                receiver_id=receiver_id,
                content=content,
                type=MessageType(message_type),
                priority=(
                    priority
                    if isinstance(priority, MessagePriority)
                    else MessagePriority(priority)
                ),
                metadata=metadata or {},
                token_count=self._count_tokens(content),
                visible_to=vis,
            )

    def broadcast_message(
        self,
        sender_id: str,
        content: Union[str, Dict[str, Any], List[Any]],
        message_type: Union[MessageType, str] = MessageType.CHAT,
        priority: Union[MessagePriority, int] = MessagePriority.NORMAL,
        metadata: Optional[Dict[str, Any]] = None,
        exclude_agents: Optional[List[str]] = None,
    ) -> str:
        """Broadcast to everyone except sender + excludes (reference
        swarmdb/ main.py:810-850)."""
        with self._lock:
            self.register_agent(sender_id)
            excl = set(exclude_agents or [])
            excl.add(sender_id)
            vis = sorted(a for a in self.registered_agents if a not in excl)
            return self.send_message(
                sender_id=sender_id,
                content=content,
                receiver_id=None,
                message_type=message_type,
                priority=priority,
                metadata=metadata,
                visible_to=vis,
            )

    def send_to_group(
        self,
        group_name: str,
        sender_id: str,
        content: Union[str, Dict[str, Any], List[Any]],
        message_type: Union[MessageType, str] = MessageType.CHAT,
        priority: Union[MessagePriority, int] = MessagePriority.NORMAL,
        metadata: Optional[Dict[str, Any]] = None,
    ) -> List[str]:
        """Group fan-out (reference swarmdb/ main.py:1229-1279: N sequential
        sends). Observable result kept (one message id per member, sender
        skipped, metadata['group'] stamped); delivery is a single batched
        enqueue instead of N independent send paths."""
        with self._lock:
            groups = self.metadata.get("agent_groups", {})
            if group_name not in groups:
                raise ValueError(f"group '{group_name}' does not exist")
            members = [a for a in groups[group_name] if a != sender_id]
            md = dict(metadata or {})
            md["group"] = group_name
            ids: List[str] = []
            for member in members:
                ids.append(
                    self.send_message(
                        sender_id=sender_id,
                        content=content,
                        receiver_id=member,
                        message_type=message_type,
                        priority=priority,
                        metadata=md,
                    )
                )
            return ids

    def send_to_group_fast(
        self,
        group_name: str,
        sender_id: str,
        content: Union[str, Dict[str, Any], List[Any]],
        message_type: Union[MessageType, str] = MessageType.CHAT,
        priority: Union[MessagePriority, int] = MessagePriority.NORMAL,
        metadata: Optional[Dict[str, Any]] = None,
    ) -> str:
        """Single-slot group fan-out (the GPU-native group path,
        BASELINE config 3): ONE message slot, the fan-out kernel appends
        the entry to member inboxes only (VIS_GROUP). Returns one message
        id — use :meth:`send_to_group` for the reference-compatible
        one-id-per-member behavior."""
        with self._lock:
            groups = self.metadata.get("agent_groups", {})
            if group_name not in groups:
                raise ValueError(f"group '{group_name}' does not exist")
            self.register_agent(sender_id)
            members = [a for a in groups[group_name] if a != sender_id]
            md = dict(metadata or {})
            md["group"] = group_name
            mtype = MessageType(message_type)
            prio = (
                priority
                if isinstance(priority, MessagePriority)
                else MessagePriority(priority)
            )
            msg = Message(
                sender_id=sender_id,
                receiver_id=None,
                content=content,
                type=mtype,
                priority=prio,
                metadata=md,
                token_count=self._count_tokens(content),
                visible_to=members,
            )
            rec, payload, overflow = self._encode_message(
                msg, bitmap_idx=self._group_bitmap(group_name, members)
            )
            rec["vis_mode"] = VIS_GROUP
            seqs = self.engine.enqueue_batch(rec, payload)
            self._register_enqueued(msg, int(seqs[0]), overflow)
            self._maybe_autosave()
            return msg.id

    def _group_bitmap(self, group_name: str, members: List[str]) -> int:
        """Cached per-(group, membership) visibility bitmap."""
        key = (group_name, tuple(members))
        cache = self.metadata.setdefault("_group_bitmaps", {})
        if key not in cache:
            cache[key] = self._bitmap_for(members)
        return cache[key]

    def resend_failed_messages(self) -> List[str]:
        """Re-send every FAILED message as a NEW message, linked via
        metadata['resent_from'] (reference swarmdb/ main.py:1096-1130)."""
        with self._lock:
            new_ids: List[str] = []
            failed = list(self._failed.items())
            for old_id, msg in failed:
                md = dict(msg.metadata)
                md.pop("error", None)
                md["resent_from"] = old_id
                try:
                    nid = self.send_message(
                        sender_id=msg.sender_id,
                        content=msg.content,
                        receiver_id=msg.receiver_id,
                        message_type=msg.type,
                        priority=msg.priority,
                        metadata=md,
                        visible_to=msg.visible_to or None,
                    )
                except Exception:
                    continue
                new_ids.append(nid)
                del self._failed[old_id]
            return new_ids

    # ------------------------------------------------------------------
    # batch hot path (new — the array API bench.py and bulk REST use)
    # ------------------------------------------------------------------

    def send_batch(self, recs: np.ndarray, payloads: bytes) -> np.ndarray:
        """Zero-per-message-Python send: REC_DTYPE array + one payload
        buffer straight into the engine (pinned staging + one kernel on
        GPU). Message ids are derived from (rank, seq). Returns seqs."""
        with tracer.span("send_batch", n=len(recs)):
            seqs = self.engine.enqueue_batch(recs, payloads)
        self._maybe_autosave()
        return seqs

    def receive_batch(
        self, agent_idxs: np.ndarray, max_per_agent: int = 100,
        priority_order: bool = False,
    ):
        """Drain many agents in one engine call (one dequeue kernel on
        GPU). Returns (counts, seqs)."""
        with tracer.span("receive_batch", n=len(agent_idxs)):
            return self.engine.receive_many(
                agent_idxs, max_per_agent, priority_order
            )

    def agent_index(self, agent_id: str) -> int:
        with self._lock:
            self.register_agent(agent_id)
            return self._agent_idx[agent_id]

    # ------------------------------------------------------------------
    # receive / read path (reference swarmdb/ main.py:521-781, 1026-1047)
    # ------------------------------------------------------------------

    def _seq_of(self, message_id: str) -> Optional[int]:
        seq = self._id_to_seq.get(message_id)
        if seq is not None:
            return seq
        parsed = parse_derived_id(message_id)
        if parsed is not None:
            rank, seq = parsed
            if rank == self.config.rank and seq < self.engine.total_messages():
                return seq
        return None

    def _messages_from_seqs(self, seqs: np.ndarray) -> List[Message]:
        if len(seqs) == 0:
            return []
        hdrs, payloads = self.engine.fetch(np.asarray(seqs, dtype=np.uint64))
        out: List[Message] = []
        for row, payload in zip(hdrs, payloads):
            out.append(self._reconstruct(row, payload))
        return out

    def _reconstruct(self, row: np.void, payload: bytes) -> Message:
        flags = int(row["flags"])
        if flags & FLAG_OVERFLOW:
            clen, payload = self._overflow.get(int(row["seq"]), (0, b""))
        else:
            clen = int(row["content_len"])
        content = decode_content(payload[:clen], bool(flags & FLAG_JSON_CONTENT))
        extras = (
            decode_extras(payload[clen:]) if flags & FLAG_HAS_EXTRAS else {}
        )
        seq = int(row["seq"])
        msg_id = extras.get("id") or derived_id(self.config.rank, seq)
        recv = int(row["receiver"])
        sidx = int(row["sender"])
        return Message(
            id=msg_id,
            sender_id=(
                self._agent_ids[sidx] if sidx < len(self._agent_ids) else f"agent{sidx}"
            ),
            receiver_id=(
                None
                if recv == BROADCAST
                else (
                    self._agent_ids[recv]
                    if recv < len(self._agent_ids)
                    else f"agent{recv}"
                )
            ),
            content=content,
            type=MessageType(TYPE_NAMES[int(row["type"])]),
            priority=MessagePriority(int(row["priority"])),
            timestamp=float(row["timestamp"]),
            status=MessageStatus(STATUS_NAMES[min(int(row["status"]), ST_FAILED)]),
            metadata=extras.get("metadata", {}),
            token_count=int(row["token_count"]),
            visible_to=extras.get("visible_to", []),
        )

    def receive_messages(
        self,
        agent_id: str,
        max_messages: int = 100,
        timeout: float = 1.0,
        priority_order: bool = False,
    ) -> List[Message]:
        """Poll the agent's inbox cursor (reference swarmdb/
        main.py:521-601). Unknown agents are auto-registered
        (main.py:538-542). The engine applies the visibility filter and
        marks messages READ; ``timeout`` bounds the wait for the first
        message (the reference's consumer-poll timeout)."""
        with self._lock:
            self.register_agent(agent_id)
            idx = self._agent_idx[agent_id]
        deadline = time.monotonic() + max(0.0, timeout)
        while True:
            seqs = self.engine.receive(idx, max_messages, priority_order)
            if len(seqs) or time.monotonic() >= deadline:
                break
            time.sleep(0.001)
        return self._messages_from_seqs(seqs)

    def get_message(self, message_id: str) -> Optional[Message]:
        """Reference swarmdb/ main.py:603-613."""
        with self._lock:
            seq = self._seq_of(message_id)
            if seq is None:
                return None
            if self.engine.get_status(seq) == ST_DELETED:
                return None
            msgs = self._messages_from_seqs(np.array([seq], dtype=np.uint64))
            return msgs[0] if msgs else None

    def get_agent_messages(
        self,
        agent_id: str,
        status: Optional[Union[MessageStatus, str]] = None,
        limit: int = 100,
        skip: int = 0,
    ) -> List[Message]:
        """Paginated newest-first inbox listing (reference swarmdb/
        main.py:615-652). Pure engine-side; no transport involved."""
        with self._lock:
            idx = self._agent_idx.get(agent_id)
            if idx is None:
                return []
        seqs = self.engine.peek_inbox(idx)[::-1]  # newest first
        # reference pagination order (main.py:640-652): skip raw inbox
        # entries first, THEN apply the status filter, THEN cap at limit
        seqs = seqs[skip:]
        if status is not None and len(seqs):
            code = _status_code(status)
            st = self.engine.statuses(seqs)
            seqs = seqs[st == code]
        seqs = seqs[:limit]
        return self._messages_from_seqs(seqs.copy())

    def mark_message_as_processed(self, message_id: str) -> bool:
        """Reference swarmdb/ main.py:654-669."""
        with self._lock:
            seq = self._seq_of(message_id)
            if seq is None:
                return False
            self.engine.set_status(seq, ST_PROCESSED)
            return True

    def update_message_status(
        self, message_id: str, status: Union[MessageStatus, str]
    ) -> bool:
        """Direct status mutation (the API's PUT /messages/{id}/status path,
        reference api.py:691-733)."""
        with self._lock:
            seq = self._seq_of(message_id)
            if seq is None:
                return False
            self.engine.set_status(seq, _status_code(status))
            return True

    def query_messages(
        self,
        sender_id: Optional[str] = None,
        receiver_id: Optional[str] = None,
        message_type: Optional[Union[MessageType, str]] = None,
        status: Optional[Union[MessageStatus, str]] = None,
        after_timestamp: Optional[float] = None,
        before_timestamp: Optional[float] = None,
        limit: int = 100,
    ) -> List[Message]:
        """Filtered newest-first scan (reference swarmdb/ main.py:671-740;
        both timestamp bounds exclusive). Vectorized on CPU, a filter
        kernel on GPU — not a Python loop over all messages."""
        with self._lock:
            skw: Dict[str, Any] = {}
            if sender_id is not None:
                i = self._agent_idx.get(sender_id)
                if i is None:
                    return []
                skw["sender"] = i
            if receiver_id is not None:
                i = self._agent_idx.get(receiver_id)
                if i is None:
                    return []
                skw["receiver"] = i
        if message_type is not None:
            skw["type_code"] = _type_code(message_type)
        if status is not None:
            skw["status"] = _status_code(status)
        seqs = self.engine.query(
            after=after_timestamp, before=before_timestamp, limit=limit, **skw
        )
        return self._messages_from_seqs(seqs)

    def search_messages(
        self, keyword: str, case_sensitive: bool = False, limit: int = 100
    ) -> List[Message]:
        """Keyword scan over message content, newest-first (reference
        swarmdb/ main.py:742-781)."""
        seqs = self.engine.search(keyword.encode("utf-8"), case_sensitive, limit)
        return self._messages_from_seqs(seqs)

    def get_conversation(
        self, agent1_id: str, agent2_id: str, limit: int = 100,
        sort: bool = False,
    ) -> List[Message]:
        """Two half-limit queries concatenated, not interleaved (reference
        swarmdb/ main.py:783-808 — kept). ``sort=True``
        opts into chronological interleaving (the behavior the reference
        presumably intended)."""
        a_to_b = self.query_messages(
            sender_id=agent1_id, receiver_id=agent2_id, limit=limit // 2
        )
        b_to_a = self.query_messages(
            sender_id=agent2_id, receiver_id=agent1_id, limit=limit // 2
        )
        out = a_to_b + b_to_a
        if sort:
            out.sort(key=lambda m: m.timestamp)
        return out

    def get_unread_message_count(self, agent_id: str) -> int:
        """Inbox entries still in DELIVERED state (reference swarmdb/
        main.py:1026-1047)."""
        with self._lock:
            idx = self._agent_idx.get(agent_id)
        if idx is None:
            return 0
        return self.engine.unread_count(idx)

    def delete_message(self, message_id: str) -> bool:
        """Tombstone (reference swarmdb/ main.py:1132-1157: pops the
        message and scrubs every inbox; the tombstone gives the same
        observable results in O(1))."""
        with self._lock:
            seq = self._seq_of(message_id)
            if seq is None:
                return False
            ok = self.engine.delete(seq)
            self._id_to_seq.pop(message_id, None)
            self._overflow.pop(seq, None)
            return ok

    # ------------------------------------------------------------------
    # express lane — persistent-kernel single-message latency plane
    # (the batched tick stays the throughput plane; this is for the
    # reference's p2p "agent waits on one message" regime where the
    # tick's ~0.4 ms floor dominates: measured p50 9.7 us on MI355X)
    # ------------------------------------------------------------------

    def express_start(self, agent_ids: List[str], slot_bytes: int = 1024,
                      max_seconds: float = 3600.0) -> None:
        """Open the express lane for a fixed set of agents. On GPU this
        starts the resident doorbell kernel (csrc k_doorbell: pinned
        mailboxes, no kernel launch per message); on the CPU engine an
        in-process queue double keeps the API testable anywhere."""
        with self._lock:
            if getattr(self, "_express", None) is not None:
                raise RuntimeError("express lane already running")
            self._express_idx = {a: i for i, a in enumerate(agent_ids)}
            for a in agent_ids:
                self.register_agent(a)
            if hasattr(self.engine, "q"):
                from .. import _swarmq  # type: ignore

                db = _swarmq.DoorbellQueue(
                    slot_bytes=slot_bytes, sub_cap=1024,
                    n_agents=len(agent_ids), ring_cap=1024,
                    device=self.config.device_index,
                )
                db.start(max_seconds)
                self._express = db
            else:
                from collections import deque

                class _CpuExpress:
                    def __init__(self, n):
                        self.rings = [deque() for _ in range(n)]

                    def send(self, receiver, sender, payload):
                        self.rings[receiver].append((sender, bytes(payload)))
                        return 0

                    def try_recv(self, agent):
                        ring = self.rings[agent]
                        return ring.popleft() if ring else None

                    def recv_spin(self, agent, timeout_us=0.0):
                        return self.try_recv(agent)

                    def stop(self):
                        pass

                    def release(self):
                        pass

                self._express = _CpuExpress(len(agent_ids))

    def express_send(self, sender_id: str, receiver_id: str,
                     content: Union[str, bytes]) -> None:
        """Single-message express send (content only — the latency plane
        carries no metadata/visibility; use send_message for those)."""
        ex = getattr(self, "_express", None)
        if ex is None:
            raise RuntimeError("express lane not started")
        payload = content.encode() if isinstance(content, str) else content
        ex.send(self._express_idx[receiver_id],
                self._express_idx.get(sender_id, 0), payload)

    def express_recv(self, agent_id: str,
                     timeout_us: float = 1e6) -> Optional[tuple]:
        """Blocking-with-deadline express receive; returns
        (sender_id, payload bytes) or None."""
        ex = getattr(self, "_express", None)
        if ex is None:
            raise RuntimeError("express lane not started")
        got = ex.recv_spin(self._express_idx[agent_id], timeout_us)
        if got is None:
            return None
        sidx, payload = got
        inv = {i: a for a, i in self._express_idx.items()}
        return inv.get(int(sidx), f"agent{sidx}"), bytes(payload)

    def express_stop(self) -> None:
        ex = getattr(self, "_express", None)
        if ex is not None:
            ex.stop()
            ex.release()
            self._express = None

    # ------------------------------------------------------------------
    # groups (reference swarmdb/ main.py:1208-1227)
    # ------------------------------------------------------------------

    def add_agent_group(self, group_name: str, agent_ids: List[str]) -> None:
        """Reference swarmdb/ main.py:1208-1227 (stored in self.metadata,
        not in the history file — SURVEY.md §8.13; persisted separately in
        a sidecar by save_message_history)."""
        with self._lock:
            self.metadata.setdefault("agent_groups", {})[group_name] = list(agent_ids)
            for a in agent_ids:
                self.register_agent(a)

    def get_agent_groups(self) -> Dict[str, List[str]]:
        with self._lock:
            return dict(self.metadata.get("agent_groups", {}))

    # ------------------------------------------------------------------
    # LLM load balancing (reference swarmdb/ main.py:1049-1094, 1281-1325)
    # ------------------------------------------------------------------

    def set_llm_load_balancing(self, enabled: bool) -> None:
        """Reference swarmdb/ main.py:1281-1291 — there the flag is dead;
        here it gates dispatch_llm_request's least-loaded selection."""
        with self._lock:
            self.llm_load_balancing = bool(enabled)

    def register_llm_backend(self, backend_id: str) -> int:
        """Register a backend in the load table (new — the mechanism the
        reference lacks; per-backend load words live device-side on GPU)."""
        with self._lock:
            if backend_id in self._llm_backend_idx:
                return self._llm_backend_idx[backend_id]
            idx = len(self._llm_backends)
            if idx >= self.config.num_backends:
                raise RuntimeError("backend capacity exceeded")
            self._llm_backends.append(backend_id)
            self._llm_backend_idx[backend_id] = idx
            return idx

    def assign_llm_backend(self, agent_id: str, backend_id: str) -> None:
        """Reference swarmdb/ main.py:1293-1311."""
        with self._lock:
            self.register_llm_backend(backend_id)
            self.metadata.setdefault("llm_backends", {})[agent_id] = backend_id

    def get_llm_backend(self, agent_id: str) -> Optional[str]:
        """Reference swarmdb/ main.py:1313-1325."""
        with self._lock:
            return self.metadata.get("llm_backends", {}).get(agent_id)

    def dispatch_llm_request(self, agent_id: Optional[str] = None) -> str:
        """Least-loaded backend selection (BASELINE config 5). If the agent
        has a pinned backend and balancing is off, use it; otherwise argmin
        over the per-backend load counters (a wavefront min-reduce kernel
        on GPU). Increments the chosen backend's in-flight count."""
        with self._lock:
            n = len(self._llm_backends)
            if n == 0:
                raise RuntimeError("no LLM backends registered")
            if agent_id is not None and not self.llm_load_balancing:
                pinned = self.get_llm_backend(agent_id)
                if pinned is not None:
                    idx = self._llm_backend_idx[pinned]
                    self.engine.backend_add_load(idx, 1)
                    return pinned
        idx = int(self.engine.dispatch_batch(1, n)[0])
        return self._llm_backends[idx]

    def complete_llm_request(self, backend_id: str) -> None:
        with self._lock:
            idx = self._llm_backend_idx[backend_id]
        self.engine.backend_add_load(idx, -1)

    def get_agent_load(self, agent_id: str) -> Dict[str, Any]:
        """Load probe (reference swarmdb/ main.py:1049-1094):
        processing_rate = messages received in the last 60 s / 60."""
        with self._lock:
            idx = self._agent_idx.get(agent_id)
        if idx is None:
            return {
                "agent_id": agent_id,
                "total_messages": 0,
                "inbox_size": 0,
                "unread_count": 0,
                "processing_rate": 0.0,
            }
        stats = self.engine.stats_arrays()
        inbox = self.engine.peek_inbox(idx)
        return {
            "agent_id": agent_id,
            "total_messages": int(stats["sent"][idx] + stats["received"][idx]),
            "inbox_size": int(len(inbox)),
            "unread_count": self.engine.unread_count(idx),
            "processing_rate": self.engine.recv_rate_window(idx, 60.0) / 60.0,
        }

    # ------------------------------------------------------------------
    # persistence & ops (reference swarmdb/ main.py:852-1024, 1159-1206,
    # 1327-1365)
    # ------------------------------------------------------------------

    def _message_dict(self, row: np.void, payload: bytes) -> Dict[str, Any]:
        """Reconstruct the wire dict straight from an engine row —
        the spill path skips pydantic construction entirely."""
        flags = int(row["flags"])
        if flags & FLAG_OVERFLOW:
            clen, payload = self._overflow.get(int(row["seq"]), (0, b""))
        else:
            clen = int(row["content_len"])
        extras = (
            decode_extras(payload[clen:]) if flags & FLAG_HAS_EXTRAS else {}
        )
        seq = int(row["seq"])
        recv = int(row["receiver"])
        sidx = int(row["sender"])
        return {
            "id": extras.get("id") or derived_id(self.config.rank, seq),
            "sender_id": (
                self._agent_ids[sidx]
                if sidx < len(self._agent_ids)
                else f"agent{sidx}"
            ),
            "receiver_id": (
                None
                if recv == BROADCAST
                else (
                    self._agent_ids[recv]
                    if recv < len(self._agent_ids)
                    else f"agent{recv}"
                )
            ),
            "content": decode_content(
                payload[:clen], bool(flags & FLAG_JSON_CONTENT)
            ),
            "type": TYPE_NAMES[int(row["type"])],
            "priority": int(row["priority"]),
            "timestamp": float(row["timestamp"]),
            "status": STATUS_NAMES[min(int(row["status"]), ST_FAILED)],
            "metadata": extras.get("metadata", {}),
            "token_count": int(row["token_count"]),
            "visible_to": extras.get("visible_to", []),
        }

    def _history_object(self) -> Dict[str, Any]:
        """The §2.1 history schema (reference swarmdb/ main.py:877-884)."""
        n = self.engine.total_messages()
        seqs = np.arange(n, dtype=np.uint64)
        st = self.engine.statuses(seqs) if n else np.empty(0, dtype=np.uint8)
        seqs = seqs[st != ST_DELETED]
        messages: Dict[str, Any] = {}
        seq_to_id: Dict[int, str] = {}
        done = 0
        while done < len(seqs):
            chunk = seqs[done : done + 16384]
            hdrs, payloads = self.engine.fetch(chunk)
            for row, payload in zip(hdrs, payloads):
                d = self._message_dict(row, payload)
                messages[d["id"]] = d
                seq_to_id[int(row["seq"])] = d["id"]
            done += len(chunk)
        with self._lock:
            inbox_obj: Dict[str, List[str]] = {}
            for agent_id, idx in self._agent_idx.items():
                entries = self.engine.peek_inbox(idx)
                inbox_obj[agent_id] = [
                    seq_to_id[int(s)] for s in entries if int(s) in seq_to_id
                ]
            registered = sorted(self.registered_agents)
        return {
            "messages": messages,
            "agent_inbox": inbox_obj,
            "registered_agents": registered,
            "timestamp": time.time(),
            "message_count": n,
        }

    def _maybe_autosave(self) -> None:
        """Auto-save trigger cadence (reference swarmdb/ main.py:492-497)
        but the serialization runs on the spill thread, off the send path
        (SURVEY.md §8.14). On GPU the device->host copy goes through pinned
        staging on a side stream (engine.fetch)."""
        if not self.config.auto_save or self._closed:
            return
        n = self.engine.total_messages()
        due = (
            time.time() - self.last_save_time > self.config.save_interval
            or (n > 0 and n % self.config.max_messages_per_file == 0)
        )
        if due and not self._spill_pending:
            self._spill_pending = True
            self._spill.submit(self._spill_save)

    def _spill_save(self) -> None:
        try:
            self.save_message_history()
        except Exception as e:  # pragma: no cover
            logger.error("background save failed: %s", e)
        finally:
            self._spill_pending = False

    def save_message_history(self) -> str:
        """Full-state JSON snapshot (reference swarmdb/ main.py:852-892).
        File name + schema byte-compatible; groups/LLM assignments go to a
        sidecar file (schema compat)."""
        history = self._history_object()
        self.save_dir.mkdir(parents=True, exist_ok=True)
        ts = datetime.now().strftime("%Y%m%d_%H%M%S")
        path = self.save_dir / f"message_history_{ts}_{history['message_count']}.json"
        # indent=2 is the reference's on-disk format; indent 0 (config)
        # selects compact JSON, which runs on the C encoder (~6x faster)
        indent = self.config.history_indent or None
        with open(path, "w") as f:
            json.dump(history, f, indent=indent)
        with self._lock:
            sidecar = {
                "agent_groups": self.metadata.get("agent_groups", {}),
                "llm_backends": self.metadata.get("llm_backends", {}),
                "agent_metadata": self.agent_metadata,
            }
        with open(self.save_dir / f"metadata_{ts}.json", "w") as f:
            json.dump(sidecar, f, indent=indent)
        self.last_save_time = time.time()
        logger.info("saved history to %s", path)
        return str(path)

    def load_message_history(self, path: Union[str, Path]) -> int:
        """Rebuild state from a history file (reference swarmdb/
        main.py:894-934): re-register agents, replay messages into the
        engine, restore statuses. Returns loaded message count."""
        with open(path) as f:
            history = json.load(f)
        with self._lock:
            for agent_id in history.get("registered_agents", []):
                self.register_agent(agent_id)
            msgs = sorted(
                (
                    Message.from_dict(d, validate=False)
                    for d in history.get("messages", {}).values()
                ),
                key=lambda m: m.timestamp,
            )
            for m in msgs:
                if m.receiver_id is not None:
                    self.register_agent(m.receiver_id)
                self.register_agent(m.sender_id)
            # one batched enqueue + one batched status restore (a
            # per-message replay would cost a kernel launch per message)
            n = len(msgs)
            recs = np.zeros(n, dtype=REC_DTYPE)
            chunks: List[bytes] = []
            off = 0
            overflow_pending: Dict[int, tuple] = {}
            for i, m in enumerate(msgs):
                content_b, is_json = encode_content(m.content)
                extras_b = encode_extras(m.id, m.metadata, m.visible_to)
                payload = content_b + extras_b
                if len(payload) > self._slot_capacity():
                    overflow_pending[i] = (len(content_b), payload)
                    recs["flags"][i] = FLAG_OVERFLOW
                    payload = b""
                    content_b = b""
                pad = (-len(payload)) % 16
                chunks.append(payload + b"\x00" * pad)
                recs["sender"][i] = self._agent_idx[m.sender_id]
                recs["receiver"][i] = (
                    BROADCAST
                    if m.receiver_id is None
                    else self._agent_idx[m.receiver_id]
                )
                recs["type"][i] = _type_code(m.type)
                recs["priority"][i] = m.priority.value
                recs["timestamp"][i] = m.timestamp
                recs["token_count"][i] = m.token_count or 0
                recs["payload_off"][i] = off
                recs["payload_len"][i] = len(payload)
                recs["content_len"][i] = len(content_b)
                recs["flags"][i] |= FLAG_HAS_EXTRAS | (
                    FLAG_JSON_CONTENT if is_json else 0
                )
                if m.visible_to:
                    recs["vis_mode"][i] = VIS_BITMAP
                    recs["bitmap"][i] = self._bitmap_for(m.visible_to)
                else:
                    recs["vis_mode"][i] = VIS_ALL
                    recs["bitmap"][i] = NO_BITMAP
                off += len(payload) + pad
            if n:
                seqs = self.engine.enqueue_batch(recs, b"".join(chunks))
                for m, s in zip(msgs, seqs):
                    self._id_to_seq[m.id] = int(s)
                for i, blob in overflow_pending.items():
                    self._overflow[int(seqs[i])] = blob
                self.engine.set_statuses(
                    seqs,
                    np.fromiter(
                        (_status_code(m.status) for m in msgs),
                        dtype=np.uint32,
                        count=n,
                    ),
                )
            return n

    # ---- binary incremental checkpoints (operational persistence;
    # the JSON history above stays the reference-compatible format) ----

    def save_checkpoint(self, path: Optional[str] = None) -> str:
        """Full binary snapshot at device-gather speed (batched pinned
        D2H, zero per-message Python — see runtime/checkpoint.py).
        Resets the delta chain."""
        from .checkpoint import save_checkpoint

        return save_checkpoint(self, path)

    def save_checkpoint_delta(self):
        """Append messages since the last checkpoint/delta to the
        base's .delta file. Returns (path, records appended)."""
        from .checkpoint import save_checkpoint_delta

        return save_checkpoint_delta(self)

    def load_checkpoint(self, path: Union[str, Path],
                        with_deltas: bool = True) -> int:
        """Replay a binary checkpoint (+ delta chain). Returns records
        loaded."""
        from .checkpoint import load_checkpoint

        return load_checkpoint(self, path, with_deltas)

    def export_as_yaml(self) -> str:
        """History object as YAML (reference swarmdb/ main.py:936-971)."""
        history = self._history_object()
        self.save_dir.mkdir(parents=True, exist_ok=True)
        ts = datetime.now().strftime("%Y%m%d_%H%M%S")
        path = self.save_dir / f"message_history_{ts}.yaml"
        with open(path, "w") as f:
            yaml.safe_dump(history, f, default_flow_style=False, sort_keys=False)
        return str(path)

    def flush_old_messages(self, older_than_seconds: Optional[float] = None) -> int:
        """Archive + tombstone messages older than the cutoff (reference
        swarmdb/ main.py:1159-1206; default 7 days). Archive format: bare
        {msg_id: msg_dict} (main.py:1184-1196)."""
        if older_than_seconds is None:
            older_than_seconds = self.config.retention_ms / 1000.0
        cutoff = time.time() - older_than_seconds
        archive: Dict[str, Any] = {}
        count = 0
        # the engine may cap a single query at its staging depth; loop
        # until the matching set is drained
        while True:
            seqs = self.engine.query(before=cutoff, limit=1 << 30)
            if len(seqs) == 0:
                break
            msgs = self._messages_from_seqs(seqs)
            with self._lock:
                for s, m in zip(seqs, msgs):
                    if self.engine.delete(int(s)):
                        archive[m.id] = m.to_dict()
                        self._id_to_seq.pop(m.id, None)
                        self._overflow.pop(int(s), None)
                        count += 1
        if count == 0:
            return 0
        archive_dir = self.save_dir / "archives"
        archive_dir.mkdir(parents=True, exist_ok=True)
        path = archive_dir / f"archive_{int(time.time())}.json"
        with open(path, "w") as f:
            json.dump(archive, f, indent=self.config.history_indent or None)
        logger.info("flushed %d old messages to %s", count, path)
        return count

    def get_stats(self) -> Dict[str, Any]:
        """Running device/engine counters instead of O(N) scans (reference
        swarmdb/ main.py:973-1024)."""
        stats = self.engine.stats_arrays()
        with self._lock:
            agents = list(self._agent_idx.items())
            active = set(self.registered_agents)
        by_type = {
            TYPE_NAMES[i]: int(c) for i, c in enumerate(stats["by_type"]) if c
        }
        by_status = {
            STATUS_NAMES[i]: int(c)
            for i, c in enumerate(stats["by_status"])
            if c and i != ST_DELETED
        }
        by_agent = {}
        for agent_id, idx in agents:
            s = int(stats["sent"][idx])
            r = int(stats["received"][idx])
            if s or r:
                by_agent[agent_id] = {"sent": s, "received": r, "total": s + r}
        return {
            "total_messages": self.engine.total_messages(),
            "active_agents": len(active),
            "messages_by_type": by_type,
            "messages_by_status": by_status,
            "messages_by_agent": by_agent,
            # inbox-ring overwrites of unread entries under slow
            # consumers (always 0 on the CPU engine)
            "messages_dropped": int(stats.get("dropped", 0)),
            "last_save_time": self.last_save_time,
        }

    def auto_scale_partitions(self) -> Dict[str, int]:
        """Partition elasticity (reference swarmdb/ main.py:1327-1365):
        recommended = max(3, ceil(agents/10)*3); partitions only grow."""
        with self._lock:
            n_agents = len(self.registered_agents)
            recommended = max(3, (n_agents + 9) // 10 * 3)
            current = self.config.num_partitions
            if recommended > current:
                self.config.num_partitions = recommended
                resize = getattr(self.engine, "resize_partitions", None)
                if resize is not None:
                    resize(recommended)
                logger.info("partitions scaled %d -> %d", current, recommended)
            return {
                "previous_partitions": current,
                "current_partitions": self.config.num_partitions,
                "registered_agents": n_agents,
            }

    # ------------------------------------------------------------------
    # lifecycle (reference swarmdb/ main.py:1367-1394)
    # ------------------------------------------------------------------

    def close(self) -> None:
        """Auto-save then release the engine (reference swarmdb/
        main.py:1367-1388)."""
        if self._closed:
            return
        self._closed = True
        self.express_stop()
        if self.config.auto_save and self.engine.total_messages() > 0:
            try:
                self.save_message_history()
            except Exception as e:  # pragma: no cover
                logger.error("save on close failed: %s", e)
        self._spill.shutdown(wait=True)
        self.engine.close()
        logger.info("SwarmsDB closed")

    def __enter__(self) -> "SwarmsDB":
        return self

    def __exit__(self, *exc) -> None:
        self.close()
