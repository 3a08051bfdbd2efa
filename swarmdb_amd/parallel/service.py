"""DistributedSwarmsDB — the agent-sharded multi-GPU service.

One instance per rank (one process per GPU, ``torch.distributed``; nccl =
RCCL over xGMI on GPU, gloo on CPU for CI). Replaces the reference's
"many workers sharing one Kafka broker" topology (SURVEY.md §2.4 row
"cross-process shared state") with:

- **control plane, tick-synchronized**: agent registrations, visibility
  bitmaps, group definitions and migrations are queued as ops and
  applied on EVERY rank in a deterministic order each tick (a fused
  tensor header exchange + a padded-u8 JSON all_gather only when ops
  exist — no pickle, quiet tick = one tiny collective), so the dense
  agent-index table, ownership table and bitmap pool stay bit-identical
  across ranks with no coordinator;
- **data plane**: outbound messages batch into a per-tick exchange
  routed by owner rank (dense agent index mod world — the same mapping
  the all-to-all router applies) via all-to-all; an agent's inbox lives
  only on its owner rank.

Usage::

    svc = DistributedSwarmsDB(config)        # inside an initialized group
    svc.register_agent("alice")              # queued
    svc.send_message("alice", "hi", receiver_id="bob")   # queued
    svc.tick()                               # ALL ranks call together
    msgs = svc.receive_messages("bob")       # on bob's owner rank

``start_ticker(interval)`` runs the tick loop on a daemon thread when the
process has no other collectives in flight.
"""

from __future__ import annotations

import json
import threading
import time
from typing import Any, Dict, List, Optional, Union

import numpy as np
import torch
import torch.distributed as dist

from ..core.config import QueueConfig
from ..core.message import Message, MessagePriority, MessageType
from ..core.wire import encode_content, encode_extras
from ..runtime.engine import (
    BROADCAST,
    FLAG_HAS_EXTRAS,
    FLAG_JSON_CONTENT,
    NO_BITMAP,
    REC_DTYPE,
    TYPE_CODES,
    VIS_ALL,
    VIS_BITMAP,
)
from ..runtime.facade import SwarmsDB
from .router import CrossGpuRouter


class DistributedSwarmsDB(SwarmsDB):
    def __init__(
        self,
        config: Optional[QueueConfig] = None,
        group: Optional[object] = None,
        **kw,
    ):
        self.group = group
        self.world = dist.get_world_size(group)
        self.rank = dist.get_rank(group)
        config = config or QueueConfig()
        config.world_size = self.world
        config.rank = self.rank
        super().__init__(config=config, **kw)
        self._ctl_ops: List[tuple] = []     # queued control ops
        self._ctl_seq = 0
        self._out_msgs: List[Message] = []      # queued sends (local)
        self._out_recs: List[np.ndarray] = []   # materialized data records
        self._out_pay: List[bytes] = []
        self._out_bytes = 0
        self._pending_meta: Dict[str, Message] = {}  # id -> msg (sender side)
        device = (
            torch.device("cuda", config.device_index)
            if config.use_gpu or (
                config.use_gpu is None and torch.cuda.is_available()
            )
            else torch.device("cpu")
        )
        self._router = CrossGpuRouter(device, group)
        # replicated ownership table (agent idx -> rank); default is
        # idx % world, rewritten by live migration ops
        self._owner_vec = np.full(config.max_agents, -1, dtype=np.int64)
        # control-plane tensors live wherever the collective backend
        # wants them (cuda for nccl/RCCL, cpu for gloo)
        self._ctl_device = device
        self._ticker: Optional[threading.Thread] = None
        self._stop = threading.Event()

    # ------------------------------------------------------------------
    # sharding
    # ------------------------------------------------------------------

    def owner_rank(self, agent_id: str) -> int:
        """Owner rank from the replicated ownership table (defaults to
        dense index mod world; live migration rewrites entries). The
        table is tick-synchronized, so every rank agrees. Requires the
        agent to be registered (and a tick to have run)."""
        idx = self._agent_idx.get(agent_id)
        if idx is None:
            raise KeyError(
                f"agent '{agent_id}' not registered yet (register + tick)"
            )
        return int(self._owner_vec[idx])

    def is_local(self, agent_id: str) -> bool:
        return self.owner_rank(agent_id) == self.rank

    # ------------------------------------------------------------------
    # control plane (tick-synchronized)
    # ------------------------------------------------------------------

    def _queue_ctl(self, op: tuple) -> None:
        with self._lock:
            self._ctl_ops.append((self._ctl_seq, *op))
            self._ctl_seq += 1

    def register_agent(self, agent_id: str) -> bool:  # type: ignore[override]
        """Queued; the dense index is assigned identically on every rank
        at the next tick. Idempotent."""
        with self._lock:
            if agent_id in self.registered_agents:
                return True
        self._queue_ctl(("register", agent_id))
        return True

    def _apply_register(self, agent_id: str) -> None:
        with self._lock:
            if agent_id in self.registered_agents:
                return
            idx = self._idx_of(agent_id, create=True)
            if self._owner_vec[idx] < 0:
                self._owner_vec[idx] = idx % self.world
            # active (inbox fan-out target) only on the owner rank; the
            # index table itself is replicated on every rank
            if self.is_local(agent_id):
                self.engine.register_agent(idx)
            self.registered_agents.add(agent_id)

    def deregister_agent(self, agent_id: str) -> bool:  # type: ignore[override]
        self._queue_ctl(("deregister", agent_id))
        return True

    def _apply_deregister(self, agent_id: str) -> None:
        with self._lock:
            if agent_id not in self.registered_agents:
                return
            self.registered_agents.discard(agent_id)
            idx = self._agent_idx[agent_id]
            if self.is_local(agent_id):
                self.engine.deregister_agent(idx)

    def migrate_agent(self, agent_id: str, new_rank: int) -> None:
        """Queue a live re-shard of one agent to ``new_rank`` (the
        re-sharding the reference approximates with Kafka partition
        growth, swarmdb/ main.py:1327-1365 — here ownership actually
        moves). Applied at the next tick on every rank: the old owner
        drains the agent's undelivered inbox and re-homes those
        messages through the same exchange, the new owner activates the
        inbox, and the replicated ownership table flips — in-flight and
        later sends route to the new rank. The agent's already-read
        HISTORY stays on the old shard (shard-local queries, as
        documented)."""
        if not (0 <= new_rank < self.world):
            raise ValueError(f"rank {new_rank} outside world {self.world}")
        self._queue_ctl(("migrate", agent_id, int(new_rank)))

    def rebalance_plan(self) -> Dict[str, int]:
        """Round-robin re-spread of all registered agents (a helper for
        operators; apply with migrate_agent + tick)."""
        with self._lock:
            agents = sorted(self.registered_agents)
        return {a: i % self.world for i, a in enumerate(agents)}

    def _apply_migrate(self, agent_id: str, new_rank: int) -> None:
        with self._lock:
            idx = self._agent_idx.get(agent_id)
            if idx is None:
                return
            old = int(self._owner_vec[idx])
            if old == new_rank:
                return
            self._owner_vec[idx] = new_rank
            if new_rank == self.rank:
                self.engine.register_agent(idx)
            if old != self.rank:
                return
            # old owner: drain undelivered entries and re-home them
            # through this tick's exchange (they route by the updated
            # ownership table)
            while True:
                seqs = self.engine.receive(idx, 4096)
                if len(seqs) == 0:
                    break
                hdrs, flat, stride = self.engine.fetch_raw_chunks(seqs)
                recs = np.zeros(len(hdrs), dtype=REC_DTYPE)
                for name in REC_DTYPE.names:
                    recs[name] = hdrs[name]
                recs["receiver"] = idx  # re-home as direct deliveries
                unsplit = getattr(self.engine, "unsplit_bitmap_handles",
                                  None)
                if unsplit is not None:
                    recs = unsplit(recs)
                lens = recs["payload_len"].astype(np.int64)
                plen16 = (lens + 15) // 16 * 16
                offs = np.zeros(len(recs), dtype=np.int64)
                np.cumsum(plen16[:-1], out=offs[1:])
                mat = flat.reshape(-1, stride)
                ar = np.arange(stride)
                mask = ar[None, :] < plen16[:, None]
                payload = mat[mask].tobytes()
                recs["payload_off"] = (
                    offs + self._out_bytes
                ).astype(np.uint64)
                self._out_recs.append(recs)
                self._out_pay.append(payload)
                self._out_bytes += len(payload)
            # deactivate locally (the agent stays registered
            # service-wide; fan-out targets only the new owner's engine)
            self.engine.deregister_agent(idx)

    def add_agent_group(self, group_name: str, agent_ids: List[str]) -> None:  # type: ignore[override]
        for a in agent_ids:
            self.register_agent(a)
        self._queue_ctl(("group", group_name, list(agent_ids)))

    def _apply_group(self, group_name: str, agent_ids: List[str]) -> None:
        with self._lock:
            self.metadata.setdefault("agent_groups", {})[group_name] = list(
                agent_ids
            )

    def _apply_bitmap(self, members: List[str]) -> int:
        """Replicated allocation: every rank allocates the same bitmap at
        the same pool index (allocators run in lockstep)."""
        bits = np.zeros(self.config.max_agents, dtype=bool)
        for a in members:
            i = self._agent_idx.get(a)
            if i is not None:
                bits[i] = True
        return self.engine.alloc_bitmap(bits)

    # ------------------------------------------------------------------
    # data plane
    # ------------------------------------------------------------------

    def send_message(  # type: ignore[override]
        self,
        sender_id: str,
        content: Union[str, Dict[str, Any], List[Any]],
        receiver_id: Optional[str] = None,
        message_type: Union[MessageType, str] = MessageType.CHAT,
        priority: Union[MessagePriority, int] = MessagePriority.NORMAL,
        metadata: Optional[Dict[str, Any]] = None,
        visible_to: Optional[List[str]] = None,
    ) -> str:
        """Queued send; delivery happens at the next tick on the
        receiver's owner rank. Returns the message id immediately (the
        id travels in the payload extras)."""
        self.register_agent(sender_id)
        if receiver_id is not None:
            self.register_agent(receiver_id)
        mtype = MessageType(message_type)
        prio = (
            priority
            if isinstance(priority, MessagePriority)
            else MessagePriority(priority)
        )
        vis = list(visible_to) if visible_to else []
        if receiver_id is None and not vis:
            with self._lock:
                vis = sorted(self.registered_agents)
        msg = Message(
            sender_id=sender_id,
            receiver_id=receiver_id,
            content=content,
            type=mtype,
            priority=prio,
            metadata=metadata or {},
            token_count=self._count_tokens(content),
            visible_to=vis,
        )
        # the single-process facade spills oversized content to a host
        # overflow store; across ranks the payload must fit a device slot
        # (the owner rank has no host copy) — fail loudly at send time.
        # Encode ONCE here and carry the bytes to materialization.
        content_b, is_json = encode_content(content)
        extras_b = encode_extras(msg.id, msg.metadata, vis)
        approx = len(content_b) + len(extras_b)
        if approx > int(self.config.slot_bytes):
            raise ValueError(
                f"content ({approx} B) exceeds the distributed slot "
                f"capacity ({self.config.slot_bytes} B); raise "
                "SWARMQ_SLOT_BYTES"
            )
        msg._wire = (content_b, is_json, extras_b)  # type: ignore[attr-defined]
        with self._lock:
            self._out_msgs.append(msg)
            self._pending_meta[msg.id] = msg
            if len(self._pending_meta) > 65536:
                for k in list(self._pending_meta)[:32768]:
                    del self._pending_meta[k]
        return msg.id

    # ------------------------------------------------------------------
    # the tick (ALL ranks must call together)
    # ------------------------------------------------------------------

    def _gather_blobs(self, blob: bytes, sizes: List[int]) -> List[bytes]:
        """Max-length-padded u8 all_gather of per-rank byte blobs whose
        lengths are already known on every rank."""
        dev = self._ctl_device
        mx = max(sizes)
        buf = torch.zeros(mx, dtype=torch.uint8, device=dev)
        if blob:
            buf[: len(blob)] = torch.frombuffer(
                bytearray(blob), dtype=torch.uint8
            ).to(dev)
        outs = [torch.empty(mx, dtype=torch.uint8, device=dev)
                for _ in range(self.world)]
        dist.all_gather(outs, buf, group=self.group)
        return [
            bytes(outs[r][: sizes[r]].cpu().numpy().tobytes())
            if sizes[r]
            else b""
            for r in range(self.world)
        ]

    def tick(self) -> int:
        """One control+data exchange round. Returns the number of
        messages ingested locally this tick.

        The tick's collective budget (replaces round 1's per-tick
        ``all_gather_object`` pickle, ~10 ms at world 2 gloo):

        1. ONE fused header all_to_all carrying (control-blob bytes,
           pending-send count), both replicated per destination — on a
           quiet tick this is the ONLY collective;
        2. only when some rank queued control ops: one padded u8
           all_gather of JSON blobs (data, not pickle);
        3. only when some rank queued sends: the router's exchange.
        """
        with self._lock:
            my_ops = self._ctl_ops
            self._ctl_ops = []
            self._ctl_seq = 0
            out_msgs, self._out_msgs = self._out_msgs, []
        # visibility bitmaps must exist in every rank's pool at the same
        # index: ship the member lists with the control gather and have
        # EVERY rank allocate them in the same order
        my_vis = [m.visible_to for m in out_msgs if m.visible_to]
        blob = b""
        if my_ops or my_vis:
            blob = json.dumps([my_ops, my_vis],
                              separators=(",", ":")).encode()
        dev = self._ctl_device
        hdr = torch.tensor(
            [len(blob), len(out_msgs)] * self.world,
            dtype=torch.int64, device=dev,
        )
        in_hdr = torch.empty_like(hdr)
        dist.all_to_all_single(in_hdr, hdr, group=self.group)
        pairs = in_hdr.reshape(self.world, 2).cpu()
        ctl_sizes = [int(x) for x in pairs[:, 0]]
        any_sends = bool(int(pairs[:, 1].sum()))
        if max(ctl_sizes) == 0:
            gathered_ops: List[list] = [[] for _ in range(self.world)]
            gathered_vis: List[list] = [[] for _ in range(self.world)]
        else:
            raws = self._gather_blobs(blob, ctl_sizes)
            gathered_ops, gathered_vis = [], []
            for raw in raws:
                if raw:
                    ops, vis = json.loads(raw)
                else:
                    ops, vis = [], []
                gathered_ops.append(ops)
                gathered_vis.append(vis)

        # apply control ops deterministically: (rank, op_seq) order; all
        # non-send ops first so sends see a consistent registry
        all_ops = []
        for r, ops in enumerate(gathered_ops):
            for op in ops or []:
                all_ops.append((r, *op))
        all_ops.sort(key=lambda t: (t[0], t[1]))
        for r, _seq, kind, *args in all_ops:
            if kind == "register":
                self._apply_register(args[0])
            elif kind == "deregister":
                self._apply_deregister(args[0])
            elif kind == "group":
                self._apply_group(args[0], args[1])
            elif kind == "migrate":
                self._apply_migrate(args[0], args[1])
        # replicated bitmap allocation (identical pool on every rank)
        my_bitmaps: List[int] = []
        for r in range(self.world):
            for vis_list in gathered_vis[r]:
                bidx = self._apply_bitmap(vis_list)
                if r == self.rank:
                    my_bitmaps.append(bidx)
        # materialize this rank's queued sends now that the tick's
        # registry/bitmap state is applied everywhere — ONE records
        # array for the whole tick (a per-message 1-row array cost
        # ~17 us of numpy dtype churn each)
        if out_msgs:
            bit_iter = iter(my_bitmaps)
            recs = np.zeros(len(out_msgs), dtype=REC_DTYPE)
            pays: List[bytes] = []
            off = self._out_bytes
            for i, msg in enumerate(out_msgs):
                content_b, is_json, extras_b = msg._wire  # type: ignore[attr-defined]
                payload = content_b + extras_b
                pad = (-len(payload)) % 16
                pays.append(payload + b"\x00" * pad)
                recs["sender"][i] = self._agent_idx[msg.sender_id]
                recs["receiver"][i] = (
                    BROADCAST
                    if msg.receiver_id is None
                    else self._agent_idx[msg.receiver_id]
                )
                recs["type"][i] = TYPE_CODES[msg.type.value]
                recs["priority"][i] = msg.priority.value
                recs["timestamp"][i] = msg.timestamp
                recs["token_count"][i] = msg.token_count or 0
                recs["payload_off"][i] = off
                recs["payload_len"][i] = len(payload)
                recs["content_len"][i] = len(content_b)
                recs["flags"][i] = FLAG_HAS_EXTRAS | (
                    FLAG_JSON_CONTENT if is_json else 0
                )
                if msg.visible_to:
                    recs["vis_mode"][i] = VIS_BITMAP
                    recs["bitmap"][i] = next(bit_iter)
                else:
                    recs["vis_mode"][i] = VIS_ALL
                    recs["bitmap"][i] = NO_BITMAP
                off += len(payload) + pad
            self._out_recs.append(recs)
            self._out_pay.append(b"".join(pays))
            self._out_bytes = off

        # control ops can generate handoff traffic (migration re-homing)
        # even when no rank queued sends
        if not any_sends and max(ctl_sizes) == 0:
            return 0  # quiet tick: skip the router exchange

        if self._out_recs:
            recs = np.concatenate(self._out_recs)
            payload = b"".join(self._out_pay)
        else:
            recs = np.empty(0, dtype=REC_DTYPE)
            payload = b""
        self._out_recs, self._out_pay, self._out_bytes = [], [], 0

        in_recs, in_pay = self._router.route(
            recs, payload, owner_of=self._owner_vec
        )
        if len(in_recs):
            seqs = self.engine.enqueue_batch(in_recs, in_pay)
            # map ids for locally-ingested compat-path messages
            # (ids live in the extras; resolved lazily via fetch)
            self._maybe_autosave()
            return len(seqs)
        return 0

    def flush(self, ticks: int = 2) -> None:
        """Run `ticks` synchronized rounds (registration + delivery)."""
        for _ in range(ticks):
            self.tick()

    def start_ticker(self, interval: float = 0.002) -> None:
        """Background tick loop (every rank must start one)."""
        if self._ticker is not None:
            return

        def loop() -> None:
            while not self._stop.is_set():
                self.tick()
                time.sleep(interval)

        self._ticker = threading.Thread(target=loop, daemon=True)
        self._ticker.start()

    def stop_ticker(self) -> None:
        self._stop.set()
        if self._ticker is not None:
            self._ticker.join(timeout=10)
            self._ticker = None

    # ------------------------------------------------------------------
    # reads are owner-local
    # ------------------------------------------------------------------

    def receive_messages(  # type: ignore[override]
        self,
        agent_id: str,
        max_messages: int = 100,
        timeout: float = 1.0,
        priority_order: bool = False,
    ):
        if agent_id not in self._agent_idx:
            # unknown agent: queue the registration (applied at the next
            # tick, reference auto-register semantics) — nothing to read
            self.register_agent(agent_id)
            return []
        if not self.is_local(agent_id):
            raise RuntimeError(
                f"agent '{agent_id}' lives on rank "
                f"{self.owner_rank(agent_id)}; poll there — or serve "
                "through the REST gateway (create_app(peer_urls=...)), "
                "which 307-redirects to the owner"
            )
        with self._lock:
            idx = self._agent_idx.get(agent_id)
        if idx is None:
            return []
        deadline = time.monotonic() + max(0.0, timeout)
        while True:
            seqs = self.engine.receive(idx, max_messages, priority_order)
            if len(seqs) or time.monotonic() >= deadline:
                break
            time.sleep(0.001)
        return self._messages_from_seqs(seqs)

    def get_message(self, message_id: str):  # type: ignore[override]
        m = super().get_message(message_id)
        if m is not None:
            return m
        with self._lock:
            return self._pending_meta.get(message_id)

    def send_to_group_fast(self, group_name, sender_id, content,
                           message_type=MessageType.CHAT,
                           priority=MessagePriority.NORMAL,
                           metadata=None):  # type: ignore[override]
        """The single-slot group fan-out is a per-shard engine
        operation; in the distributed service group sends go through the
        tick exchange as per-member messages instead (one id per member,
        reference semantics)."""
        ids = self.send_to_group(group_name, sender_id, content,
                                 message_type=message_type,
                                 priority=priority, metadata=metadata)
        return ids[0] if ids else None

    # NOTE: query/search/stats/history operate on THIS rank's shard.
    # Aggregate views are the caller's concern (run them on every rank
    # and merge); the data plane keeps each message on exactly one rank.

    def close(self) -> None:  # type: ignore[override]
        self.stop_ticker()
        super().close()
