"""Queue / device configuration.

Replaces the reference's ``KafkaConfig`` (reference swarmdb/ main.py:114-127)
with a config for the GPU-resident queue. The transport-tunable names keep
their reference semantics (partitions, retention, poll timeout); the device
tier (ring sizes, slot bytes, staging depth) is new — sized for MI355X
(288 GB HBM3E per GPU).
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Optional


@dataclass
class QueueConfig:
    """Configuration for the message queue runtime.

    Reference analog: ``KafkaConfig`` (swarmdb/ main.py:114-127). Kafka
    bootstrap/group options become device-queue options; names that carried
    observable semantics (num_partitions, retention_ms, consumer_timeout_ms)
    keep them.
    """

    # --- transport-semantics tier (reference-visible knobs) ---
    base_topic: str = "agent_messages"
    num_partitions: int = 3              # ring shards per GPU
    replication_factor: int = 1          # kept for API compat; unused on GPU
    retention_ms: int = 7 * 24 * 60 * 60 * 1000
    consumer_timeout_ms: int = 1000
    auto_offset_reset: str = "earliest"

    # --- device tier (MI355X) ---
    device_index: int = 0
    max_agents: int = 8192               # capacity of the device registry
    slot_bytes: int = 2048               # fixed message slot (header+payload)
    num_slots: int = 1 << 20             # ring capacity in slots (2 GiB @2KB)
    inbox_capacity: int = 1 << 16        # per-agent inbox ring entries (u64)
    recv_window: int = 4096              # entries examined per dequeue call
    #                                      (pow2; small windows keep many
    #                                      dequeue workgroups resident)
    staging_batch: int = 16384           # max messages per enqueue batch
    num_bitmaps: int = 4096              # visibility-bitmap pool depth
    #                                      (epoch-tagged ring: recycled
    #                                      entries are detected exactly)
    num_backends: int = 64               # LLM backend table capacity
    use_gpu: Optional[bool] = None       # None = auto-detect

    # --- persistence tier (reference swarmdb/ main.py:156-166, 221-230) ---
    log_file: Optional[str] = None       # rotating file log (reference
    #                                      loguru sink, swarmdb/ main.py:170-189)
    save_dir: str = "message_history"
    history_indent: int = 2              # reference snapshot format uses
    #                                      indent=2 (main.py:886); 0 opts
    #                                      into compact JSON, which uses
    #                                      Python's C encoder (~6x faster
    #                                      serialization — same structure,
    #                                      different whitespace)
    auto_save: bool = True
    save_interval: float = 300.0
    max_messages_per_file: int = 10000

    # --- cross-GPU tier ---
    world_size: int = 1
    rank: int = 0

    @classmethod
    def from_env(cls, **overrides) -> "QueueConfig":
        """Build from environment variables.

        Honors the reference env names where meaningful (api.py:38-52):
        KAFKA_NUM_PARTITIONS, KAFKA_TOPIC_PREFIX, MESSAGE_HISTORY_DIR,
        SAVE_INTERVAL_SECONDS; plus the SWARMQ_* device tier.
        """
        env = os.environ
        kw = dict(
            base_topic=env.get("KAFKA_TOPIC_PREFIX", "agent_messaging_") + "messages",
            num_partitions=int(env.get("KAFKA_NUM_PARTITIONS", "3")),
            replication_factor=int(env.get("KAFKA_REPLICATION_FACTOR", "1")),
            save_dir=env.get("MESSAGE_HISTORY_DIR", "message_history"),
            log_file=env.get("LOG_FILE") or None,
            save_interval=float(env.get("SAVE_INTERVAL_SECONDS", "300")),
            max_agents=int(env.get("SWARMQ_MAX_AGENTS", "8192")),
            slot_bytes=int(env.get("SWARMQ_SLOT_BYTES", "2048")),
            num_slots=int(env.get("SWARMQ_NUM_SLOTS", str(1 << 20))),
            inbox_capacity=int(env.get("SWARMQ_INBOX_CAPACITY", str(1 << 16))),
            staging_batch=int(env.get("SWARMQ_STAGING_BATCH", "16384")),
            num_bitmaps=int(env.get("SWARMQ_NUM_BITMAPS", "4096")),
            history_indent=int(env.get("SWARMDB_HISTORY_INDENT", "2")),
            device_index=int(env.get("SWARMQ_DEVICE", "0")),
        )
        kw.update(overrides)
        return cls(**kw)
