"""Stable hashing for partition / shard routing.

The reference routes with Python's salted builtin ``hash`` (swarmdb/
main.py:309-312), which is nondeterministic across processes (SURVEY.md
§8.6). Cross-process and cross-GPU routing needs a stable hash; FNV-1a is
also what the device-side kernels implement (csrc/swarmq_common.h) so host
and device agree on agent->partition and agent->GPU mapping.
"""

from __future__ import annotations

FNV_OFFSET = 0xCBF29CE484222325
FNV_PRIME = 0x100000001B3
_MASK64 = (1 << 64) - 1


def fnv1a64(data: bytes) -> int:
    """64-bit FNV-1a. Must match fnv1a64() in csrc/swarmq_common.h."""
    h = FNV_OFFSET
    for b in data:
        h ^= b
        h = (h * FNV_PRIME) & _MASK64
    return h


def stable_hash(s: str) -> int:
    return fnv1a64(s.encode("utf-8"))


def partition_for(agent_id: str, num_partitions: int) -> int:
    """agent -> partition (reference semantics: hash % partitions,
    swarmdb/ main.py:309-312, made deterministic)."""
    return stable_hash(agent_id) % max(1, num_partitions)


def mix64(h: int) -> int:
    """splitmix64 finalizer — FNV-1a's raw bits distribute poorly under
    small moduli on short keys; this avalanches them."""
    h = (h ^ (h >> 30)) * 0xBF58476D1CE4E5B9 & _MASK64
    h = (h ^ (h >> 27)) * 0x94D049BB133111EB & _MASK64
    return h ^ (h >> 31)


def shard_for(agent_id: str, world_size: int) -> int:
    """agent -> GPU rank for cross-GPU sharding (mixed so it decorrelates
    from partition_for's low-bit modulo)."""
    if world_size <= 1:
        return 0
    return mix64(stable_hash(agent_id)) % world_size
