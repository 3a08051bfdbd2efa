"""Unit tests: JWT, stable hashing, wire encoding."""

import time

import pytest

from swarmdb_amd.core import wire
from swarmdb_amd.utils import jwt as jwtlib
from swarmdb_amd.utils.hashing import fnv1a64, partition_for, shard_for, stable_hash


# ---- jwt ----

def test_jwt_round_trip():
    tok = jwtlib.encode({"sub": "alice", "exp": time.time() + 60}, "secret")
    assert jwtlib.decode(tok, "secret")["sub"] == "alice"


def test_jwt_bad_signature():
    tok = jwtlib.encode({"sub": "alice"}, "secret")
    with pytest.raises(jwtlib.JWTError):
        jwtlib.decode(tok, "other-secret")


def test_jwt_expired():
    tok = jwtlib.encode({"sub": "a", "exp": time.time() - 1}, "s")
    with pytest.raises(jwtlib.JWTError, match="expired"):
        jwtlib.decode(tok, "s")


def test_jwt_malformed():
    for bad in ["", "a.b", "a.b.c.d", "!!!.???.###"]:
        with pytest.raises(jwtlib.JWTError):
            jwtlib.decode(bad, "s")


def test_jwt_tamper_payload():
    tok = jwtlib.encode({"sub": "user"}, "s")
    h, b, sig = tok.split(".")
    evil = jwtlib.encode({"sub": "admin"}, "s").split(".")[1]
    with pytest.raises(jwtlib.JWTError):
        jwtlib.decode(f"{h}.{evil}.{sig}", "s")


# ---- hashing ----

def test_fnv1a_known_vectors():
    # standard FNV-1a 64-bit test vectors
    assert fnv1a64(b"") == 0xCBF29CE484222325
    assert fnv1a64(b"a") == 0xAF63DC4C8601EC8C
    assert fnv1a64(b"foobar") == 0x85944171F73967E8


def test_partition_stability_and_range():
    # deterministic across calls/processes (unlike builtin hash,
    # SURVEY.md §8.6) and in range
    for agent in ["agent1", "agent2", "x" * 100]:
        p = partition_for(agent, 7)
        assert p == partition_for(agent, 7)
        assert 0 <= p < 7
    assert partition_for("anything", 1) == 0


def test_shard_decorrelated_from_partition():
    # shard uses high bits: agents in one partition spread over shards
    shards = {shard_for(f"a{i}", 8) for i in range(100)}
    assert len(shards) == 8
    assert shard_for("x", 1) == 0


# ---- wire ----

def test_derived_id_round_trip():
    for rank, seq in [(0, 0), (3, 12345), (255, (1 << 48) - 1)]:
        did = wire.derived_id(rank, seq)
        assert wire.parse_derived_id(did) == (rank, seq)


def test_derived_id_never_matches_uuid4():
    import uuid

    for _ in range(20):
        assert wire.parse_derived_id(str(uuid.uuid4())) is None


def test_content_encoding():
    b, j = wire.encode_content("plain text")
    assert b == b"plain text" and not j
    assert wire.decode_content(b, j) == "plain text"
    b, j = wire.encode_content({"k": [1, 2]})
    assert j
    assert wire.decode_content(b, j) == {"k": [1, 2]}
    b, j = wire.encode_content([1, "two"])
    assert wire.decode_content(b, j) == [1, "two"]


def test_extras_encoding():
    assert wire.encode_extras(None, {}, []) == b""
    assert wire.decode_extras(b"") == {}
    b = wire.encode_extras("id1", {"m": 1}, ["a"])
    d = wire.decode_extras(b)
    assert d == {"id": "id1", "metadata": {"m": 1}, "visible_to": ["a"]}


# ---- config ----

def test_queue_config_from_env(monkeypatch):
    from swarmdb_amd.core.config import QueueConfig

    monkeypatch.setenv("KAFKA_TOPIC_PREFIX", "team_")
    monkeypatch.setenv("KAFKA_NUM_PARTITIONS", "9")
    monkeypatch.setenv("MESSAGE_HISTORY_DIR", "/tmp/hist")
    monkeypatch.setenv("SAVE_INTERVAL_SECONDS", "60")
    monkeypatch.setenv("SWARMQ_MAX_AGENTS", "256")
    monkeypatch.setenv("SWARMQ_SLOT_BYTES", "4096")
    cfg = QueueConfig.from_env(auto_save=False)
    assert cfg.base_topic == "team_messages"
    assert cfg.num_partitions == 9
    assert cfg.save_dir == "/tmp/hist"
    assert cfg.save_interval == 60.0
    assert cfg.max_agents == 256
    assert cfg.slot_bytes == 4096
    assert cfg.auto_save is False  # override wins
