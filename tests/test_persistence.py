"""Persistence / checkpoint-resume tests (SURVEY.md §5.4, §2.1 file formats)."""

import json
import time
from pathlib import Path

import pytest
import yaml

from swarmdb_amd import MessageStatus, QueueConfig, SwarmsDB


def _mk(tmp_path, **kw):
    cfg = QueueConfig(use_gpu=False, save_dir=str(tmp_path / "hist"), **kw)
    return SwarmsDB(config=cfg)


def test_history_file_schema(tmp_path):
    db = _mk(tmp_path)
    id1 = db.send_message("a", "hello", receiver_id="b")
    id2 = db.broadcast_message("a", {"k": 1})
    path = db.save_message_history()
    data = json.loads(Path(path).read_text())
    # exact top-level schema (reference swarmdb/ main.py:877-884)
    assert set(data.keys()) == {
        "messages", "agent_inbox", "registered_agents", "timestamp",
        "message_count",
    }
    assert set(data["messages"].keys()) == {id1, id2}
    d1 = data["messages"][id1]
    assert set(d1.keys()) == {
        "id", "sender_id", "receiver_id", "content", "type", "priority",
        "timestamp", "status", "metadata", "token_count", "visible_to",
    }
    assert d1["type"] == "chat" and d1["priority"] == 1
    assert data["agent_inbox"]["b"] == [id1, id2]
    # broadcast lands in the sender's inbox too (reference behavior,
    # SURVEY.md §8.11)
    assert data["agent_inbox"]["a"] == [id2]
    assert sorted(data["registered_agents"]) == ["a", "b"]
    assert data["message_count"] == 2
    # file name pattern message_history_{ts}_{count}.json
    assert Path(path).name.startswith("message_history_")
    assert Path(path).name.endswith("_2.json")
    db.config.auto_save = False
    db.close()


def test_load_message_history_round_trip(tmp_path):
    db = _mk(tmp_path)
    id1 = db.send_message("a", "persisted", receiver_id="b",
                          metadata={"k": "v"})
    db.receive_messages("b", timeout=0)
    db.mark_message_as_processed(id1)
    path = db.save_message_history()
    db.config.auto_save = False
    db.close()

    db2 = _mk(tmp_path)
    n = db2.load_message_history(path)
    assert n == 1
    assert db2.registered_agents == {"a", "b"}
    m = db2.get_message(id1)
    assert m is not None
    assert m.content == "persisted"
    assert m.metadata == {"k": "v"}
    assert m.status == MessageStatus.PROCESSED
    db2.config.auto_save = False
    db2.close()


def test_yaml_export(tmp_path):
    db = _mk(tmp_path)
    mid = db.send_message("a", "yaml me", receiver_id="b")
    path = db.export_as_yaml()
    data = yaml.safe_load(Path(path).read_text())
    assert mid in data["messages"]
    assert data["messages"][mid]["content"] == "yaml me"
    db.config.auto_save = False
    db.close()


def test_flush_old_messages_archive(tmp_path):
    db = _mk(tmp_path)
    old_id = db.send_message("a", "ancient", receiver_id="b")
    # backdate it by rewriting the engine header timestamp
    seq = db._id_to_seq[old_id]
    db.engine._hdr["timestamp"][seq] = time.time() - 10 * 86400
    new_id = db.send_message("a", "fresh", receiver_id="b")
    flushed = db.flush_old_messages()  # default 7-day cutoff
    assert flushed == 1
    assert db.get_message(old_id) is None
    assert db.get_message(new_id) is not None
    archives = list((Path(db.save_dir) / "archives").glob("archive_*.json"))
    assert len(archives) == 1
    arch = json.loads(archives[0].read_text())
    # archive format: bare {msg_id: msg_dict} (main.py:1184-1196)
    assert list(arch.keys()) == [old_id]
    assert arch[old_id]["content"] == "ancient"
    db.config.auto_save = False
    db.close()


def test_autosave_trigger_by_count(tmp_path):
    db = _mk(tmp_path, max_messages_per_file=10, save_interval=1e9)
    for i in range(10):
        db.send_message("a", f"m{i}", receiver_id="b")
    db._spill.shutdown(wait=True)  # let the background save finish
    files = list(Path(db.config.save_dir).glob("message_history_*.json"))
    assert len(files) >= 1
    db.config.auto_save = False
    db._spill = __import__("concurrent.futures", fromlist=["ThreadPoolExecutor"]).ThreadPoolExecutor(max_workers=1)
    db.close()


def test_close_saves_history(tmp_path):
    db = _mk(tmp_path)
    db.send_message("a", "x", receiver_id="b")
    db.close()
    files = list(Path(db.config.save_dir).glob("message_history_*.json"))
    assert len(files) == 1


def test_groups_sidecar_persisted(tmp_path):
    db = _mk(tmp_path)
    db.add_agent_group("g", ["a", "b"])
    db.assign_llm_backend("a", "backendX")
    db.send_message("a", "x", receiver_id="b")
    db.save_message_history()
    sidecars = list(Path(db.config.save_dir).glob("metadata_*.json"))
    assert sidecars
    side = json.loads(sidecars[0].read_text())
    assert side["agent_groups"] == {"g": ["a", "b"]}
    assert side["llm_backends"] == {"a": "backendX"}
    db.config.auto_save = False
    db.close()


def test_oversized_content_survives_save_load(tmp_path):
    db = _mk(tmp_path)
    big = "Q" * (db.config.slot_bytes * 2)
    mid = db.send_message("a", big, receiver_id="b")
    path = db.save_message_history()
    db.config.auto_save = False
    db.close()
    db2 = _mk(tmp_path)
    db2.load_message_history(path)
    assert db2.get_message(mid).content == big
    assert db2.receive_messages("b", timeout=0)[0].content == big
    db2.config.auto_save = False
    db2.close()


def test_binary_checkpoint_roundtrip(tmp_path):
    """Binary base + delta checkpoint: full state (messages, statuses,
    visibility, groups, overflow) replays into a fresh facade."""
    cfg = QueueConfig(use_gpu=False, save_dir=str(tmp_path), auto_save=False,
                      max_agents=64, slot_bytes=512)
    db = SwarmsDB(config=cfg)
    for a in ["alice", "bob", "carol"]:
        db.register_agent(a)
    m1 = db.send_message("alice", "plain one", receiver_id="bob")
    m2 = db.send_message("alice", {"k": [1, 2]}, receiver_id="carol",
                         priority=3, metadata={"tag": "x"})
    db.broadcast_message("bob", "to everyone")
    db.send_message("alice", "secret", receiver_id=None,
                    visible_to=["carol"])
    big = "B" * 2000  # exceeds slot_bytes -> host overflow store
    mo = db.send_message("bob", big, receiver_id="alice")
    db.mark_message_as_processed(m1)
    db.add_agent_group("team", ["alice", "bob"])

    base = db.save_checkpoint()
    # post-base traffic -> delta segment
    m3 = db.send_message("carol", "after base", receiver_id="bob")
    path_delta, n_delta = db.save_checkpoint_delta()
    assert n_delta == 1
    # empty delta appends nothing
    assert db.save_checkpoint_delta()[1] == 0

    db2 = SwarmsDB(config=QueueConfig(
        use_gpu=False, save_dir=str(tmp_path), auto_save=False,
        max_agents=64, slot_bytes=512))
    loaded = db2.load_checkpoint(base)
    assert loaded == 6
    assert db2.registered_agents == {"alice", "bob", "carol"}
    assert db2.get_agent_groups() == {"team": ["alice", "bob"]}

    got_bob = db2.receive_messages("bob", timeout=0)
    contents = [m.content for m in got_bob]
    assert "plain one" in contents
    assert "after base" in contents  # the delta replayed too
    # bob sent the broadcast: excluded from its visible_to
    assert "to everyone" not in contents
    assert "secret" not in contents  # visibility respected after reload
    got_carol = db2.receive_messages("carol", timeout=0)
    assert {m.content if isinstance(m.content, str) else "dict"
            for m in got_carol} == {"dict", "to everyone", "secret"}
    # overflow payload survived; alice sees the broadcast
    got_alice = db2.receive_messages("alice", timeout=0)
    acontents = [m.content for m in got_alice]
    assert big in acontents and "to everyone" in acontents
    # statuses restored: m1 was processed
    msgs = db2.query_messages(status="processed")
    assert [m.content for m in msgs] == ["plain one"]
    db.close()
    db2.close()


def test_binary_checkpoint_replay_into_nonempty_facade(tmp_path):
    """The replay remaps dense agent indices: loading into a facade
    whose index table differs (pre-existing agents in another order)
    must still deliver to the right agents and rebuild visibility."""
    cfg = dict(use_gpu=False, save_dir=str(tmp_path), auto_save=False,
               max_agents=64, slot_bytes=512)
    db = SwarmsDB(config=QueueConfig(**cfg))
    for a in ["alice", "bob", "carol"]:
        db.register_agent(a)  # alice=0, bob=1, carol=2
    db.send_message("alice", "for bob", receiver_id="bob")
    db.send_message("bob", "members only", receiver_id=None,
                    visible_to=["alice"])
    base = db.save_checkpoint()

    db2 = SwarmsDB(config=QueueConfig(**cfg))
    # conflicting pre-existing table: carol=0, zed=1, bob=2, alice=3
    for a in ["carol", "zed", "bob", "alice"]:
        db2.register_agent(a)
    db2.send_message("zed", "pre-existing", receiver_id="bob")
    loaded = db2.load_checkpoint(base)
    assert loaded == 2
    got_bob = sorted(m.content for m in db2.receive_messages("bob",
                                                             timeout=0))
    assert got_bob == ["for bob", "pre-existing"]
    got_alice = [m.content for m in db2.receive_messages("alice",
                                                         timeout=0)]
    assert got_alice == ["members only"]
    # the restricted message never reaches non-members
    assert db2.receive_messages("zed", timeout=0) == []
    assert db2.receive_messages("carol", timeout=0) == []
    db.close()
    db2.close()
