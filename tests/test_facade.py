"""Core-runtime (SwarmsDB facade over CpuEngine) behavior tests.

Covers the SwarmsDB method surface of SURVEY.md §2.2 — the contract the
API layer programs against. This is BASELINE config 1's plumbing.
"""

import json
import time

import numpy as np
import pytest

from swarmdb_amd import (
    Message,
    MessagePriority,
    MessageStatus,
    MessageType,
    QueueConfig,
    SwarmsDB,
)


def test_register_is_idempotent(tmp_db):
    assert tmp_db.register_agent("a1")
    assert tmp_db.register_agent("a1")
    assert "a1" in tmp_db.registered_agents


def test_deregister(tmp_db):
    tmp_db.register_agent("a1")
    assert tmp_db.deregister_agent("a1")
    assert not tmp_db.deregister_agent("a1")
    assert "a1" not in tmp_db.registered_agents


def test_send_receive_p2p(tmp_db):
    mid = tmp_db.send_message("alice", "hello bob", receiver_id="bob")
    assert isinstance(mid, str)
    msgs = tmp_db.receive_messages("bob", timeout=0)
    assert len(msgs) == 1
    m = msgs[0]
    assert m.id == mid
    assert m.sender_id == "alice"
    assert m.receiver_id == "bob"
    assert m.content == "hello bob"
    assert m.status == MessageStatus.READ
    # second receive drains nothing (cursor advanced)
    assert tmp_db.receive_messages("bob", timeout=0) == []


def test_send_auto_registers(tmp_db):
    tmp_db.send_message("s", "x", receiver_id="r")
    assert {"s", "r"} <= tmp_db.registered_agents


def test_status_lifecycle(tmp_db):
    mid = tmp_db.send_message("a", "x", receiver_id="b")
    assert tmp_db.get_message(mid).status == MessageStatus.DELIVERED
    tmp_db.receive_messages("b", timeout=0)
    assert tmp_db.get_message(mid).status == MessageStatus.READ
    assert tmp_db.mark_message_as_processed(mid)
    assert tmp_db.get_message(mid).status == MessageStatus.PROCESSED
    assert not tmp_db.mark_message_as_processed("nonexistent")


def test_broadcast_visible_to_excludes_sender(tmp_db):
    for a in ["a", "b", "c"]:
        tmp_db.register_agent(a)
    mid = tmp_db.broadcast_message("a", "hello all")
    m = tmp_db.get_message(mid)
    assert m.receiver_id is None
    assert set(m.visible_to) == {"b", "c"}
    # b and c receive it; sender a does not (reference swarmdb/
    # main.py:834-839 + 579-585)
    assert [x.id for x in tmp_db.receive_messages("b", timeout=0)] == [mid]
    assert [x.id for x in tmp_db.receive_messages("c", timeout=0)] == [mid]
    assert tmp_db.receive_messages("a", timeout=0) == []


def test_broadcast_exclude_agents(tmp_db):
    for a in ["a", "b", "c", "d"]:
        tmp_db.register_agent(a)
    mid = tmp_db.broadcast_message("a", "hi", exclude_agents=["c"])
    assert tmp_db.receive_messages("b", timeout=0)[0].id == mid
    assert tmp_db.receive_messages("c", timeout=0) == []
    assert tmp_db.receive_messages("d", timeout=0)[0].id == mid


def test_plain_send_broadcast_includes_sender_in_visible(tmp_db):
    # raw send_message(receiver=None) gives visible_to = ALL registered,
    # including the sender (reference swarmdb/ main.py:449-450)
    for a in ["a", "b"]:
        tmp_db.register_agent(a)
    mid = tmp_db.send_message("a", "x", receiver_id=None)
    m = tmp_db.get_message(mid)
    assert set(m.visible_to) == {"a", "b"}
    assert tmp_db.receive_messages("a", timeout=0)[0].id == mid


def test_visible_to_restriction_on_p2p(tmp_db):
    for a in ["a", "b"]:
        tmp_db.register_agent(a)
    tmp_db.send_message("a", "secret", receiver_id="b", visible_to=["someone_else"])
    assert tmp_db.receive_messages("b", timeout=0) == []


def test_get_agent_messages_pagination_and_filter(tmp_db):
    ids = [tmp_db.send_message("a", f"m{i}", receiver_id="b") for i in range(5)]
    # newest first
    got = tmp_db.get_agent_messages("b")
    assert [m.id for m in got] == list(reversed(ids))
    got = tmp_db.get_agent_messages("b", limit=2, skip=1)
    assert [m.id for m in got] == [ids[3], ids[2]]
    tmp_db.receive_messages("b", max_messages=2, timeout=0)
    read = tmp_db.get_agent_messages("b", status=MessageStatus.READ)
    assert {m.id for m in read} == set(ids[:2])
    delivered = tmp_db.get_agent_messages("b", status="delivered")
    assert {m.id for m in delivered} == set(ids[2:])
    assert tmp_db.get_agent_messages("unknown") == []


def test_query_messages_filters(tmp_db):
    t0 = time.time()
    id1 = tmp_db.send_message("a", "x", receiver_id="b",
                              message_type=MessageType.COMMAND)
    id2 = tmp_db.send_message("b", "y", receiver_id="a",
                              priority=MessagePriority.HIGH)
    id3 = tmp_db.send_message("a", "z", receiver_id="c")
    assert {m.id for m in tmp_db.query_messages(sender_id="a")} == {id1, id3}
    assert [m.id for m in tmp_db.query_messages(receiver_id="a")] == [id2]
    assert [m.id for m in tmp_db.query_messages(message_type="command")] == [id1]
    assert {m.id for m in tmp_db.query_messages(status="delivered")} == {id1, id2, id3}
    # exclusive bounds (reference swarmdb/ main.py:725-735)
    mid_ts = tmp_db.get_message(id2).timestamp
    after = tmp_db.query_messages(after_timestamp=mid_ts)
    assert {m.id for m in after} == {id3}
    before = tmp_db.query_messages(before_timestamp=mid_ts)
    assert {m.id for m in before} == {id1}
    # newest first + limit
    q = tmp_db.query_messages(limit=2)
    assert [m.id for m in q] == [id3, id2]
    # unknown sender -> empty
    assert tmp_db.query_messages(sender_id="ghost") == []


def test_search_messages(tmp_db):
    id1 = tmp_db.send_message("a", "the quick brown fox", receiver_id="b")
    id2 = tmp_db.send_message("a", {"note": "QUICK reply"}, receiver_id="b")
    tmp_db.send_message("a", "nothing here", receiver_id="b")
    hits = tmp_db.search_messages("quick")
    assert {m.id for m in hits} == {id1, id2}
    hits = tmp_db.search_messages("quick", case_sensitive=True)
    assert {m.id for m in hits} == {id1}
    # keyword matching an agent id must NOT hit (content-only scan)
    assert tmp_db.search_messages("alicebob") == []


def test_search_does_not_match_metadata_or_ids(tmp_db):
    tmp_db.send_message("needle_agent", "clean content", receiver_id="b",
                        metadata={"needlemeta": 1})
    assert tmp_db.search_messages("needle") == []


def test_get_conversation_halved_limit(tmp_db):
    a2b = [tmp_db.send_message("a", f"a{i}", receiver_id="b") for i in range(3)]
    b2a = [tmp_db.send_message("b", f"b{i}", receiver_id="a") for i in range(3)]
    conv = tmp_db.get_conversation("a", "b", limit=4)
    # limit//2 each direction, newest-first per direction, concatenated
    assert [m.id for m in conv] == [a2b[2], a2b[1], b2a[2], b2a[1]]


def test_unread_count(tmp_db):
    for i in range(3):
        tmp_db.send_message("a", f"m{i}", receiver_id="b")
    assert tmp_db.get_unread_message_count("b") == 3
    tmp_db.receive_messages("b", max_messages=2, timeout=0)
    assert tmp_db.get_unread_message_count("b") == 1
    assert tmp_db.get_unread_message_count("ghost") == 0


def test_delete_message(tmp_db):
    mid = tmp_db.send_message("a", "x", receiver_id="b")
    assert tmp_db.delete_message(mid)
    assert tmp_db.get_message(mid) is None
    assert not tmp_db.delete_message(mid)
    assert tmp_db.receive_messages("b", timeout=0) == []
    assert tmp_db.get_agent_messages("b") == []


def test_groups(tmp_db):
    tmp_db.add_agent_group("team", ["a", "b", "c"])
    assert tmp_db.get_agent_groups() == {"team": ["a", "b", "c"]}
    ids = tmp_db.send_to_group("team", "a", "hello team")
    assert len(ids) == 2  # sender skipped
    for member in ["b", "c"]:
        msgs = tmp_db.receive_messages(member, timeout=0)
        assert len(msgs) == 1
        assert msgs[0].metadata["group"] == "team"
    assert tmp_db.receive_messages("a", timeout=0) == []
    with pytest.raises(ValueError):
        tmp_db.send_to_group("nope", "a", "x")


def test_priority_order_receive(tmp_db):
    ids = []
    for p in [MessagePriority.LOW, MessagePriority.CRITICAL,
              MessagePriority.NORMAL, MessagePriority.HIGH]:
        ids.append(tmp_db.send_message("a", f"p{p.value}", receiver_id="b",
                                       priority=p))
    msgs = tmp_db.receive_messages("b", timeout=0, priority_order=True)
    assert [m.priority for m in msgs] == [
        MessagePriority.CRITICAL, MessagePriority.HIGH,
        MessagePriority.NORMAL, MessagePriority.LOW,
    ]


def test_priority_partial_drain_keeps_rest(tmp_db):
    for p in [0, 3, 1, 2]:
        tmp_db.send_message("a", f"p{p}", receiver_id="b",
                            priority=MessagePriority(p))
    first = tmp_db.receive_messages("b", max_messages=2, timeout=0,
                                    priority_order=True)
    assert [m.priority.value for m in first] == [3, 2]
    rest = tmp_db.receive_messages("b", timeout=0, priority_order=True)
    assert [m.priority.value for m in rest] == [1, 0]


def test_token_counter_hook(tmp_path):
    cfg = QueueConfig(use_gpu=False, save_dir=str(tmp_path), auto_save=False)
    db = SwarmsDB(config=cfg, token_counter=lambda s: len(s.split()))
    mid = db.send_message("a", "one two three", receiver_id="b")
    assert db.get_message(mid).token_count == 3
    mid2 = db.send_message("a", {"k": "v"}, receiver_id="b")
    assert db.get_message(mid2).token_count == len(json.dumps({"k": "v"}).split())
    db.close()


def test_stats(tmp_db):
    tmp_db.send_message("a", "x", receiver_id="b")
    tmp_db.send_message("a", "y", receiver_id="b",
                        message_type=MessageType.COMMAND)
    tmp_db.receive_messages("b", timeout=0)
    s = tmp_db.get_stats()
    assert s["total_messages"] == 2
    assert s["active_agents"] == 2
    assert s["messages_by_type"] == {"chat": 1, "command": 1}
    assert s["messages_by_status"] == {"read": 2}
    assert s["messages_by_agent"]["a"] == {"sent": 2, "received": 0, "total": 2}
    assert s["messages_by_agent"]["b"] == {"sent": 0, "received": 2, "total": 2}


def test_agent_load(tmp_db):
    tmp_db.send_message("a", "x", receiver_id="b")
    tmp_db.send_message("a", "y", receiver_id="b")
    tmp_db.receive_messages("b", max_messages=1, timeout=0)
    load = tmp_db.get_agent_load("b")
    assert load["inbox_size"] == 2
    assert load["unread_count"] == 1
    assert load["total_messages"] == 1  # received only; b sent nothing
    assert load["processing_rate"] > 0


def test_auto_scale_partitions(tmp_db):
    for i in range(25):
        tmp_db.register_agent(f"agent{i}")
    out = tmp_db.auto_scale_partitions()
    assert out["current_partitions"] == 9  # ceil(25/10)*3
    # partitions never shrink
    for i in range(25):
        tmp_db.deregister_agent(f"agent{i}")
    out = tmp_db.auto_scale_partitions()
    assert out["current_partitions"] == 9


def test_llm_backend_assignment(tmp_db):
    tmp_db.assign_llm_backend("agent1", "backend-A")
    assert tmp_db.get_llm_backend("agent1") == "backend-A"
    assert tmp_db.get_llm_backend("ghost") is None


def test_llm_least_loaded_dispatch(tmp_db):
    for b in ["b0", "b1", "b2"]:
        tmp_db.register_llm_backend(b)
    tmp_db.set_llm_load_balancing(True)
    # load up b0 and b1
    assert tmp_db.dispatch_llm_request() in {"b0", "b1", "b2"}
    counts = {"b0": 0, "b1": 0, "b2": 0}
    for _ in range(9):
        counts[tmp_db.dispatch_llm_request()] += 1
    # least-loaded keeps them balanced
    assert max(counts.values()) - min(counts.values()) <= 1
    tmp_db.complete_llm_request("b0")
    assert tmp_db.dispatch_llm_request() == "b0"


def test_llm_pinned_backend_without_balancing(tmp_db):
    tmp_db.register_llm_backend("pinned")
    tmp_db.register_llm_backend("other")
    tmp_db.assign_llm_backend("agent1", "pinned")
    tmp_db.set_llm_load_balancing(False)
    for _ in range(3):
        assert tmp_db.dispatch_llm_request("agent1") == "pinned"


def test_batch_send_receive(tmp_db):
    from swarmdb_amd.runtime.engine import REC_DTYPE, VIS_ALL, NO_BITMAP

    a = tmp_db.agent_index("a")
    b = tmp_db.agent_index("b")
    n = 100
    payload = b"x" * 64
    recs = np.zeros(n, dtype=REC_DTYPE)
    recs["sender"] = a
    recs["receiver"] = b
    recs["type"] = 0
    recs["priority"] = 1
    recs["timestamp"] = time.time()
    recs["payload_off"] = np.arange(n, dtype=np.uint64) * len(payload)
    recs["payload_len"] = len(payload)
    recs["content_len"] = len(payload)
    recs["vis_mode"] = VIS_ALL
    recs["bitmap"] = NO_BITMAP
    seqs = tmp_db.send_batch(recs, payload * n)
    assert len(seqs) == n
    counts, got = tmp_db.receive_batch(np.array([b]), max_per_agent=1000)
    assert counts[0] == n
    assert set(got.tolist()) == set(seqs.tolist())
    # derived ids resolve through get_message
    m = tmp_db.get_message(tmp_db.get_stats() and
                           __import__("swarmdb_amd.core.wire", fromlist=["derived_id"]).derived_id(0, int(seqs[0])))
    assert m is not None
    assert m.content == "x" * 64


def test_resend_failed(tmp_db, monkeypatch):
    calls = {"n": 0}
    orig = tmp_db.engine.enqueue_batch

    def flaky(recs, payloads):
        if calls["n"] == 0:
            calls["n"] += 1
            raise RuntimeError("ring full")
        return orig(recs, payloads)

    monkeypatch.setattr(tmp_db.engine, "enqueue_batch", flaky)
    with pytest.raises(RuntimeError):
        tmp_db.send_message("a", "will fail", receiver_id="b")
    new_ids = tmp_db.resend_failed_messages()
    assert len(new_ids) == 1
    m = tmp_db.get_message(new_ids[0])
    assert m.content == "will fail"
    assert "resent_from" in m.metadata
    # second resend is a no-op
    assert tmp_db.resend_failed_messages() == []


def test_send_to_group_fast(tmp_db):
    tmp_db.add_agent_group("team", ["a", "b", "c", "d"])
    mid = tmp_db.send_to_group_fast("team", "a", "one slot fanout",
                                    priority=MessagePriority.HIGH)
    # one message id; delivered to members only, not the sender
    for member in ["b", "c", "d"]:
        got = tmp_db.receive_messages(member, timeout=0)
        assert [m.id for m in got] == [mid]
        assert got[0].metadata["group"] == "team"
        assert got[0].priority == MessagePriority.HIGH
    assert tmp_db.receive_messages("a", timeout=0) == []
    # non-members never see it in their inbox (group fan-out filters at
    # append, unlike broadcast)
    tmp_db.register_agent("outsider")
    assert tmp_db.get_agent_messages("outsider") == []
    # member inboxes list it
    assert [m.id for m in tmp_db.get_agent_messages("b")] == [mid]
    with pytest.raises(ValueError):
        tmp_db.send_to_group_fast("nope", "a", "x")


def test_error_lane_malformed_batch(tmp_db):
    """Malformed batch records park as FAILED (the error-lane analog of
    the reference's _errors topic), valid ones deliver normally."""
    from swarmdb_amd.runtime.engine import (
        BROADCAST, NO_BITMAP, REC_DTYPE, ST_FAILED, VIS_ALL,
    )

    a = tmp_db.agent_index("a")
    b = tmp_db.agent_index("b")
    recs = np.zeros(4, dtype=REC_DTYPE)
    recs["sender"] = a
    recs["receiver"] = b
    recs["vis_mode"] = VIS_ALL
    recs["bitmap"] = NO_BITMAP
    recs["payload_len"] = 16
    recs["content_len"] = 16
    recs["payload_off"] = np.arange(4, dtype=np.uint64) * 16
    recs["receiver"][1] = 999999        # bad receiver (not BROADCAST)
    recs["type"][2] = 200               # bad type
    recs["payload_len"][3] = 1 << 24    # oversize payload
    seqs = tmp_db.send_batch(recs, b"y" * 64)
    st = [tmp_db.engine.get_status(int(s)) for s in seqs]
    assert st[0] != ST_FAILED
    assert st[1] == ST_FAILED and st[2] == ST_FAILED and st[3] == ST_FAILED
    # only the valid one delivers
    counts, got = tmp_db.receive_batch(np.array([b]), max_per_agent=10)
    assert counts[0] == 1 and got[0] == seqs[0]
    # failed ones are queryable by status
    failed = tmp_db.engine.query(status=ST_FAILED, limit=10)
    assert set(failed.tolist()) == set(seqs[1:].tolist())
    s = tmp_db.get_stats()
    assert s["messages_by_status"].get("failed") == 3


def test_oversized_content_overflow_store(tmp_db):
    """Content larger than a device slot routes normally; the payload
    lives in the host-side overflow store (SURVEY.md §7 hard part 2)."""
    big = "Z" * (tmp_db.config.slot_bytes * 3)
    mid = tmp_db.send_message("a", big, receiver_id="b",
                              metadata={"note": "huge"})
    got = tmp_db.receive_messages("b", timeout=0)
    assert len(got) == 1
    assert got[0].id == mid
    assert got[0].content == big
    assert got[0].metadata == {"note": "huge"}
    # round-trips through get_message and history save
    assert tmp_db.get_message(mid).content == big
    path = tmp_db.save_message_history()
    data = json.loads(open(path).read())
    assert data["messages"][mid]["content"] == big
    # dict/list content too
    big_dict = {"payload": ["x" * 1000] * 10}
    mid2 = tmp_db.send_message("a", big_dict, receiver_id="b")
    assert tmp_db.receive_messages("b", timeout=0)[0].content == big_dict
    assert tmp_db.get_message(mid2).content == big_dict


def test_receive_timeout_waits_for_message(tmp_db):
    """receive_messages(timeout>0) polls until a message arrives
    (reference consumer-poll semantics, swarmdb/ main.py:553-558)."""
    import threading
    import time as _t

    tmp_db.register_agent("waiter")

    def late_send():
        _t.sleep(0.15)
        tmp_db.send_message("someone", "late delivery", receiver_id="waiter")

    t = threading.Thread(target=late_send)
    start = _t.monotonic()
    t.start()
    msgs = tmp_db.receive_messages("waiter", timeout=2.0)
    elapsed = _t.monotonic() - start
    t.join()
    assert len(msgs) == 1
    assert msgs[0].content == "late delivery"
    assert 0.1 < elapsed < 1.5  # returned as soon as it arrived
    # and an empty timeout expires without messages
    start = _t.monotonic()
    assert tmp_db.receive_messages("waiter", timeout=0.1) == []
    assert _t.monotonic() - start >= 0.09


def test_derived_id_from_other_rank_not_found(tmp_db):
    from swarmdb_amd.core.wire import derived_id

    tmp_db.send_message("a", "x", receiver_id="b")
    # rank 7's derived id never resolves on rank 0
    assert tmp_db.get_message(derived_id(7, 0)) is None
    # out-of-range seq on our rank doesn't resolve either
    assert tmp_db.get_message(derived_id(0, 999999)) is None


def test_unregistered_agent_receive_empty(tmp_db):
    # engine-level receive for an index that never registered
    assert len(tmp_db.engine.receive(37, 10)) == 0
    assert tmp_db.engine.unread_count(37) == 0
    assert len(tmp_db.engine.peek_inbox(37)) == 0


def test_compat_host_maps_prune_with_retention(tmp_path):
    """_id_to_seq/_overflow must not outlive the engine's retention
    horizon (round-1 review: unbounded host maps leak RAM while the
    device ring evicts)."""
    cfg = QueueConfig(use_gpu=False, save_dir=str(tmp_path), auto_save=False,
                      max_agents=64)
    db = SwarmsDB(config=cfg)
    db._PRUNE_EVERY = 8  # tight cadence for the test
    ids = [db.send_message("a", f"m{i}", receiver_id="b") for i in range(32)]
    assert len(db._id_to_seq) == 32
    # simulate the device ring evicting the first 16 seqs
    db.engine.evict_base = lambda: 16
    for i in range(db._PRUNE_EVERY + 1):
        db.send_message("a", f"late{i}", receiver_id="b")
    live = set(db._id_to_seq.values())
    assert all(s >= 16 for s in live)
    # ids below the horizon are gone; newer ones survive
    assert ids[0] not in db._id_to_seq
    assert db._id_to_seq.get(ids[-1]) is not None
    db.close()


def test_express_lane_cpu_double(tmp_path):
    """Express-lane API contract on the CPU double (the GPU path is the
    doorbell persistent kernel, covered by the gpu-marked test)."""
    cfg = QueueConfig(use_gpu=False, save_dir=str(tmp_path), auto_save=False,
                      max_agents=64)
    db = SwarmsDB(config=cfg)
    db.express_start(["a", "b", "c"])
    assert db.express_recv("b", timeout_us=0) is None
    db.express_send("a", "b", "fast one")
    db.express_send("c", "b", b"\x00binary\xff")
    s1 = db.express_recv("b")
    s2 = db.express_recv("b")
    assert s1 == ("a", b"fast one")
    assert s2 == ("c", b"\x00binary\xff")
    with pytest.raises(RuntimeError):
        db.express_start(["x"])  # already running
    db.express_stop()
    with pytest.raises(RuntimeError):
        db.express_send("a", "b", "nope")
    db.close()


def test_get_conversation_sort_flag(tmp_path):
    cfg = QueueConfig(use_gpu=False, save_dir=str(tmp_path),
                      auto_save=False, max_agents=64)
    db = SwarmsDB(config=cfg)
    db.send_message("a", "1", receiver_id="b")
    time.sleep(0.002)
    db.send_message("b", "2", receiver_id="a")
    time.sleep(0.002)
    db.send_message("a", "3", receiver_id="b")
    # default keeps the reference's unmerged concatenation
    plain = [m.content for m in db.get_conversation("a", "b", limit=10)]
    assert plain == ["3", "1", "2"]
    # sort=True interleaves chronologically
    merged = [m.content for m in db.get_conversation("a", "b", limit=10,
                                                     sort=True)]
    assert merged == ["1", "2", "3"]
    db.close()
