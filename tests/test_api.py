"""API-layer tests: all 19 reference endpoints + auth/permission matrix
 against the CPU engine via FastAPI TestClient."""

import json
from pathlib import Path

import pytest
from fastapi.testclient import TestClient

from swarmdb_amd import QueueConfig, SwarmsDB
from swarmdb_amd.api.app import ApiSettings, create_app


@pytest.fixture()
def client(tmp_path):
    cfg = QueueConfig(use_gpu=False, save_dir=str(tmp_path / "hist"),
                      max_agents=256)
    db = SwarmsDB(config=cfg)
    app = create_app(db=db, settings=ApiSettings())
    with TestClient(app) as c:
        c.db = db
        yield c
    db.config.auto_save = False


def token(client, username):
    r = client.post("/auth/token",
                    json={"username": username, "password": "pw"})
    assert r.status_code == 200, r.text
    body = r.json()
    assert body["token_type"] == "bearer"
    return body["access_token"]


def auth(client, username):
    return {"Authorization": f"Bearer {token(client, username)}"}


def test_token_requires_nonempty_credentials(client):
    r = client.post("/auth/token", json={"username": "", "password": "x"})
    assert r.status_code == 400
    r = client.post("/auth/token", json={"username": "a", "password": ""})
    assert r.status_code == 400
    # any non-empty pair accepted (reference api.py:373-380)
    r = client.post("/auth/token", json={"username": "anyone", "password": "pw"})
    assert r.status_code == 200


def test_endpoints_require_auth(client):
    for method, path in [
        ("post", "/agents/register"),
        ("delete", "/agents/x"),
        ("post", "/messages"),
        ("post", "/messages/broadcast"),
        ("get", "/messages/xyz"),
        ("get", "/messages"),
        ("get", "/agents/x/messages"),
        ("post", "/agents/receive"),
        ("put", "/messages/x/status?status=read"),
        ("post", "/groups"),
        ("post", "/groups/message"),
        ("get", "/stats"),
        ("post", "/admin/save"),
        ("post", "/admin/flush"),
        ("post", "/admin/resend_failed"),
        ("post", "/admin/scale_partitions"),
    ]:
        r = getattr(client, method)(path)
        assert r.status_code == 401, f"{method} {path} -> {r.status_code}"


def test_invalid_token_rejected(client):
    r = client.get("/messages", headers={"Authorization": "Bearer garbage"})
    assert r.status_code == 401


def test_register_self_and_admin(client):
    h = auth(client, "agent1")
    r = client.post("/agents/register", headers=h,
                    json={"agent_id": "agent1", "description": "test bot",
                          "capabilities": ["chat"]})
    assert r.status_code == 201
    assert r.json() == {"status": "registered", "agent_id": "agent1"}
    # non-admin cannot register someone else
    r = client.post("/agents/register", headers=h, json={"agent_id": "other"})
    assert r.status_code == 403
    # admin can
    r = client.post("/agents/register", headers=auth(client, "admin"),
                    json={"agent_id": "other"})
    assert r.status_code == 201
    # metadata stashed
    assert client.db.agent_metadata["agent1"]["description"] == "test bot"


def test_deregister(client):
    h = auth(client, "agent1")
    client.post("/agents/register", headers=h, json={"agent_id": "agent1"})
    r = client.delete("/agents/agent1", headers=h)
    assert r.status_code == 200
    r = client.delete("/agents/agent1", headers=h)
    assert r.status_code == 404
    r = client.delete("/agents/someone", headers=h)
    assert r.status_code == 403


def test_send_and_get_message(client):
    h = auth(client, "alice")
    r = client.post("/messages", headers=h,
                    json={"receiver_id": "bob", "content": "hi bob",
                          "message_type": "chat", "priority": 2})
    assert r.status_code == 200
    body = r.json()
    assert body["sender_id"] == "alice"
    assert body["receiver_id"] == "bob"
    assert body["content"] == "hi bob"
    assert body["type"] == "chat"
    assert body["priority"] == 2
    assert body["status"] == "delivered"
    mid = body["id"]

    # sender can read it
    assert client.get(f"/messages/{mid}", headers=h).status_code == 200
    # receiver can read it
    hb = auth(client, "bob")
    assert client.get(f"/messages/{mid}", headers=hb).status_code == 200
    # a third party cannot
    hc = auth(client, "carol")
    assert client.get(f"/messages/{mid}", headers=hc).status_code == 403
    # admin can
    assert client.get(f"/messages/{mid}",
                      headers=auth(client, "admin")).status_code == 200
    # 404 for unknown id
    assert client.get("/messages/nope", headers=h).status_code == 404


def test_receive_flow(client):
    ha, hb = auth(client, "alice"), auth(client, "bob")
    client.post("/messages", headers=ha,
                json={"receiver_id": "bob", "content": "one"})
    client.post("/messages", headers=ha,
                json={"receiver_id": "bob", "content": {"n": 2}})
    r = client.post("/agents/receive?timeout=0", headers=hb)
    assert r.status_code == 200
    msgs = r.json()
    assert [m["content"] for m in msgs] == ["one", {"n": 2}]
    assert all(m["status"] == "read" for m in msgs)
    # drained
    assert client.post("/agents/receive?timeout=0", headers=hb).json() == []


def test_broadcast_endpoint(client):
    for a in ["a", "b", "c"]:
        client.post("/agents/register", headers=auth(client, a),
                    json={"agent_id": a})
    r = client.post("/messages/broadcast", headers=auth(client, "a"),
                    json={"content": "all hands", "exclude_agents": ["c"]})
    assert r.status_code == 200
    body = r.json()
    assert body["status"] == "broadcast"
    mid = body["message_id"]
    got_b = client.post("/agents/receive?timeout=0", headers=auth(client, "b")).json()
    assert [m["id"] for m in got_b] == [mid]
    assert client.post("/agents/receive?timeout=0",
                       headers=auth(client, "c")).json() == []


def test_query_scope_enforcement(client):
    ha, hb = auth(client, "alice"), auth(client, "bob")
    client.post("/messages", headers=ha,
                json={"receiver_id": "bob", "content": "x"})
    client.post("/messages", headers=hb,
                json={"receiver_id": "alice", "content": "y"})
    # non-admin with no filter defaults to own sent messages
    r = client.get("/messages", headers=ha)
    assert r.status_code == 200
    assert [m["content"] for m in r.json()] == ["x"]
    # non-admin cannot query someone else's traffic
    r = client.get("/messages?sender_id=bob", headers=ha)
    assert r.status_code == 403
    # receiver scope allowed
    r = client.get("/messages?receiver_id=alice", headers=ha)
    assert [m["content"] for m in r.json()] == ["y"]
    # admin sees all
    r = client.get("/messages", headers=auth(client, "admin"))
    assert len(r.json()) == 2
    # filters pass through
    r = client.get("/messages?status=delivered&limit=1",
                   headers=auth(client, "admin"))
    assert len(r.json()) == 1


def test_agent_messages_endpoint(client):
    ha, hb = auth(client, "alice"), auth(client, "bob")
    for i in range(3):
        client.post("/messages", headers=ha,
                    json={"receiver_id": "bob", "content": f"m{i}"})
    r = client.get("/agents/bob/messages", headers=hb)
    assert [m["content"] for m in r.json()] == ["m2", "m1", "m0"]
    r = client.get("/agents/bob/messages?limit=1&skip=1", headers=hb)
    assert [m["content"] for m in r.json()] == ["m1"]
    assert client.get("/agents/bob/messages", headers=ha).status_code == 403
    assert client.get("/agents/bob/messages",
                      headers=auth(client, "admin")).status_code == 200


def test_update_status_permissions(client):
    ha, hb = auth(client, "alice"), auth(client, "bob")
    mid = client.post("/messages", headers=ha,
                      json={"receiver_id": "bob", "content": "x"}).json()["id"]
    # sender (not receiver) cannot update
    r = client.put(f"/messages/{mid}/status?status=read", headers=ha)
    assert r.status_code == 403
    r = client.put(f"/messages/{mid}/status?status=processed", headers=hb)
    assert r.status_code == 200
    assert client.get(f"/messages/{mid}", headers=hb).json()["status"] == "processed"
    assert client.put("/messages/nope/status?status=read",
                      headers=hb).status_code == 404


def test_group_flow(client):
    h = auth(client, "lead")
    r = client.post("/groups", headers=h,
                    json={"group_name": "team",
                          "agent_ids": ["lead", "m1", "m2"]})
    assert r.status_code == 201
    r = client.post("/groups/message", headers=h,
                    json={"group_name": "team", "content": "standup"})
    assert r.status_code == 200
    body = r.json()
    assert body["status"] == "sent"
    assert len(body["message_ids"]) == 2
    r = client.post("/groups/message", headers=h,
                    json={"group_name": "ghost", "content": "x"})
    assert r.status_code == 404


def test_health_no_auth(client):
    r = client.get("/health")
    assert r.status_code == 200
    body = r.json()
    assert body["status"] == "healthy"
    assert body["engine"] == "CpuEngine"
    assert body["engine_connected"] is True


def test_stats_admin_only(client):
    assert client.get("/stats", headers=auth(client, "pleb")).status_code == 403
    client.post("/messages", headers=auth(client, "a"),
                json={"receiver_id": "b", "content": "x"})
    r = client.get("/stats", headers=auth(client, "admin"))
    assert r.status_code == 200
    assert r.json()["total_messages"] == 1


def test_admin_save_flush_resend_scale(client):
    ha = auth(client, "admin")
    hx = auth(client, "pleb")
    for path in ["/admin/save", "/admin/flush", "/admin/resend_failed",
                 "/admin/scale_partitions"]:
        assert client.post(path, headers=hx).status_code == 403
    client.post("/messages", headers=auth(client, "a"),
                json={"receiver_id": "b", "content": "x"})
    r = client.post("/admin/save", headers=ha)
    assert r.status_code == 200
    assert Path(r.json()["path"]).exists()
    r = client.post("/admin/flush?older_than=0.0", headers=ha)
    assert r.status_code == 200
    assert r.json()["messages_flushed"] == 1
    r = client.post("/admin/resend_failed", headers=ha)
    assert r.json() == {"status": "resent", "message_ids": []}
    r = client.post("/admin/scale_partitions", headers=ha)
    assert r.status_code == 200
    assert "current_partitions" in r.json()


def test_admin_load_endpoint(client):
    ha = auth(client, "admin")
    client.post("/messages", headers=auth(client, "a"),
                json={"receiver_id": "b", "content": "persist me"})
    path = client.post("/admin/save", headers=ha).json()["path"]
    r = client.post(f"/admin/load?path={path}", headers=ha)
    assert r.status_code == 200
    # paths outside the save dir are rejected (no arbitrary-file ingest)
    r = client.post("/admin/load?path=/etc/passwd", headers=ha)
    assert r.status_code == 403
    # missing files *under* the save dir are a plain 404
    missing = str(client.db.save_dir / "nonexistent.json")
    r = client.post(f"/admin/load?path={missing}", headers=ha)
    assert r.status_code == 404


def test_llm_routes(client):
    ha = auth(client, "admin")
    r = client.post("/llm/dispatch", headers=ha)
    assert r.status_code == 503  # no backends yet
    for b in ["gpu0", "gpu1"]:
        assert client.post(f"/llm/backends/{b}", headers=ha).status_code == 200
    client.db.set_llm_load_balancing(True)
    picks = [client.post("/llm/dispatch", headers=ha).json()["backend_id"]
             for _ in range(4)]
    assert sorted(set(picks)) == ["gpu0", "gpu1"]
    r = client.post("/llm/complete/gpu0", headers=ha)
    assert r.status_code == 200


def test_load_and_unread_routes(client):
    ha, hb = auth(client, "alice"), auth(client, "bob")
    client.post("/messages", headers=ha,
                json={"receiver_id": "bob", "content": "x"})
    r = client.get("/agents/bob/unread_count", headers=hb)
    assert r.json()["unread_count"] == 1
    r = client.get("/agents/bob/load", headers=hb)
    assert r.json()["inbox_size"] == 1
    assert client.get("/agents/bob/load", headers=ha).status_code == 403


def test_rate_limiter():
    import swarmdb_amd.api.app as appmod

    rl = appmod.RateLimiter(limit_per_minute=3)
    assert all(rl.allow("1.2.3.4") for _ in range(3))
    assert not rl.allow("1.2.3.4")
    assert rl.allow("5.6.7.8")  # other IPs unaffected


def test_metrics_endpoint(client):
    client.post("/messages", headers=auth(client, "a"),
                json={"receiver_id": "b", "content": "x"})
    r = client.get("/metrics")
    assert r.status_code == 200
    body = r.text
    assert "swarmdb_messages_total 1.0" in body
    assert 'swarmdb_messages_by_status{status="delivered"} 1.0' in body
    assert "swarmdb_registered_agents 2.0" in body


def test_trace_endpoint(client):
    ha = auth(client, "admin")
    assert client.get("/admin/trace", headers=auth(client, "x")).status_code == 403
    r = client.get("/admin/trace?enable=true", headers=ha)
    assert r.json()["enabled"] is True
    import numpy as np

    from swarmdb_amd.runtime.engine import NO_BITMAP, REC_DTYPE, VIS_ALL

    a = client.db.agent_index("a")
    recs = np.zeros(4, dtype=REC_DTYPE)
    recs["sender"] = a
    recs["receiver"] = a
    recs["payload_len"] = 8
    recs["content_len"] = 8
    recs["payload_off"] = np.arange(4, dtype=np.uint64) * 8
    recs["vis_mode"] = VIS_ALL
    recs["bitmap"] = NO_BITMAP
    client.db.send_batch(recs, b"x" * 32)
    r = client.get("/admin/trace", headers=ha)
    assert "send_batch" in r.json()["ops"]
    r = client.get("/admin/trace?enable=false", headers=ha)
    assert r.json()["enabled"] is False


def test_batch_send_endpoint(client):
    ha, hb = auth(client, "alice"), auth(client, "bob")
    client.post("/agents/register", headers=hb, json={"agent_id": "bob"})
    msgs = [{"receiver_id": "bob", "content": f"bulk {i}"} for i in range(5)]
    msgs.append({"receiver_id": None, "content": {"kind": "announce"}})
    r = client.post("/messages/batch", headers=ha, json=msgs)
    assert r.status_code == 200, r.text
    ids = r.json()["message_ids"]
    assert len(ids) == 6
    got = client.post("/agents/receive?timeout=0&max_messages=100",
                      headers=hb).json()
    # 5 p2p + 1 broadcast
    assert len(got) == 6
    assert got[-1]["content"] == {"kind": "announce"}
    # derived ids resolve through GET
    r = client.get(f"/messages/{ids[0]}", headers=ha)
    assert r.status_code == 200
    assert r.json()["content"] == "bulk 0"
    # empty batch
    r = client.post("/messages/batch", headers=ha, json=[])
    assert r.json()["message_ids"] == []


def test_search_endpoint(client):
    ha, hb, hc = auth(client, "alice"), auth(client, "bob"), auth(client, "carol")
    client.post("/messages", headers=ha,
                json={"receiver_id": "bob", "content": "wombat sighting"})
    client.post("/messages", headers=hc,
                json={"receiver_id": "dave", "content": "wombat relocation"})
    # non-admin sees only own traffic
    r = client.get("/messages/search/?keyword=wombat", headers=hb)
    assert [m["content"] for m in r.json()] == ["wombat sighting"]
    # admin sees all
    r = client.get("/messages/search/?keyword=wombat",
                   headers=auth(client, "admin"))
    assert len(r.json()) == 2
    # no auth
    assert client.get("/messages/search/?keyword=x").status_code == 401


def test_micro_batched_sends(tmp_path):
    """With a batch window, concurrent sends coalesce into one engine
    batch; responses stay per-request correct."""
    import threading

    cfg = QueueConfig(use_gpu=False, save_dir=str(tmp_path / "h"),
                      max_agents=256, auto_save=False)
    db = SwarmsDB(config=cfg)
    calls = {"n": 0}
    orig = db.engine.enqueue_batch

    def counting(recs, payloads):
        calls["n"] += 1
        return orig(recs, payloads)

    db.engine.enqueue_batch = counting
    app = create_app(db=db, settings=ApiSettings(), batch_window=0.01)
    with TestClient(app) as c:
        tok = c.post("/auth/token",
                     json={"username": "alice", "password": "x"}).json()
        h = {"Authorization": f"Bearer {tok['access_token']}"}
        results = []

        def one(i):
            r = c.post("/messages", headers=h,
                       json={"receiver_id": "bob", "content": f"m{i}"})
            results.append(r)

        threads = [threading.Thread(target=one, args=(i,)) for i in range(12)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        assert all(r.status_code == 200 for r in results)
        ids = {r.json()["id"] for r in results}
        assert len(ids) == 12
        # engine batches << request count (coalescing happened)
        assert calls["n"] < 12, calls["n"]
        tok2 = c.post("/auth/token",
                      json={"username": "bob", "password": "x"}).json()
        h2 = {"Authorization": f"Bearer {tok2['access_token']}"}
        got = c.post("/agents/receive?timeout=0&max_messages=100",
                     headers=h2).json()
        assert {m["id"] for m in got} == ids
    db.config.auto_save = False


def test_batch_send_respects_visible_to(client):
    """A batch item restricted via visible_to must not reach agents
    outside the list (round-1 advisor finding: the fast path hardcoded
    VIS_ALL and leaked restricted broadcasts to everyone)."""
    ha = auth(client, "alice")
    hb = auth(client, "bob")
    he = auth(client, "eve")
    for name, h in [("bob", hb), ("eve", he)]:
        client.post("/agents/register", headers=h, json={"agent_id": name})
    msgs = [
        {"receiver_id": None, "content": "secret", "visible_to": ["bob"]},
        {"receiver_id": "bob", "content": "tagged",
         "metadata": {"k": "v"}},
    ]
    r = client.post("/messages/batch", headers=ha, json=msgs)
    assert r.status_code == 200, r.text
    got_bob = client.post(
        "/agents/receive?timeout=0&max_messages=100", headers=hb
    ).json()
    got_eve = client.post(
        "/agents/receive?timeout=0&max_messages=100", headers=he
    ).json()
    assert sorted(m["content"] for m in got_bob) == ["secret", "tagged"]
    assert [m["content"] for m in got_eve] == []  # eve never sees it
    tagged = [m for m in got_bob if m["content"] == "tagged"][0]
    assert tagged["metadata"] == {"k": "v"}


def test_admin_login_blocked_outside_development(tmp_path, monkeypatch):
    """Open demo auth must not mint admin outside development mode when
    no credential validator is installed (round-1 advisor finding)."""
    monkeypatch.setenv("API_ENV", "production")
    cfg = QueueConfig(use_gpu=False, save_dir=str(tmp_path / "hist"),
                      max_agents=64, auto_save=False)
    db = SwarmsDB(config=cfg)
    app = create_app(db=db, settings=ApiSettings())
    with TestClient(app) as c:
        r = c.post("/auth/token",
                   json={"username": "admin", "password": "pw"})
        assert r.status_code == 403
        # non-admin demo logins still work (documented demo behavior)
        r = c.post("/auth/token",
                   json={"username": "alice", "password": "pw"})
        assert r.status_code == 200
    # with a validator, admin can authenticate in production
    app2 = create_app(
        db=db,
        settings=ApiSettings(),
        credential_validator=lambda u, p: p == "s3cret",
    )
    with TestClient(app2) as c:
        r = c.post("/auth/token",
                   json={"username": "admin", "password": "s3cret"})
        assert r.status_code == 200
        r = c.post("/auth/token",
                   json={"username": "admin", "password": "wrong"})
        assert r.status_code == 401


def test_agent_messages_pagination_matches_reference_order(client):
    """skip applies to raw newest-first inbox entries BEFORE the status
    filter (reference main.py:640-652)."""
    ha, hb = auth(client, "a"), auth(client, "b")
    for i in range(5):
        client.post("/messages", headers=ha,
                    json={"receiver_id": "b", "content": f"m{i}"})
    # read everything, then mark m4 (newest) processed
    got = client.post("/agents/receive?timeout=0&max_messages=100",
                      headers=hb).json()
    assert len(got) == 5
    newest = [m for m in got if m["content"] == "m4"][0]
    client.put(f"/messages/{newest['id']}/status?status=processed",
               headers=hb)
    # newest-first raw order: m4(processed), m3..m0(read)
    # skip=1 drops m4 BEFORE filtering; status=read then yields m3..m0
    r = client.get("/agents/b/messages?status=read&skip=1&limit=10",
                   headers=hb)
    assert [m["content"] for m in r.json()] == ["m3", "m2", "m1", "m0"]
    # skip=0 with status=processed yields only m4
    r = client.get("/agents/b/messages?status=processed&skip=0&limit=10",
                   headers=hb)
    assert [m["content"] for m in r.json()] == ["m4"]
    # skip=1 with status=processed: m4 is skipped as a RAW entry -> empty
    r = client.get("/agents/b/messages?status=processed&skip=1&limit=10",
                   headers=hb)
    assert r.json() == []


def test_admin_checkpoint_routes(client):
    ha = auth(client, "admin")
    client.post("/messages", headers=auth(client, "a"),
                json={"receiver_id": "b", "content": "ckpt me"})
    # no base yet -> delta conflicts
    r = client.post("/admin/checkpoint/delta", headers=ha)
    assert r.status_code == 409
    r = client.post("/admin/checkpoint", headers=ha)
    assert r.status_code == 200
    path = r.json()["path"]
    client.post("/messages", headers=auth(client, "a"),
                json={"receiver_id": "b", "content": "delta me"})
    r = client.post("/admin/checkpoint/delta", headers=ha)
    assert r.status_code == 200 and r.json()["messages"] == 1
    r = client.post(f"/admin/checkpoint/load?path={path}", headers=ha)
    assert r.status_code == 200 and r.json()["messages"] == 2
    # path constraint mirrors /admin/load
    r = client.post("/admin/checkpoint/load?path=/etc/passwd", headers=ha)
    assert r.status_code == 403
    # non-admin rejected
    r = client.post("/admin/checkpoint", headers=auth(client, "a"))
    assert r.status_code == 403


def test_search_route_scoping(client):
    """/messages/search/: device-scan route; non-admins only see their
    own traffic, admins see all."""
    ha, hb = auth(client, "alice"), auth(client, "bob")
    hc = auth(client, "carol")
    client.post("/messages", headers=ha,
                json={"receiver_id": "bob", "content": "zebra sighting"})
    client.post("/messages", headers=hb,
                json={"receiver_id": "alice", "content": "no zebras here"})
    client.post("/messages", headers=hc,
                json={"receiver_id": "carol2", "content": "zebra private"})
    # participants see their own zebra traffic
    r = client.get("/messages/search/?keyword=zebra", headers=ha)
    assert r.status_code == 200
    assert sorted(m["content"] for m in r.json()) == [
        "no zebras here", "zebra sighting"]
    # case-insensitive by default
    r = client.get("/messages/search/?keyword=ZEBRA", headers=ha)
    assert len(r.json()) == 2
    # case-sensitive opt-in
    r = client.get("/messages/search/?keyword=ZEBRA&case_sensitive=true",
                   headers=ha)
    assert r.json() == []
    # admin sees everything
    r = client.get("/messages/search/?keyword=zebra",
                   headers=auth(client, "admin"))
    assert len(r.json()) == 3
    # carol's private message is invisible to alice
    assert all("private" not in m["content"]
               for m in client.get("/messages/search/?keyword=zebra",
                                   headers=ha).json())


def test_export_yaml_route(client):
    ha = auth(client, "admin")
    client.post("/messages", headers=auth(client, "a"),
                json={"receiver_id": "b", "content": "yaml via rest"})
    r = client.post("/admin/export_yaml", headers=ha)
    assert r.status_code == 200
    import yaml as _yaml
    data = _yaml.safe_load(Path(r.json()["path"]).read_text())
    contents = [m["content"] for m in data["messages"].values()]
    assert "yaml via rest" in contents
    # non-admin rejected
    r = client.post("/admin/export_yaml", headers=auth(client, "a"))
    assert r.status_code == 403
