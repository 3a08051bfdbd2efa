"""Client SDK round trip against the in-process app (httpx transport)."""

import pytest
from fastapi.testclient import TestClient

from swarmdb_amd import QueueConfig, SwarmsDB
from swarmdb_amd.api.app import ApiSettings, create_app
from swarmdb_amd.client import SwarmDBClient


@pytest.fixture()
def served_client_factory(tmp_path):
    cfg = QueueConfig(use_gpu=False, save_dir=str(tmp_path / "h"),
                      max_agents=128, auto_save=False)
    db = SwarmsDB(config=cfg)
    app = create_app(db=db, settings=ApiSettings())

    made = []

    def make(agent_id: str) -> SwarmDBClient:
        c = SwarmDBClient("http://testserver", agent_id=agent_id)
        # route the SDK through the in-process app (TestClient is an
        # httpx.Client against the ASGI app)
        c._http = TestClient(app)
        made.append(c)
        return c

    yield make
    for c in made:
        c.close()
    db.config.auto_save = False


def test_client_end_to_end(served_client_factory):
    alice = served_client_factory("alice")
    bob = served_client_factory("bob")

    assert alice.health()["status"] == "healthy"
    alice.register(description="sender bot")
    bob.register()

    mid = alice.send("bob", "hello bob", priority=2)
    got = bob.receive(timeout=0)
    assert [m["id"] for m in got] == [mid]
    assert got[0]["content"] == "hello bob"
    assert got[0]["priority"] == 2

    ids = alice.send_batch(
        [{"receiver_id": "bob", "content": f"bulk {i}"} for i in range(4)]
    )
    assert len(ids) == 4
    assert len(bob.receive(timeout=0)) == 4

    bid = alice.broadcast({"note": "all"})
    assert bob.receive(timeout=0)[0]["id"] == bid

    assert alice.get_message(mid)["content"] == "hello bob"
    assert bob.unread_count() == 0
    assert bob.load()["inbox_size"] == 6

    hits = alice.search("hello")
    assert [m["id"] for m in hits] == [mid]

    alice.create_group("duo", ["alice", "bob"])
    gids = alice.send_to_group("duo", "group ping")
    assert len(gids) == 1
    assert bob.mark_processed(gids[0])["new_status"] == "processed"

    mine = alice.my_messages()
    # the inbox listing shows alice's own broadcast (reference behavior,
    # SURVEY.md §8.11) even though receive() filters it
    assert [m["id"] for m in mine] == [bid]
    q = alice.query(sender_id="alice", limit=3)
    assert len(q) == 3


def test_client_admin_checkpoint(served_client_factory):
    admin = served_client_factory("admin")
    alice = served_client_factory("alice")
    alice.register()
    alice.send("admin", "persist me")
    out = admin.admin_checkpoint()
    assert out["status"] == "checkpointed"
    alice.send("admin", "delta me")
    assert admin.admin_checkpoint_delta()["messages"] == 1
    loaded = admin.admin_checkpoint_load(out["path"])
    assert loaded["messages"] == 2
