"""Host-side thread-safety stress (the reference mutated
all shared dicts from 4 threads per worker without locks; our facade and
CPU engine take explicit locks — prove no lost updates or crashes under
contention)."""

import threading

import numpy as np

from swarmdb_amd import QueueConfig, SwarmsDB


def test_concurrent_senders_receivers(tmp_path):
    cfg = QueueConfig(use_gpu=False, auto_save=False, max_agents=128,
                      save_dir=str(tmp_path))
    db = SwarmsDB(config=cfg)
    n_threads = 8
    per_thread = 50
    errors = []
    received = []
    rlock = threading.Lock()

    def sender(tid):
        try:
            for i in range(per_thread):
                db.send_message(f"s{tid}", f"msg {tid}/{i}",
                                receiver_id=f"r{tid % 4}")
        except Exception as e:  # pragma: no cover
            errors.append(e)

    def receiver(rid):
        try:
            got = 0
            for _ in range(200):
                msgs = db.receive_messages(f"r{rid}", max_messages=50,
                                           timeout=0.01)
                got += len(msgs)
                if got and not msgs:
                    break
            with rlock:
                received.append(got)
        except Exception as e:  # pragma: no cover
            errors.append(e)

    for rid in range(4):
        db.register_agent(f"r{rid}")
    senders = [threading.Thread(target=sender, args=(t,))
               for t in range(n_threads)]
    for t in senders:
        t.start()
    receivers = [threading.Thread(target=receiver, args=(r,))
                 for r in range(4)]
    for t in receivers:
        t.start()
    for t in senders + receivers:
        t.join(timeout=60)
    assert not errors, errors

    total_sent = n_threads * per_thread
    # drain anything receivers missed after senders finished
    tail = sum(
        len(db.receive_messages(f"r{r}", max_messages=10000, timeout=0))
        for r in range(4)
    )
    assert sum(received) + tail == total_sent
    stats = db.get_stats()
    assert stats["total_messages"] == total_sent
    by_status = stats["messages_by_status"]
    assert sum(by_status.values()) == total_sent
    db.close()


def test_concurrent_registry_and_stats(tmp_path):
    cfg = QueueConfig(use_gpu=False, auto_save=False, max_agents=512,
                      save_dir=str(tmp_path))
    db = SwarmsDB(config=cfg)
    errors = []

    def worker(tid):
        try:
            for i in range(40):
                db.register_agent(f"agent{tid}_{i}")
                db.send_message(f"agent{tid}_{i}", "x",
                                receiver_id=f"agent{tid}_{i}")
                db.get_stats()
                db.get_agent_load(f"agent{tid}_{i}")
        except Exception as e:  # pragma: no cover
            errors.append(e)

    threads = [threading.Thread(target=worker, args=(t,)) for t in range(6)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    assert not errors, errors
    assert len(db.registered_agents) == 6 * 40
    assert db.get_stats()["total_messages"] == 6 * 40
    db.close()
