import sys, struct, time
sys.path.insert(0, "/root/repo")
from swarmdb_amd import _swarmq

db = _swarmq.DoorbellQueue(slot_bytes=256, sub_cap=256, n_agents=4,
                           ring_cap=64, device=0)
db.start(60.0)
got = 0
nxt = 0
def drain():
    global got, nxt
    while True:
        m = db.try_recv(1)
        if m is None:
            return
        v = struct.unpack("<I", bytes(m[1])[:4])[0]
        assert v == nxt, (v, nxt)
        nxt += 1; got += 1
try:
    for i in range(50000):
        while i - got >= 48:
            m = db.recv_spin(1, timeout_us=2e6)
            assert m is not None
            v = struct.unpack("<I", bytes(m[1])[:4])[0]
            assert v == nxt, (v, nxt)
            nxt += 1; got += 1
        db.send(receiver=1, sender=0,
                payload=struct.pack("<I", i).ljust(32, b"."))
        drain()
    print("OK", got)
except RuntimeError as e:
    print("STALL at i=?", "got", got, "nxt", nxt)
    print("head", db.head(), "consumed", db.consumed(),
          "delivered[1]", db.delivered_count(1),
          "read_pos[1]", db.read_pos(1), "exited", db.exited())
finally:
    db.stop(); db.release()
