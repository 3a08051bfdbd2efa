import sys, time, json
sys.path.insert(0, "/root/repo")
import numpy as np
from swarmdb_amd import QueueConfig
from swarmdb_amd.runtime.gpu_engine import GpuEngine
from swarmdb_amd.runtime.engine import REC_DTYPE, NO_BITMAP, VIS_ALL

n_msgs = 1 << 20
cfg = QueueConfig(use_gpu=True, auto_save=False, max_agents=1024,
                  num_slots=n_msgs, slot_bytes=1024 + 64,
                  staging_batch=1 << 16, inbox_capacity=1 << 12)
eng = GpuEngine(cfg)
rng = np.random.default_rng(0)
agents = np.arange(64, dtype=np.uint32)
for a in agents: eng.register_agent(int(a))
batch, stride = 1 << 16, 1024
for b in range(n_msgs // batch):
    recs = np.zeros(batch, dtype=REC_DTYPE)
    recs["sender"] = rng.choice(agents, batch)
    recs["receiver"] = rng.choice(agents, batch)
    recs["vis_mode"] = VIS_ALL; recs["bitmap"] = NO_BITMAP
    recs["payload_len"] = stride; recs["content_len"] = stride
    recs["payload_off"] = np.arange(batch, dtype=np.uint64) * stride
    pay = bytearray(rng.integers(32, 127, batch * stride, dtype=np.uint8).tobytes())
    for i in range(0, batch, 4096):
        off = int(recs["payload_off"][i]) + 100
        pay[off:off+12] = b"NEEDLE-%05d" % (b % 10)
    eng.enqueue_batch(recs, bytes(pay))
total = n_msgs * 1088  # full slot region bytes

def bench(needle, label, expect_hits):
    ts = []
    for _ in range(20):
        s = time.perf_counter()
        hits = eng.q.search_range(0, n_msgs, needle, False, 16384)
        ts.append(time.perf_counter() - s)
        if expect_hits: assert len(hits) > 0
    p50 = float(np.median(ts))
    print(label, round(total / p50 / 1e9, 1), "GB/s", round(p50*1e3,3), "ms")

bench(b"NEEDLE-00003", "normal      ", True)
bench(b"\x01\x02\x03\x04", "impossible  ", False)   # zero candidates
bench(b"\x01" + b"EEDLE", "no-first-byte", False)   # first byte absent
bench(b"N\x01EDLE", "first-only  ", False)          # n0 hits, n1 never
eng.close()
