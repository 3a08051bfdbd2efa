#!/usr/bin/env python3
"""Express-lane endurance: continuous ping-pong through the persistent
doorbell kernel for N seconds; per-minute latency percentiles show the
resident kernel holds its latency over its lifetime."""
import json
import sys
import time
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from swarmdb_amd import _swarmq  # noqa: E402

seconds = float(sys.argv[1]) if len(sys.argv) > 1 else 240.0
db = _swarmq.DoorbellQueue(slot_bytes=1024, sub_cap=256, n_agents=2,
                           ring_cap=64, device=0)
db.start(seconds + 60.0)
pay = b"x" * 256
windows = []
cur = []
t0 = time.perf_counter()
next_cut = 60.0
n = 0
try:
    while True:
        el = time.perf_counter() - t0
        if el >= seconds:
            break
        if el >= next_cut:
            windows.append(cur)
            cur = []
            next_cut += 60.0
        s = time.perf_counter()
        db.send(receiver=1, sender=0, payload=pay)
        assert db.recv_spin(1, timeout_us=2e6) is not None
        cur.append(time.perf_counter() - s)
        n += 1
    windows.append(cur)
finally:
    db.stop()
    db.release()
out = {
    "name": "express-endurance",
    "seconds": round(time.perf_counter() - t0, 1),
    "messages": n,
    "per_minute_p50_us": [round(float(np.median(w) * 1e6), 1)
                          for w in windows if w],
    "per_minute_p99_us": [round(float(np.percentile(w, 99) * 1e6), 1)
                          for w in windows if w],
}
print(json.dumps(out))
