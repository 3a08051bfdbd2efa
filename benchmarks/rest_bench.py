#!/usr/bin/env python3
"""REST serving benchmark — end-to-end through a real uvicorn process.

Note: this environment has no uvloop/httptools, so uvicorn runs the
pure-Python asyncio+h11 stack — a localhost round trip costs ~3 ms here
regardless of the handler (a bare /health measures the same). Treat
send_req_per_s as an ASGI-stack floor, batch_msgs_per_s as the service's
real ingestion capability (the reference's FastAPI tier had the same
stack in front of Kafka, plus a 300 req/min rate limit).

The server runs in its own process (its own GIL); the client drives it
with an async httpx client at a fixed concurrency. Three request shapes:

  send      POST /messages          (per-message compat path)
  batch     POST /messages/batch    (bulk ingestion, one kernel per call)
  receive   POST /agents/receive    (consumer poll)

Usage: python -m benchmarks.rest_bench [--seconds S] [--batch N] [--conc C]
"""

from __future__ import annotations

import argparse
import asyncio
import json
import os
import subprocess
import sys
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))


async def run_phases(base: str, seconds: float, batch: int, conc: int) -> dict:
    import httpx

    async with httpx.AsyncClient(base_url=base, timeout=30.0) as c:
        for _ in range(200):
            try:
                if (await c.get("/health")).status_code == 200:
                    break
            except httpx.TransportError:
                await asyncio.sleep(0.1)

        tok = (await c.post(
            "/auth/token", json={"username": "bench", "password": "x"}
        )).json()["access_token"]
        hdr = {"Authorization": f"Bearer {tok}"}
        await c.post("/agents/register", headers=hdr,
                     json={"agent_id": "bench"})
        tok2 = (await c.post(
            "/auth/token", json={"username": "sink", "password": "x"}
        )).json()["access_token"]
        hdr2 = {"Authorization": f"Bearer {tok2}"}
        await c.post("/agents/register", headers=hdr2,
                     json={"agent_id": "sink"})

        results = {}

        async def hammer(path: str, headers: dict, body, until: float) -> int:
            n = 0
            while time.perf_counter() < until:
                r = await c.post(path, headers=headers, json=body)
                assert r.status_code == 200, r.text
                n += 1
            return n

        # phase 1: per-message sends at `conc` in flight
        until = time.perf_counter() + seconds
        t0 = time.perf_counter()
        counts = await asyncio.gather(*[
            hammer("/messages", hdr,
                   {"receiver_id": "sink", "content": "x" * 256}, until)
            for _ in range(conc)
        ])
        results["send_req_per_s"] = round(sum(counts) / (time.perf_counter() - t0), 1)

        # phase 2: bulk sends (4 in flight is plenty; each carries `batch`)
        body = [{"receiver_id": "sink", "content": "y" * 256}
                for _ in range(batch)]
        until = time.perf_counter() + seconds
        t0 = time.perf_counter()
        counts = await asyncio.gather(*[
            hammer("/messages/batch", hdr, body, until) for _ in range(4)
        ])
        results["batch_msgs_per_s"] = round(
            sum(counts) * batch / (time.perf_counter() - t0), 1
        )

        # phase 3: receive drain (single consumer identity)
        n = 0
        t0 = time.perf_counter()
        while time.perf_counter() - t0 < seconds:
            r = await c.post("/agents/receive?timeout=0&max_messages=1000",
                             headers=hdr2)
            msgs = r.json()
            n += len(msgs)
            if not msgs:
                break
        dt = time.perf_counter() - t0
        results["receive_msgs_per_s"] = round(n / dt, 1) if n else 0.0
        return results


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=4.0)
    ap.add_argument("--batch", type=int, default=512)
    ap.add_argument("--conc", type=int, default=64)
    ap.add_argument("--port", type=int, default=8311)
    args = ap.parse_args()

    env = dict(os.environ)
    env.update(
        RATE_LIMIT_PER_MINUTE="1000000000",
        SWARMQ_MAX_AGENTS="1024",
        MESSAGE_HISTORY_DIR="/tmp/swarmdb_rest_bench_hist",
        SAVE_INTERVAL_SECONDS="1000000",
        PYTHONPATH=str(REPO),
    )
    server = subprocess.Popen(
        [sys.executable, "-m", "uvicorn", "swarmdb_amd.api.app:get_app",
         "--factory", "--host", "127.0.0.1", "--port", str(args.port),
         "--log-level", "error"],
        env=env, cwd=str(REPO),
    )
    try:
        results = asyncio.run(
            run_phases(f"http://127.0.0.1:{args.port}", args.seconds,
                       args.batch, args.conc)
        )
        gpu = False
        try:
            import torch

            gpu = torch.cuda.is_available()
        except Exception:
            pass
        print(json.dumps({
            "name": "rest-serving",
            "engine": "gpu" if gpu else "cpu",
            "batch": args.batch,
            "concurrency": args.conc,
            **results,
        }))
    finally:
        server.terminate()
        try:
            server.wait(timeout=10)
        except subprocess.TimeoutExpired:
            server.kill()
            server.wait(timeout=10)
    return 0


if __name__ == "__main__":
    sys.exit(main())
