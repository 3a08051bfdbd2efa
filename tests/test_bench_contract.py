"""bench.py driver-contract tests: JSON schema, single-process run, and
the exact torchrun multi-rank launch path the driver uses (gloo on CPU)."""

import json
import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent

REQUIRED_KEYS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


def _last_json_line(stdout: str) -> dict:
    for line in reversed(stdout.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{stdout}")


def test_bench_single_process_contract():
    proc = subprocess.run(
        [sys.executable, "bench.py", "--steps", "3", "--warmup", "1",
         "--batch", "512"],
        capture_output=True, text=True, timeout=300, cwd=str(REPO),
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    out = _last_json_line(proc.stdout)
    assert REQUIRED_KEYS <= set(out.keys()), out.keys()
    assert out["value"] > 0
    assert out["higher_is_better"] is True
    assert out["scaling"] == "weak"
    assert out["data"] == "synthetic"
    assert out["steps"] == 3 and out["warmup"] == 1
    assert {"model", "global_batch", "seq_len", "parallelism"} <= set(
        out["config"].keys()
    )


def test_bench_torchrun_world2():
    """The driver's exact launch shape: torch.distributed.run, nnodes=1,
    nproc-per-node 2, master-addr 127.0.0.1 (gloo fallback on CPU)."""
    env = dict(os.environ)
    env.setdefault("GLOO_SOCKET_IFNAME", "lo")
    proc = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", "29517",
            "bench.py", "--gpus", "2", "--steps", "2", "--warmup", "1",
            "--batch", "256", "--agents", "64",
        ],
        capture_output=True, text=True, timeout=300, env=env, cwd=str(REPO),
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    out = _last_json_line(proc.stdout)
    assert out["config"]["agents"] == 128  # 64 per rank, weak scaling
    assert out["config"]["global_batch"] == 512
    assert "all-to-all" in out["config"]["parallelism"]
    assert out["value"] > 0


def test_bench_broadcast_mode():
    proc = subprocess.run(
        [sys.executable, "bench.py", "--steps", "3", "--warmup", "1",
         "--batch", "256", "--agents", "64", "--bcast-frac", "0.05"],
        capture_output=True, text=True, timeout=300, cwd=str(REPO),
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    out = _last_json_line(proc.stdout)
    assert out["value"] > 0
    # fan-out amplifies deliveries; no loss warning expected
    assert "loss" not in proc.stderr
