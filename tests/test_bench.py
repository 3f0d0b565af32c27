"""bench.py contract tests (CPU mode)."""

from __future__ import annotations

import json
import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def last_json(stdout: str) -> dict:
    for line in reversed(stdout.strip().splitlines()):
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON in {stdout!r}")


def test_bench_cpu_single():
    r = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--device", "cpu",
         "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr
    out = last_json(r.stdout)
    assert out["metric"] == "colocated_train_samples_per_s_total"
    assert out["value"] > 0
    assert out["n_gpus"] == 1
    assert out["steps"] == 2 and out["warmup"] == 1
    assert out["higher_is_better"] is True
    assert out["scaling"] == "weak"
    assert out["data"] == "synthetic"
    assert "parallelism" in out["config"]


def test_bench_cpu_two_ranks():
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29517", str(REPO / "bench.py"),
         "--device", "cpu", "--gpus", "2", "--steps", "2",
         "--warmup", "1"],
        capture_output=True, text=True, timeout=600, env=env,
        cwd=str(REPO))
    assert r.returncode == 0, r.stderr[-3000:]
    out = last_json(r.stdout)
    assert out["n_gpus"] == 2
    assert out["config"]["global_batch"] == 8  # 4 per rank on cpu
    assert len(out["config"]["per_rank_seconds"]) == 2
