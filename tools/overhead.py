#!/usr/bin/env python3
"""Interposition-overhead A/B: stock vs nvshare-amd (solo client).

Reproduces the reference's headline "~1% slowdown" experiment
 on MI355X: run each workload stock, then under the
interposer (standalone: managed memory only), then under interposer +
scheduler, and report the slowdown ratios.

Usage (on a GPU box):
    python tools/overhead.py [--quick] [--out profiles/overhead.json]
"""

from __future__ import annotations

import argparse
import json
import os
import subprocess
import sys
import tempfile
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

from nvshare_amd.colocate import _parse_json_line, workload_cmd  # noqa: E402
from nvshare_amd.env import client_env  # noqa: E402
from nvshare_amd.scheduler import SchedulerDaemon  # noqa: E402


def run_one(cmd, env, timeout=1200):
    r = subprocess.run(cmd, env=env, capture_output=True, text=True,
                       timeout=timeout)
    if r.returncode != 0:
        raise RuntimeError(f"{cmd} failed: {r.stderr[-2000:]}")
    res = _parse_json_line(r.stdout)
    assert res is not None, r.stdout
    return res


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--quick", action="store_true")
    ap.add_argument("--out", default="profiles/overhead.json")
    ap.add_argument("--repeats", type=int, default=1)
    args = ap.parse_args()

    if args.quick:
        workloads = {
            "pytorch_add": workload_cmd("pytorch_add", "--gb", "4",
                                        "--iters", "300"),
            "matmul": workload_cmd("matmul", "--gb", "4", "--iters", "8"),
            "train_resnet50": workload_cmd(
                "train_resnet", "--batch", "32", "--steps", "20",
                "--warmup", "5"),
        }
    else:
        workloads = {
            "pytorch_add": workload_cmd("pytorch_add", "--gb", "12",
                                        "--iters", "1000"),
            "matmul": workload_cmd("matmul", "--gb", "12", "--iters",
                                   "20"),
            "train_resnet50": workload_cmd(
                "train_resnet", "--batch", "64", "--steps", "60",
                "--warmup", "10"),
        }

    sock_dir = tempfile.mkdtemp(prefix="nvs-ovh-", dir="/tmp")
    results: dict = {"workloads": {}, "host": os.uname().nodename}
    with SchedulerDaemon(sock_dir=sock_dir, tq=30):
        for name, cmd in workloads.items():
            row = {}
            stock_env = dict(os.environ)
            row["stock"] = min(
                run_one(cmd, stock_env)["seconds"]
                for _ in range(args.repeats))
            um_env = client_env(sock_dir=sock_dir, standalone=True)
            row["managed"] = min(
                run_one(cmd, um_env)["seconds"]
                for _ in range(args.repeats))
            sched_env = client_env(sock_dir=sock_dir)
            row["managed+sched"] = min(
                run_one(cmd, sched_env)["seconds"]
                for _ in range(args.repeats))
            row["slowdown_managed"] = row["managed"] / row["stock"]
            row["slowdown_full"] = row["managed+sched"] / row["stock"]
            results["workloads"][name] = row
            print(f"{name}: stock={row['stock']:.2f}s "
                  f"managed={row['managed']:.2f}s "
                  f"(x{row['slowdown_managed']:.4f}) "
                  f"full={row['managed+sched']:.2f}s "
                  f"(x{row['slowdown_full']:.4f})", flush=True)

    ratios = [w["slowdown_full"] for w in results["workloads"].values()]
    results["mean_slowdown_full"] = sum(ratios) / len(ratios)
    results["max_slowdown_full"] = max(ratios)
    out = Path(args.out)
    out.parent.mkdir(parents=True, exist_ok=True)
    out.write_text(json.dumps(results, indent=2))
    print(json.dumps({"mean_slowdown": results["mean_slowdown_full"],
                      "max_slowdown": results["max_slowdown_full"]}))


if __name__ == "__main__":
    main()
