#!/usr/bin/env python3
"""Thrashing microbenchmark: ResNet training at growing batch sizes,
solo and 2x parallel (the reference ran this as the
"dogbreed" Keras ResNet152 notebook on a 16 GB P100; batch 64 at 2x
parallel thrashed ~15x).

On a 288 GB MI355X the memory wall is created with tools/squatter.py;
batch sizes sweep the combined working set across the remaining HBM.

Usage (GPU box):
    python tools/thrashbench.py --squat-leave-gb 24 \
        --batches 32,64,128 --out profiles/thrashbench.json
"""

from __future__ import annotations

import argparse
import glob as _glob
import json
import subprocess
import sys
import tempfile
import threading
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

from nvshare_amd import ctl  # noqa: E402
from nvshare_amd.colocate import run_colocated, workload_cmd  # noqa: E402
from nvshare_amd.scheduler import SchedulerDaemon  # noqa: E402


class VramSampler:
    """Peak VRAM (MiB) over a window, from sysfs (thesis Tables
    11.7/11.8 report peak GPU memory per scenario)."""

    def __init__(self, period_s: float = 0.5):
        self.paths = _glob.glob(
            "/sys/class/drm/card*/device/mem_info_vram_used")
        self.peak = 0
        self.period = period_s
        self._stop = threading.Event()
        self._t = threading.Thread(target=self._run, daemon=True)

    def _read(self) -> int:
        total = 0
        for p in self.paths:
            try:
                with open(p) as f:
                    total += int(f.read().strip())
            except OSError:
                pass
        return total

    def _run(self) -> None:
        while not self._stop.is_set():
            v = self._read()
            if v > self.peak:
                self.peak = v
            self._stop.wait(self.period)

    def __enter__(self):
        self._t.start()
        return self

    def __exit__(self, *exc):
        self._stop.set()
        self._t.join(timeout=2)

    @property
    def peak_mib(self) -> int:
        return self.peak // (1024 * 1024)


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--batches", default="32,64,128")
    ap.add_argument("--steps", type=int, default=60)
    ap.add_argument("--image", type=int, default=224)
    ap.add_argument("--squat-leave-gb", type=float, default=0.0)
    ap.add_argument("--tq", type=int, default=30)
    ap.add_argument("--include-off", action="store_true",
                    help="also run the scheduler-off arm (may DNF)")
    ap.add_argument("--off-timeout", type=float, default=600.0)
    ap.add_argument("--timeout", type=float, default=1800.0)
    ap.add_argument("--out", default="profiles/thrashbench.json")
    args = ap.parse_args()

    squat = None
    if args.squat_leave_gb > 0:
        squat = subprocess.Popen(
            [sys.executable, str(REPO / "tools" / "squatter.py"),
             "--leave-gb", str(args.squat_leave_gb),
             "--seconds", "86400"],
            stdout=subprocess.PIPE, text=True)
        line = squat.stdout.readline()
        assert "SQUATTING" in line, line
        print(line.strip(), flush=True)

    sock_dir = tempfile.mkdtemp(prefix="nvs-thrash-", dir="/tmp")
    out: dict = {"model": args.model, "steps": args.steps,
                 "squat_leave_gb": args.squat_leave_gb,
                 "tq": args.tq, "rows": []}
    env_kwargs = {"oversubscribe": True}

    with SchedulerDaemon(sock_dir=sock_dir, tq=args.tq):
        for batch in (int(b) for b in args.batches.split(",")):
            cmd = workload_cmd(
                "train_resnet", "--model", args.model, "--batch",
                str(batch), "--image", str(args.image), "--steps",
                str(args.steps), "--warmup", "5")
            row: dict = {"batch": batch}
            with VramSampler() as vs:
                solo = run_colocated([cmd], sock_dir=sock_dir,
                                     env_kwargs=env_kwargs,
                                     timeout=args.timeout)
            row["solo_s"] = solo.makespan if solo.ok else None
            row["solo_peak_mib"] = vs.peak_mib
            with VramSampler() as vs:
                two = run_colocated([cmd, cmd], sock_dir=sock_dir,
                                    env_kwargs=env_kwargs,
                                    timeout=args.timeout)
            row["parallel2_s"] = two.makespan if two.ok else None
            row["parallel2_peak_mib"] = vs.peak_mib
            if row["solo_s"] and row["parallel2_s"]:
                row["parallel_vs_serial"] = (
                    row["parallel2_s"] / (2 * row["solo_s"]))
            if args.include_off:
                ctl.set_scheduling(False, sock_dir)
                with VramSampler() as vs:
                    off = run_colocated([cmd, cmd], sock_dir=sock_dir,
                                        env_kwargs=env_kwargs,
                                        timeout=args.off_timeout)
                row["parallel2_sched_off_s"] = (
                    off.makespan if off.ok else None)
                row["parallel2_sched_off_peak_mib"] = vs.peak_mib
                ctl.set_scheduling(True, sock_dir)
            out["rows"].append(row)
            print(json.dumps(row), flush=True)

    if squat is not None:
        squat.terminate()

    o = Path(args.out)
    o.parent.mkdir(parents=True, exist_ok=True)
    o.write_text(json.dumps(out, indent=2))


if __name__ == "__main__":
    main()
