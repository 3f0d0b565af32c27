#!/usr/bin/env python3
"""Co-location / anti-thrashing experiment matrix.

Reproduces the reference's headline table on MI355X: solo vs serial vs
2x-parallel completion times, scheduler on/off, across TQ values, at
configurable (fake-total) oversubscription.  The shape to match:
  - scheduler-off at >1x oversubscription thrashes;
  - scheduler-on co-location beats serial execution;
  - tiny TQ degrades into page-fault domination.

Usage (GPU box):
    python tools/tq_sweep.py --mode fit   --out profiles/colo_fit.json
    python tools/tq_sweep.py --mode oversub --fake-total-mib 16384 \
        --out profiles/colo_oversub.json
"""

from __future__ import annotations

import argparse
import json
import subprocess
import sys
import tempfile
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

from nvshare_amd import ctl  # noqa: E402
from nvshare_amd.colocate import run_colocated, workload_cmd  # noqa: E402
from nvshare_amd.scheduler import SchedulerDaemon  # noqa: E402


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--mode", choices=["fit", "oversub"], default="fit")
    ap.add_argument("--fake-total-mib", type=int, default=0,
                    help="advertised capacity; workload sized to "
                    "oversubscribe it in --mode oversub")
    ap.add_argument("--gb", type=float, default=0.0,
                    help="per-job working set (default: mode-derived)")
    ap.add_argument("--iters", type=int, default=300)
    ap.add_argument("--think-every", type=int, default=0,
                    help="pytorch_add think phases (emulate *_50 "
                         "GPU/CPU-mixed jobs)")
    ap.add_argument("--think-s", type=float, default=2.0)
    ap.add_argument("--tqs", default="30,10,2",
                    help="comma list of TQ values to sweep")
    ap.add_argument("--jobs", type=int, default=2)
    ap.add_argument("--skip-off", action="store_true",
                    help="skip the scheduler-off (thrashing) arm")
    ap.add_argument("--off-timeout", type=float, default=0.0,
                    help="separate (shorter) timeout for the "
                         "scheduler-off thrash arm")
    ap.add_argument("--squat-leave-gb", type=float, default=0.0,
                    help="pin HBM with tools/squatter.py leaving this "
                         "many GB free (REAL oversubscription pressure)")
    ap.add_argument("--migration-assist", action="store_true",
                    help="enable NVSHARE_PREFETCH + NVSHARE_EVICT")
    ap.add_argument("--timeout", type=float, default=1800.0)
    ap.add_argument("--out", default="profiles/colo.json")
    args = ap.parse_args()

    if args.gb <= 0:
        if args.mode == "oversub":
            # 2 jobs x ~0.65*fake-total => ~1.3x oversubscription.
            assert args.fake_total_mib > 0, "--fake-total-mib required"
            args.gb = args.fake_total_mib * 0.65 / 1024
        else:
            args.gb = 8.0

    cmd = workload_cmd("pytorch_add", "--gb", str(args.gb), "--iters",
                       str(args.iters))
    if args.think_every:
        cmd += ["--think-every", str(args.think_every), "--think-s",
                str(args.think_s)]
    env_kwargs = {}
    if args.fake_total_mib:
        env_kwargs = {"fake_total_mib": args.fake_total_mib,
                      "oversubscribe": True,
                      "reserve_mib": max(64, args.fake_total_mib // 32)}
    if args.squat_leave_gb > 0:
        # Real pressure: clients are capped by *advertised* capacity
        # which no longer matches free HBM, so let them oversubscribe.
        env_kwargs.setdefault("oversubscribe", True)
    if args.migration_assist:
        env_kwargs["prefetch"] = True
        env_kwargs["evict"] = True
    if args.think_every:
        # Idle probes must be finer than the think phases for early
        # release to overlap one job's CPU time with the other's GPU.
        env_kwargs["extra"] = {"NVSHARE_RELEASE_INTERVAL_S": "1"}

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

    sock_dir = tempfile.mkdtemp(prefix="nvs-tq-", dir="/tmp")
    out: dict = {"mode": args.mode, "gb_per_job": args.gb,
                 "iters": args.iters, "jobs": args.jobs,
                 "fake_total_mib": args.fake_total_mib,
                 "squat_leave_gb": args.squat_leave_gb,
                 "migration_assist": args.migration_assist, "rows": {}}

    with SchedulerDaemon(sock_dir=sock_dir, tq=30):
        # solo + serial baseline (scheduler on, but only one client)
        solo = run_colocated([cmd], sock_dir=sock_dir,
                             env_kwargs=env_kwargs,
                             timeout=args.timeout)
        assert solo.ok, solo.jobs[0].stderr[-2000:]
        out["rows"]["solo"] = solo.makespan
        out["rows"]["serial_est"] = solo.makespan * args.jobs
        print(f"solo: {solo.makespan:.1f}s", flush=True)

        for tq in [int(x) for x in args.tqs.split(",") if x]:
            ctl.set_tq(tq, sock_dir)
            r = run_colocated([cmd] * args.jobs, sock_dir=sock_dir,
                              env_kwargs=env_kwargs,
                              timeout=args.timeout)
            key = f"parallel_tq{tq}"
            out["rows"][key] = r.makespan if r.ok else None
            print(f"{key}: "
                  f"{'%.1fs' % r.makespan if r.ok else 'FAILED'}",
                  flush=True)

        if not args.skip_off:
            ctl.set_scheduling(False, sock_dir)
            time.sleep(0.5)
            r = run_colocated([cmd] * args.jobs, sock_dir=sock_dir,
                              env_kwargs=env_kwargs,
                              timeout=args.off_timeout or args.timeout)
            out["rows"]["parallel_sched_off"] = (
                r.makespan if r.ok else None)
            out["rows"]["parallel_sched_off_dnf"] = not r.ok
            ctl.set_scheduling(True, sock_dir)
            print(f"sched-off: "
                  f"{'%.1fs' % r.makespan if r.ok else 'TIMEOUT/DNF'}",
                  flush=True)

    if squat is not None:
        squat.terminate()

    o = Path(args.out)
    o.parent.mkdir(parents=True, exist_ok=True)
    o.write_text(json.dumps(out, indent=2))
    print(json.dumps(out["rows"], indent=2))


if __name__ == "__main__":
    main()
