"""Whole-system CPU scenario: one daemon arbitrating two GPUs, three
interposed clients, memory pressure, preemption, live ctl.

Exercises in ONE run what the unit suites cover piecewise — the
closest no-GPU analogue of the reference's two-container co-location
demo (README.md:310-356) extended to the multi-GPU scheduler.
"""

from __future__ import annotations

import os
import subprocess
import threading
import time

from nvshare_amd import ctl
from nvshare_amd.env import client_env


def run_client(artifacts, sock_dir, gpu, label, results,
               alloc_mib=100, allocs=4, iters=400):
    env = client_env(sock_dir=sock_dir, use_stub=True, debug=True,
                     reserve_mib=64)
    env["NVSTUB_TOTAL_MIB"] = "600"
    env["NVSTUB_LOG"] = os.path.join(sock_dir, f"{label}.events")
    env["NVSHARE_GPU"] = str(gpu)
    env["NVSHARE_POD_NAME"] = label
    env["NVSHARE_RELEASE_INTERVAL_MS"] = "100"
    t0 = time.monotonic()
    r = subprocess.run(
        [str(artifacts.hipclient), "--allocs", str(allocs),
         "--alloc-mib", str(alloc_mib), "--iters", str(iters),
         "--sleep-us", "5000", "--sync-every", "50"],
        env=env, capture_output=True, text=True, timeout=120)
    results[label] = (r, time.monotonic() - t0)


def test_two_gpus_three_clients_with_pressure(artifacts, sched,
                                              sock_dir):
    results: dict = {}
    # a0+b0 share gpu0 and oversubscribe it (2x400 > 600 MiB);
    # c1 runs alone on gpu1 and must NOT be serialized against them.
    ts = [
        threading.Thread(target=run_client,
                         args=(artifacts, sock_dir, 0, "a0", results)),
        threading.Thread(target=run_client,
                         args=(artifacts, sock_dir, 0, "b0", results)),
        threading.Thread(target=run_client,
                         args=(artifacts, sock_dir, 1, "c1", results)),
    ]
    t_start = time.monotonic()
    for t in ts:
        t.start()
    # While they run: live status + a TQ change through nvsharectl's
    # python twin.
    time.sleep(1.0)
    st = ctl.status(sock_dir)
    assert st.clients >= 2
    ctl.set_tq(2, sock_dir)

    for t in ts:
        t.join()
    makespan = time.monotonic() - t_start

    for label, (r, secs) in results.items():
        assert r.returncode == 0, (label, r.stdout, r.stderr[-1500:])
        assert "PASS" in r.stdout, label

    # gpu1's solo client must overlap with gpu0's pair: if the
    # scheduler wrongly serialized all three, the makespan would be
    # ~sum of all three run times.
    total = sum(secs for _, secs in results.values())
    assert makespan < 0.8 * total, (makespan, total)

    log = sched.log_text()
    assert "on gpu0" in log and "on gpu1" in log
    # gpu0 was preempted at least once (TQ=1s fixture, two clients).
    assert "DROP_LOCK" in log
    # Pressure on gpu0 (800 MiB tracked vs 600 capacity) reached the
    # clients: at least one eviction-to-host in their stub events.
    evicts = 0
    for label in ("a0", "b0"):
        path = os.path.join(sock_dir, f"{label}.events")
        if os.path.exists(path):
            with open(path) as f:
                evicts += f.read().count("hipMemPrefetchAsync_cpu")
    assert evicts > 0
    # Everything tears down cleanly: daemon still answers.
    st = ctl.status(sock_dir)
    assert st.clients == 0 or st.clients <= 1  # stragglers drain
