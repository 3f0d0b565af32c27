"""Randomized soak: many churning clients against one daemon (no GPU).

20 stub clients with random workloads arrive and depart while TQ
changes and scheduling toggles; everything must finish and the daemon
must stay consistent.
"""

from __future__ import annotations

import random
import subprocess
import time

from nvshare_amd import ctl
from nvshare_amd.env import client_env


def test_soak_churn(artifacts, sched, sock_dir):
    rng = random.Random(42)
    env_base = client_env(sock_dir=sock_dir, use_stub=True,
                          reserve_mib=64)
    env_base["NVSTUB_TOTAL_MIB"] = "4096"

    procs = []
    for i in range(20):
        env = dict(env_base)
        env["NVSHARE_POD_NAME"] = f"soak{i}"
        env["NVSTUB_KERNEL_US"] = str(rng.choice([0, 200, 1000]))
        p = subprocess.Popen(
            [str(artifacts.hipclient),
             "--allocs", str(rng.randint(1, 3)),
             "--alloc-mib", str(rng.choice([4, 16, 64])),
             "--iters", str(rng.randint(50, 400)),
             "--sync-every", str(rng.choice([0, 25]))],
            env=env, stdout=subprocess.PIPE,
            stderr=subprocess.DEVNULL, text=True)
        procs.append(p)
        if i % 4 == 0:
            time.sleep(0.05)

    # Poke the daemon while they run.
    for tq in (2, 1, 5, 1):
        time.sleep(0.3)
        ctl.set_tq(tq, sock_dir)
    ctl.set_scheduling(False, sock_dir)
    time.sleep(0.3)
    ctl.set_scheduling(True, sock_dir)

    deadline = time.monotonic() + 120
    for p in procs:
        out, _ = p.communicate(timeout=max(5, deadline -
                                           time.monotonic()))
        assert p.returncode == 0, out
        assert "PASS" in out

    time.sleep(0.5)
    st = ctl.status(sock_dir)
    assert st.clients == 0
    assert st.queued == 0
    assert st.scheduling_on
