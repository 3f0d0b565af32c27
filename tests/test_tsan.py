"""ThreadSanitizer run of the scheduler daemon (race detection).

The reference had no sanitizer coverage; here the
daemon's full lock lifecycle runs under TSan and the test fails on any
reported race.
"""

from __future__ import annotations

import os
import signal
import subprocess
import time
from pathlib import Path

import pytest

from nvshare_amd import proto

REPO = Path(__file__).resolve().parent.parent
TSAN_BIN = REPO / "src" / "build-tsan" / "nvshare-scheduler"


@pytest.fixture(scope="module")
def tsan_bin():
    r = subprocess.run(["make", "-s", "tsan"], cwd=REPO,
                       capture_output=True, text=True)
    if r.returncode != 0 or not TSAN_BIN.exists():
        pytest.skip(f"tsan build unavailable: {r.stderr[-500:]}")
    return TSAN_BIN


def test_scheduler_lifecycle_under_tsan(tsan_bin, sock_dir):
    env = dict(os.environ)
    env["NVSHARE_SOCK_DIR"] = sock_dir
    env["NVSHARE_TQ"] = "1"
    env["NVSHARE_DEBUG"] = "1"
    env["TSAN_OPTIONS"] = "exitcode=66 halt_on_error=0"
    log = open(os.path.join(sock_dir, "tsan.log"), "wb")
    p = subprocess.Popen([str(tsan_bin)], env=env, stdout=log,
                         stderr=log)
    try:
        deadline = time.monotonic() + 10
        spath = proto.scheduler_path(sock_dir)
        while not os.path.exists(spath):
            assert time.monotonic() < deadline, "socket never appeared"
            assert p.poll() is None
            time.sleep(0.05)

        # Drive a full multi-client lifecycle incl. TQ preemption.
        a = proto.Client(sock_dir=sock_dir, pod_name="a").connect()
        a.register()
        b = proto.Client(sock_dir=sock_dir, pod_name="b").connect()
        b.register()
        a.send(proto.REQ_LOCK)
        assert a.recv(5).type == proto.LOCK_OK
        b.send(proto.REQ_LOCK)
        assert a.recv(5).type == proto.DROP_LOCK
        a.send(proto.LOCK_RELEASED)
        assert b.recv(5).type == proto.LOCK_OK
        a.close()  # eviction path
        b.send(proto.LOCK_RELEASED)
        b.close()
        time.sleep(0.5)
    finally:
        p.send_signal(signal.SIGTERM)
        try:
            rc = p.wait(timeout=10)
        except subprocess.TimeoutExpired:
            p.kill()
            rc = p.wait(timeout=5)
        log.close()

    text = Path(sock_dir, "tsan.log").read_text(errors="replace")
    assert "WARNING: ThreadSanitizer" not in text, text[-4000:]
    assert rc != 66, "TSan reported races"


TSAN_DIR = REPO / "src" / "build-tsan"


def test_client_library_under_tsan(tsan_bin, sock_dir):
    """The interposer + client runtime (gate rwlock, client thread,
    early-release thread, allocation tracker, free cache) run a
    two-client preemption workload under ThreadSanitizer."""
    if not (TSAN_DIR / "libnvshare.so").exists():
        pytest.skip("tsan libnvshare unavailable")
    from nvshare_amd.scheduler import SchedulerDaemon

    with SchedulerDaemon(sock_dir=sock_dir, tq=1, debug=False):
        procs = []
        for i in range(2):
            env = dict(os.environ)
            env["LD_PRELOAD"] = str(TSAN_DIR / "libnvshare.so")
            env["LD_LIBRARY_PATH"] = str(TSAN_DIR)
            env["NVSHARE_SOCK_DIR"] = sock_dir
            env["NVSHARE_POD_NAME"] = f"tsan{i}"
            env["NVSHARE_RELEASE_INTERVAL_MS"] = "100"
            env["NVSHARE_RESERVE_MIB"] = "64"
            env["NVSTUB_TOTAL_MIB"] = "1024"
            env["HSA_XNACK"] = "1"
            env["TSAN_OPTIONS"] = "exitcode=66 halt_on_error=0"
            p = subprocess.Popen(
                [str(TSAN_DIR / "hipclient"), "--allocs", "4",
                 "--alloc-mib", "8", "--iters", "400", "--sleep-us",
                 "2000", "--sync-every", "50"],
                env=env, stdout=subprocess.PIPE,
                stderr=subprocess.PIPE, text=True)
            procs.append(p)
        for p in procs:
            out, err = p.communicate(timeout=120)
            assert "WARNING: ThreadSanitizer" not in err, err[-4000:]
            assert p.returncode == 0, (p.returncode, out, err[-2000:])
