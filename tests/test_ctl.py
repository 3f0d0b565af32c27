"""nvsharectl CLI binary tests (no GPU)."""

from __future__ import annotations

import os
import subprocess


def run_ctl(artifacts, sock_dir, *args):
    env = dict(os.environ)
    env["NVSHARE_SOCK_DIR"] = sock_dir
    return subprocess.run([str(artifacts.ctl), *args], env=env,
                          capture_output=True, text=True, timeout=30)


def test_help(artifacts, sock_dir):
    r = run_ctl(artifacts, sock_dir, "-h")
    assert r.returncode == 0
    assert "Usage" in r.stderr or "Usage" in r.stdout


def test_set_tq_and_status(artifacts, sched, sock_dir):
    r = run_ctl(artifacts, sock_dir, "-T", "11")
    assert r.returncode == 0, r.stderr
    r = run_ctl(artifacts, sock_dir, "-q")
    assert r.returncode == 0, r.stderr
    assert "tq: 11 s" in r.stdout


def test_scheduler_toggle(artifacts, sched, sock_dir):
    r = run_ctl(artifacts, sock_dir, "-S", "off")
    assert r.returncode == 0, r.stderr
    r = run_ctl(artifacts, sock_dir, "-q")
    assert "scheduling: off" in r.stdout
    r = run_ctl(artifacts, sock_dir, "-S", "on")
    assert r.returncode == 0
    r = run_ctl(artifacts, sock_dir, "-q")
    assert "scheduling: on" in r.stdout


def test_invalid_tq(artifacts, sock_dir):
    r = run_ctl(artifacts, sock_dir, "-T", "0")
    assert r.returncode != 0
    r = run_ctl(artifacts, sock_dir, "-T", "abc")
    assert r.returncode != 0


def test_invalid_sched_arg(artifacts, sock_dir):
    r = run_ctl(artifacts, sock_dir, "-S", "maybe")
    assert r.returncode != 0


def test_no_scheduler_error(artifacts, sock_dir):
    r = run_ctl(artifacts, sock_dir, "-q")
    assert r.returncode != 0
    assert "cannot connect" in r.stderr
