"""nvshare-run launcher tests (CPU)."""

from __future__ import annotations

import subprocess
import sys


def run_launcher(*args, timeout=60):
    return subprocess.run(
        [sys.executable, "-m", "nvshare_amd.run", *args],
        capture_output=True, text=True, timeout=timeout)


def test_injects_client_env():
    r = run_launcher("--standalone", "--debug", "--", "env")
    assert r.returncode == 0, r.stderr
    env = dict(line.split("=", 1) for line in r.stdout.splitlines()
               if "=" in line)
    assert env["HSA_XNACK"] == "1"
    assert "libnvshare.so" in env["LD_PRELOAD"]
    assert env["NVSHARE_STANDALONE"] == "1"
    assert env["NVSHARE_DEBUG"] == "1"


def test_knobs():
    r = run_launcher("--oversubscribe", "--fake-total-mib", "512",
                     "--reserve-mib", "64", "--prefetch", "--", "env")
    env = dict(line.split("=", 1) for line in r.stdout.splitlines()
               if "=" in line)
    assert env["NVSHARE_ENABLE_SINGLE_OVERSUB"] == "1"
    assert env["NVSHARE_FAKE_TOTAL_MIB"] == "512"
    assert env["NVSHARE_RESERVE_MIB"] == "64"
    assert env["NVSHARE_PREFETCH"] == "1"


def test_no_command_errors():
    r = run_launcher("--standalone")
    assert r.returncode != 0
