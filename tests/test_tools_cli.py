"""Smoke tests: every ops tool parses --help on CPU (no GPU)."""

from __future__ import annotations

import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent

TOOLS = [
    "overhead.py", "tq_sweep.py", "faultbench.py", "restorebench.py",
    "launchbench.py", "squatter.py", "coverage_check.py",
    "thrashbench.py", "nvshare_top.py", "nanhunt.py",
]


@pytest.mark.parametrize("tool", TOOLS)
def test_tool_help(tool):
    r = subprocess.run(
        [sys.executable, str(REPO / "tools" / tool), "--help"],
        capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stderr
    assert "usage" in r.stdout.lower()


def test_nvshare_top_once(sched, sock_dir):
    """nvshare-top renders a live status line against a real daemon."""
    r = subprocess.run(
        [sys.executable, str(REPO / "tools" / "nvshare_top.py"),
         "--sock-dir", sock_dir, "--once"],
        capture_output=True, text=True, timeout=30)
    assert r.returncode == 0, r.stderr
    assert "scheduling=on" in r.stdout
    assert "clients=0" in r.stdout
