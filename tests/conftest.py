"""Shared fixtures for nvshare-amd tests."""

from __future__ import annotations

import shutil
import sys
import tempfile
from pathlib import Path

import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO_ROOT))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: requires a real MI355X GPU (run with -m gpu)")


@pytest.fixture(scope="session")
def artifacts():
    from nvshare_amd.paths import ensure_built

    return ensure_built()


@pytest.fixture
def sock_dir():
    # Unix socket paths are limited to ~108 bytes; pytest tmp_path can
    # exceed that, so make a short one under /tmp.
    d = tempfile.mkdtemp(prefix="nvs-", dir="/tmp")
    yield d
    shutil.rmtree(d, ignore_errors=True)


@pytest.fixture
def sched(artifacts, sock_dir):
    from nvshare_amd.scheduler import SchedulerDaemon

    daemon = SchedulerDaemon(sock_dir=sock_dir, tq=1, debug=True)
    daemon.start()
    yield daemon
    daemon.stop()
