"""Launch and supervise the nvshare-scheduler daemon."""

from __future__ import annotations

import os
import signal
import subprocess
import time
from pathlib import Path

from nvshare_amd import proto
from nvshare_amd.paths import ensure_built


class SchedulerDaemon:
    """Runs nvshare-scheduler as a child process.

    Usage:
        with SchedulerDaemon(sock_dir="/tmp/x", tq=5) as sched:
            ... clients connect via NVSHARE_SOCK_DIR=/tmp/x ...
    """

    def __init__(self, sock_dir: str, tq: int | None = None,
                 debug: bool = True, sched_off: bool = False,
                 log_path: str | None = None):
        self.sock_dir = sock_dir
        self.tq = tq
        self.debug = debug
        self.sched_off = sched_off
        self.log_path = log_path or os.path.join(sock_dir, "scheduler.log")
        self.proc: subprocess.Popen | None = None

    @property
    def socket_path(self) -> str:
        return proto.scheduler_path(self.sock_dir)

    def start(self, ready_timeout: float = 10.0) -> "SchedulerDaemon":
        art = ensure_built()
        Path(self.sock_dir).mkdir(parents=True, exist_ok=True)
        env = dict(os.environ)
        env["NVSHARE_SOCK_DIR"] = self.sock_dir
        if self.tq is not None:
            env["NVSHARE_TQ"] = str(self.tq)
        env["NVSHARE_DEBUG"] = "1" if self.debug else "0"
        if self.sched_off:
            env["NVSHARE_SCHED_OFF"] = "1"
        self._log_f = open(self.log_path, "ab")
        self.proc = subprocess.Popen(
            [str(art.scheduler)], env=env,
            stdout=self._log_f, stderr=self._log_f,
        )
        deadline = time.monotonic() + ready_timeout
        while time.monotonic() < deadline:
            if os.path.exists(self.socket_path):
                return self
            if self.proc.poll() is not None:
                raise RuntimeError(
                    f"nvshare-scheduler exited rc={self.proc.returncode}; "
                    f"see {self.log_path}")
            time.sleep(0.02)
        raise TimeoutError("scheduler socket did not appear")

    def stop(self) -> None:
        if self.proc is not None and self.proc.poll() is None:
            self.proc.send_signal(signal.SIGTERM)
            try:
                self.proc.wait(timeout=5)
            except subprocess.TimeoutExpired:
                self.proc.kill()
                self.proc.wait(timeout=5)
        if getattr(self, "_log_f", None):
            self._log_f.close()
            self._log_f = None
        self.proc = None

    def log_text(self) -> str:
        try:
            return Path(self.log_path).read_text(errors="replace")
        except FileNotFoundError:
            return ""

    def __enter__(self) -> "SchedulerDaemon":
        return self.start()

    def __exit__(self, *exc) -> None:
        self.stop()
