"""Co-location runner: N client jobs sharing one GPU via nvshare-amd.

Launches N subprocesses under the interposer against one scheduler,
collects per-job wall times and the overall makespan — the measurement
unit for the headline metric (BASELINE.json: "makespan + per-job
slowdown% for N co-located jobs on 1 MI355X").
"""

from __future__ import annotations

import json
import subprocess
import sys
import time
from dataclasses import dataclass

from nvshare_amd.env import client_env


@dataclass
class JobResult:
    label: str
    returncode: int
    seconds: float
    stdout: str
    stderr: str
    result: dict | None = None  # parsed JSON line, if any

    @property
    def ok(self) -> bool:
        return self.returncode == 0 and "PASS" in self.stdout


@dataclass
class ColocationResult:
    jobs: list[JobResult]
    makespan: float
    started: float = 0.0

    @property
    def ok(self) -> bool:
        return all(j.ok for j in self.jobs)


def _parse_json_line(stdout: str) -> dict | None:
    for line in reversed(stdout.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            try:
                return json.loads(line)
            except json.JSONDecodeError:
                continue
    return None


def run_colocated(
    cmds: list[list[str]],
    sock_dir: str | None = None,
    env_kwargs: dict | None = None,
    timeout: float = 3600.0,
    stagger_s: float = 0.0,
) -> ColocationResult:
    """Run all cmds concurrently as nvshare clients; wait for all."""
    env_kwargs = dict(env_kwargs or {})
    procs: list[tuple[str, subprocess.Popen, float]] = []
    t_start = time.monotonic()
    for i, cmd in enumerate(cmds):
        env = client_env(sock_dir=sock_dir, **env_kwargs)
        env.setdefault("NVSHARE_POD_NAME", f"job{i}")
        p = subprocess.Popen(cmd, env=env, stdout=subprocess.PIPE,
                             stderr=subprocess.PIPE, text=True)
        procs.append((f"job{i}", p, time.monotonic()))
        if stagger_s > 0:
            time.sleep(stagger_s)

    jobs: list[JobResult] = []
    deadline = time.monotonic() + timeout
    for label, p, t0 in procs:
        left = max(1.0, deadline - time.monotonic())
        try:
            out, err = p.communicate(timeout=left)
        except subprocess.TimeoutExpired:
            p.kill()
            out, err = p.communicate()
            jobs.append(JobResult(label, -9, time.monotonic() - t0,
                                  out, err))
            continue
        jobs.append(JobResult(label, p.returncode,
                              time.monotonic() - t0, out, err,
                              _parse_json_line(out)))
    makespan = time.monotonic() - t_start
    return ColocationResult(jobs, makespan, t_start)


def workload_cmd(module: str, *args: str) -> list[str]:
    return [sys.executable, "-m", f"nvshare_amd.workloads.{module}",
            *args]
