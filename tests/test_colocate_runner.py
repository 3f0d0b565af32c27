"""Co-location runner unit tests (plain CPU commands, no GPU)."""

from __future__ import annotations

import sys

from nvshare_amd.colocate import (
    ColocationResult, _parse_json_line, run_colocated, workload_cmd,
)


def test_parse_json_line():
    out = "noise\nPASS 1.0\n{\"a\": 1}\n"
    assert _parse_json_line(out) == {"a": 1}
    assert _parse_json_line("no json here") is None
    # last JSON line wins
    out = "{\"a\": 1}\n{\"b\": 2}\n"
    assert _parse_json_line(out) == {"b": 2}


def test_workload_cmd():
    cmd = workload_cmd("matmul", "--gb", "1")
    assert cmd[0] == sys.executable
    assert "nvshare_amd.workloads.matmul" in cmd
    assert cmd[-2:] == ["--gb", "1"]


def test_run_colocated_success(artifacts):
    code = "import json; print('PASS 0.1'); print(json.dumps({'x': 7}))"
    cmd = [sys.executable, "-c", code]
    res = run_colocated([cmd, cmd], env_kwargs={"standalone": True})
    assert isinstance(res, ColocationResult)
    assert res.ok
    assert len(res.jobs) == 2
    assert all(j.result == {"x": 7} for j in res.jobs)
    assert res.makespan > 0


def test_run_colocated_failure_reported(artifacts):
    good = [sys.executable, "-c", "print('PASS 0.0')"]
    bad = [sys.executable, "-c", "import sys; sys.exit(3)"]
    res = run_colocated([good, bad], env_kwargs={"standalone": True})
    assert not res.ok
    assert res.jobs[0].ok and not res.jobs[1].ok
    assert res.jobs[1].returncode == 3


def test_run_colocated_timeout(artifacts):
    slow = [sys.executable, "-c", "import time; time.sleep(30)"]
    res = run_colocated([slow], env_kwargs={"standalone": True},
                        timeout=2)
    assert not res.ok
    assert res.jobs[0].returncode == -9
