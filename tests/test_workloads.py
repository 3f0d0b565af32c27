"""Workload self-checks on CPU (tiny sizes)."""

from __future__ import annotations

import json
import subprocess
import sys


def run_module(mod, *args, timeout=300):
    return subprocess.run(
        [sys.executable, "-m", f"nvshare_amd.workloads.{mod}", *args],
        capture_output=True, text=True, timeout=timeout)


def last_json(stdout: str) -> dict:
    for line in reversed(stdout.strip().splitlines()):
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in: {stdout!r}")


def test_pytorch_add_cpu():
    r = run_module("pytorch_add", "--device", "cpu", "--gb", "0.01",
                   "--iters", "5")
    assert r.returncode == 0, r.stderr
    assert "PASS" in r.stdout
    res = last_json(r.stdout)
    assert res["iters"] == 5


def test_matmul_cpu():
    r = run_module("matmul", "--device", "cpu", "--gb", "0.01",
                   "--iters", "2")
    assert r.returncode == 0, r.stderr
    assert "PASS" in r.stdout


def test_train_tiny_resnet_cpu():
    r = run_module("train_resnet", "--device", "cpu", "--model", "tiny",
                   "--batch", "2", "--image", "32", "--steps", "2",
                   "--warmup", "1", "--dtype", "float32",
                   "--num-classes", "10")
    assert r.returncode == 0, r.stderr
    res = last_json(r.stdout)
    assert res["steps_per_s"] > 0
    assert res["loss"] == res["loss"]  # not NaN


def test_infer_burst_cpu():
    r = run_module("infer_burst", "--device", "cpu", "--model", "tiny",
                   "--batch", "1", "--image", "32", "--bursts", "2",
                   "--infers-per-burst", "2", "--think-s", "0.05")
    assert r.returncode == 0, r.stderr
    res = last_json(r.stdout)
    assert res["infers"] == 4


def test_resnet50_shape_cpu():
    import torch

    from nvshare_amd.workloads.resnet import resnet50

    m = resnet50(num_classes=10)
    out = m(torch.randn(2, 3, 64, 64))
    assert out.shape == (2, 10)
    n_params = sum(p.numel() for p in m.parameters())
    # ResNet-50 has ~25.6M params at 1000 classes; ~23.5M at 10.
    assert 20e6 < n_params < 30e6
