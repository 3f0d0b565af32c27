"""Shared helpers for workloads."""

from __future__ import annotations

import argparse
import json
import sys
import time


def add_common_args(ap: argparse.ArgumentParser) -> None:
    ap.add_argument("--device", default="cuda",
                    help="cuda (default) or cpu")
    ap.add_argument("--label", default="", help="job label for logs")


def emit(result: dict) -> None:
    """Print PASS + the machine-readable result line."""
    print(f"PASS {result.get('seconds', 0.0):.3f}", flush=True)
    print(json.dumps(result), flush=True)


class Timer:
    def __enter__(self):
        self.t0 = time.monotonic()
        return self

    def __exit__(self, *exc):
        self.seconds = time.monotonic() - self.t0


def sync(device: str) -> None:
    if device.startswith("cuda"):
        import torch

        torch.cuda.synchronize()


def die(msg: str) -> None:
    print(f"FAIL {msg}", file=sys.stderr, flush=True)
    sys.exit(1)
