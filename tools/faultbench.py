#!/usr/bin/env python3
"""gfx950 XNACK/HMM demand-paging microbenchmark.

Quantifies the data plane nvshare-amd delegates to the driver
(SURVEY.md §7 "hard parts" #1): how fast do pages migrate
host<->device under first-touch faulting vs explicit
hipMemPrefetchAsync, and what does an oversubscribed round-trip cost?
These numbers size the TQ default and justify NVSHARE_PREFETCH.

Usage (GPU box): python tools/faultbench.py [--gb 4] [--out profiles/faultbench.json]
"""

from __future__ import annotations

import argparse
import ctypes
import json
import sys
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

from nvshare_amd import hiputil  # noqa: E402


def timed(f):
    t0 = time.monotonic()
    f()
    return time.monotonic() - t0


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gb", type=float, default=4.0)
    ap.add_argument("--out", default="profiles/faultbench.json")
    args = ap.parse_args()

    h = hiputil.load()
    nbytes = int(args.gb * (1 << 30))
    n = nbytes // 4
    res = {"gb": args.gb}

    # 1. Device-first-touch of a fresh managed buffer (pages populate
    #    on device).
    ptr = h.malloc_managed(nbytes)
    res["gpu_first_touch_gbs"] = args.gb / timed(
        lambda: (h.touch_pages(ptr, n, 1, 1.0), h.sync()))

    # 2. Steady-state re-touch (resident on device).
    res["gpu_resident_touch_gbs"] = args.gb / timed(
        lambda: (h.touch_pages(ptr, n, 1, 1.0), h.sync()))

    # 3. Host touch (migrates everything to host over PCIe/HMM).
    arr = (ctypes.c_float * n).from_address(ptr)

    def host_touch():
        step = 1024  # one touch per 4 KiB page
        for i in range(0, n, step):
            arr[i] = 2.0
    res["host_pull_gbs"] = args.gb / timed(host_touch)

    # 4. GPU re-touch after host steal: pure XNACK refault path.
    res["gpu_refault_gbs"] = args.gb / timed(
        lambda: (h.touch_pages(ptr, n, 1, 1.0), h.sync()))

    # 5. Same migration, but with an explicit prefetch first (what
    #    NVSHARE_PREFETCH=1 does on LOCK_OK).
    host_touch()
    res["gpu_prefetch_then_touch_gbs"] = args.gb / timed(
        lambda: (h.prefetch(ptr, nbytes, 0), h.touch_pages(ptr, n, 1, 1.0),
                 h.sync()))
    h.free(ptr)

    # 6. Plain (non-managed) HBM write bandwidth for scale.
    ptr2 = h.malloc(nbytes)
    h.touch_pages(ptr2, n, 1, 1.0)
    h.sync()
    res["hbm_touch_gbs"] = args.gb / timed(
        lambda: (h.touch_pages(ptr2, n, 1, 1.0), h.sync()))
    h.free(ptr2)

    res["prefetch_speedup_vs_refault"] = (
        res["gpu_prefetch_then_touch_gbs"] / res["gpu_refault_gbs"])
    out = Path(args.out)
    out.parent.mkdir(parents=True, exist_ok=True)
    out.write_text(json.dumps(res, indent=2))
    print(json.dumps(res, indent=2))


if __name__ == "__main__":
    main()
