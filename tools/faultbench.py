#!/usr/bin/env python3
"""gfx950 XNACK/HMM demand-paging microbenchmark.

Quantifies the data plane nvshare-amd delegates to the driver
(SURVEY.md §7 "hard parts" #1) and sizes the interposer's policy knobs.
Round-1 finding on MI355X: naive managed memory device-first-touch
runs at ~0.09 GB/s (fine-grain XNACK fault path), vs ~2 TB/s once
resident — so eager population (prefetch at alloc) and/or coarse-grain
advise are mandatory, not optional.  This tool measures every arm:

  arm "naive":     hipMallocManaged, device first touch
  arm "prefetch":  + hipMemPrefetchAsync(device) before first touch
  arm "coarse":    + hipMemAdvise(SetCoarseGrain) before first touch
  arm "pref+coarse"
and for each: resident re-touch, host pull-back, GPU refault, and
prefetch-restore rates, plus plain-HBM and host-RAM baselines.

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


def host_touch(arr, n):
    step = 1024  # one touch per 4 KiB page
    for i in range(0, n, step):
        arr[i] = 2.0


def run_arm(h, nbytes, n, gb, *, coarse=False, prefetch=False,
            preferred=False):
    res = {}
    ptr = h.malloc_managed(nbytes)
    arr = (ctypes.c_float * n).from_address(ptr)
    try:
        if coarse:
            h.advise(ptr, nbytes, h.ADVISE_COARSE_GRAIN, 0)
        if preferred:
            h.advise(ptr, nbytes, h.ADVISE_PREFERRED_LOCATION, 0)
        if prefetch:
            res["alloc_prefetch_s"] = timed(
                lambda: (h.prefetch(ptr, nbytes, 0), h.sync()))
        res["first_touch_gbs"] = gb / timed(
            lambda: (h.touch_pages(ptr, n, 1, 1.0), h.sync()))
        res["resident_touch_gbs"] = gb / timed(
            lambda: (h.touch_pages(ptr, n, 1, 1.0), h.sync()))
        res["host_pull_gbs"] = gb / timed(lambda: host_touch(arr, n))
        res["gpu_refault_gbs"] = gb / timed(
            lambda: (h.touch_pages(ptr, n, 1, 1.0), h.sync()))
        host_touch(arr, n)
        res["prefetch_restore_gbs"] = gb / timed(
            lambda: (h.prefetch(ptr, nbytes, 0),
                     h.touch_pages(ptr, n, 1, 1.0), h.sync()))
    finally:
        h.free(ptr)
    return res


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gb", type=float, default=4.0)
    ap.add_argument("--out", default="profiles/faultbench.json")
    ap.add_argument("--arms", default="naive,prefetch,coarse,pref+coarse")
    args = ap.parse_args()

    h = hiputil.load()
    nbytes = int(args.gb * (1 << 30))
    n = nbytes // 4
    res: dict = {"gb": args.gb, "arms": {}}

    arm_flags = {
        "naive": {},
        "prefetch": {"prefetch": True},
        "coarse": {"coarse": True},
        "pref+coarse": {"coarse": True, "prefetch": True},
        "preferred": {"preferred": True},
    }
    for arm in args.arms.split(","):
        res["arms"][arm] = run_arm(h, nbytes, n, args.gb,
                                   **arm_flags[arm])
        print(arm, json.dumps(res["arms"][arm]), flush=True)

    # plain HBM baseline
    ptr2 = h.malloc(nbytes)
    h.touch_pages(ptr2, n, 1, 1.0)
    h.sync()
    res["hbm_touch_gbs"] = args.gb / timed(
        lambda: (h.touch_pages(ptr2, n, 1, 1.0), h.sync()))
    h.free(ptr2)

    out = Path(args.out)
    out.parent.mkdir(parents=True, exist_ok=True)
    out.write_text(json.dumps(res, indent=2))
    print(json.dumps(res, indent=2))


if __name__ == "__main__":
    main()
