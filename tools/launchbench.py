#!/usr/bin/env python3
"""Per-launch overhead: plain vs managed(+advise/prefetch) pointers.

ResNet-50 under the managed conversion showed ~1.29x while matmul
showed ~1.02x; the difference tracks launch density, suggesting a
per-launch runtime cost for kernels referencing managed memory.  This
isolates it.

Usage (GPU box): python tools/launchbench.py [--n 20000]
"""

from __future__ import annotations

import argparse
import json
import sys
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

from nvshare_amd import hiputil  # noqa: E402


def per_launch_us(h, ptr, n):
    h.lib.nvs_launch_burst(None, 0)  # warm nothing
    # warmup
    assert h.lib.nvs_launch_burst(ptr, 200) == 0
    t0 = time.monotonic()
    assert h.lib.nvs_launch_burst(ptr, n) == 0
    return (time.monotonic() - t0) / n * 1e6


def main() -> None:
    import ctypes

    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=20000)
    ap.add_argument("--out", default="profiles/launchbench.json")
    args = ap.parse_args()

    h = hiputil.load()
    res = {}

    p = h.malloc(1 << 20)
    res["plain_us"] = per_launch_us(h, ctypes.c_void_p(p), args.n)
    h.free(p)

    p = h.malloc_managed(1 << 20)
    h.touch_pages(p, 256, 1, 1.0)
    h.sync()
    res["managed_finegrain_us"] = per_launch_us(
        h, ctypes.c_void_p(p), args.n)
    h.free(p)

    p = h.malloc_managed(1 << 20)
    h.advise(p, 1 << 20, h.ADVISE_COARSE_GRAIN, 0)
    h.prefetch(p, 1 << 20, 0)
    h.sync()
    res["managed_coarse_prefetched_us"] = per_launch_us(
        h, ctypes.c_void_p(p), args.n)
    h.free(p)

    # Does the penalty scale with the NUMBER of live managed ranges?
    # (a torch caching allocator holds dozens of segments)
    for count in (16, 64, 256):
        ptrs = []
        for _ in range(count):
            q = h.malloc_managed(1 << 20)
            h.advise(q, 1 << 20, h.ADVISE_COARSE_GRAIN, 0)
            h.prefetch(q, 1 << 20, 0)
            ptrs.append(q)
        h.sync()
        res[f"managed_{count}_ranges_us"] = per_launch_us(
            h, ctypes.c_void_p(ptrs[0]), args.n)
        for q in ptrs:
            h.free(q)

    # Same count of PLAIN allocations for the control arm.
    ptrs = [h.malloc(1 << 20) for _ in range(256)]
    res["plain_256_ranges_us"] = per_launch_us(
        h, ctypes.c_void_p(ptrs[0]), args.n)
    for q in ptrs:
        h.free(q)

    res["managed_penalty_us"] = (res["managed_coarse_prefetched_us"]
                                 - res["plain_us"])
    out = Path(args.out)
    out.parent.mkdir(parents=True, exist_ok=True)
    out.write_text(json.dumps(res, indent=2))
    print(json.dumps(res, indent=2))


if __name__ == "__main__":
    main()
