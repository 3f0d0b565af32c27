#!/usr/bin/env python3
"""Working-set restore bandwidth: how fast can a preempted client's
managed memory come back to the device?

The post-preemption restore path bounds the useful TQ under
oversubscription (docs/tuning.md).  Measures hipMemPrefetchAsync
restore with 1/2/4/8 concurrent streams and several chunk sizes, after
the buffer has been pulled to host.

Usage (GPU box): python tools/restorebench.py [--gb 4] [--out profiles/restorebench.json]
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


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gb", type=float, default=4.0)
    ap.add_argument("--out", default="profiles/restorebench.json")
    args = ap.parse_args()

    h = hiputil.load()
    nbytes = int(args.gb * (1 << 30))
    n = nbytes // 4
    res: dict = {"gb": args.gb, "rows": {}}

    ptr = h.malloc_managed(nbytes)
    arr = (ctypes.c_float * n).from_address(ptr)
    h.advise(ptr, nbytes, h.ADVISE_COARSE_GRAIN, 0)
    h.prefetch(ptr, nbytes, 0)
    h.touch_pages(ptr, n, 1, 1.0)
    h.sync()

    def pull_to_host():
        # Evict via explicit prefetch to CPU (what NVSHARE_EVICT would
        # do on DROP_LOCK), much faster than CPU-touch faulting.
        t0 = time.monotonic()
        h.prefetch(ptr, nbytes, -1)  # hipCpuDeviceId
        h.sync()
        return time.monotonic() - t0

    res["rows"]["evict_prefetch_to_host_gbs"] = args.gb / pull_to_host()

    for streams in (1, 2, 4, 8):
        for chunk_mib in (64, 256, 1024):
            pull_to_host()
            t0 = time.monotonic()
            h.prefetch_chunked(ptr, nbytes, 0, streams,
                               chunk_mib << 20)
            dt = time.monotonic() - t0
            key = f"restore_s{streams}_c{chunk_mib}MiB_gbs"
            res["rows"][key] = args.gb / dt
            print(key, round(args.gb / dt, 2), flush=True)

    # Verify data integrity after all the migration.
    assert arr[0] == 1.0 and arr[n - 1] == 1.0
    h.free(ptr)

    out = Path(args.out)
    out.parent.mkdir(parents=True, exist_ok=True)
    out.write_text(json.dumps(res, indent=2))
    print(json.dumps(res, indent=2))


if __name__ == "__main__":
    main()
