"""Elementwise-add workload (port of reference tests/pytorch-add.py).

Allocates two NxN fp32 tensors and repeatedly adds them in place,
verifying the result.  The reference used N=28000 (~6.3 GB) x 4000
iters, N=14000 x 40000 for the -small variant (pytorch-add.py:28-37);
defaults here are sized for a 288 GB MI355X via --gb.
"""

from __future__ import annotations

import argparse
import math

from nvshare_amd.workloads.common import Timer, add_common_args, die, emit, sync


def main(argv: list[str] | None = None) -> None:
    ap = argparse.ArgumentParser()
    add_common_args(ap)
    ap.add_argument("--gb", type=float, default=12.0,
                    help="approx total working set in GiB (two tensors)")
    ap.add_argument("--iters", type=int, default=1000)
    ap.add_argument("--think-every", type=int, default=0,
                    help="sleep --think-s after every N iters (emulates "
                    "the reference's GPU/CPU-mixed *_50 workloads, "
                    "thesis Table 12.1)")
    ap.add_argument("--think-s", type=float, default=2.0)
    args = ap.parse_args(argv)

    import torch

    import time

    n = int(math.sqrt(args.gb * (1 << 30) / 2 / 4))
    dev = torch.device(args.device)
    with Timer() as t:
        x = torch.ones((n, n), dtype=torch.float32, device=dev)
        y = torch.zeros((n, n), dtype=torch.float32, device=dev)
        for i in range(args.iters):
            y.add_(x)
            if args.think_every and (i + 1) % args.think_every == 0:
                sync(args.device)
                time.sleep(args.think_s)
        sync(args.device)
        expect = float(args.iters)
        got = y[n // 2, n // 2].item()
    if got != expect:
        die(f"pytorch_add: got {got}, expected {expect}")
    emit({
        "workload": "pytorch_add", "label": args.label,
        "seconds": t.seconds, "n": n, "iters": args.iters,
        "device": args.device,
        "gib": 2 * n * n * 4 / (1 << 30),
    })


if __name__ == "__main__":
    main()
