"""Matmul workload (port of reference tests/tf-matmul.py, on PyTorch).

Repeated NxN matmul with a correctness probe.  The reference used
N=35000 fp32 x 10 iters (tf-matmul.py:37-51); MI355X default is sized
by --gb and runs on rocBLAS through PyTorch.
"""

from __future__ import annotations

import argparse
import math

from nvshare_amd.workloads.common import Timer, add_common_args, die, emit, sync


def main(argv: list[str] | None = None) -> None:
    ap = argparse.ArgumentParser()
    add_common_args(ap)
    ap.add_argument("--gb", type=float, default=12.0,
                    help="approx working set in GiB (three tensors)")
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--dtype", default="float32",
                    choices=["float32", "bfloat16"])
    args = ap.parse_args(argv)

    import torch

    dtype = getattr(torch, args.dtype)
    esz = torch.tensor([], dtype=dtype).element_size()
    n = int(math.sqrt(args.gb * (1 << 30) / 3 / esz))
    dev = torch.device(args.device)
    with Timer() as t:
        # a is identity-scaled so the product is checkable exactly.
        a = torch.eye(n, dtype=dtype, device=dev) * 2
        b = torch.ones((n, n), dtype=dtype, device=dev)
        c = b
        for _ in range(args.iters):
            c = a @ b
        sync(args.device)
        got = float(c[n // 2, n // 2])
    if got != 2.0:
        die(f"matmul: got {got}, expected 2.0")
    flops = 2 * (n ** 3) * args.iters
    emit({
        "workload": "matmul", "label": args.label,
        "seconds": t.seconds, "n": n, "iters": args.iters,
        "device": args.device, "dtype": args.dtype,
        "tflops": flops / t.seconds / 1e12 if t.seconds > 0 else 0.0,
    })


if __name__ == "__main__":
    main()
