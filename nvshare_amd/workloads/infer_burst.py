"""Bursty inference job (BASELINE.json config #5: Jupyter-style pods).

Alternates short inference bursts with idle think-time, the pattern
nvshare's early-release path is built for: during think-time the client
voluntarily releases the GPU lock so other pods run.
"""

from __future__ import annotations

import argparse
import time

from nvshare_amd.workloads.common import Timer, add_common_args, emit, sync


def main(argv: list[str] | None = None) -> None:
    ap = argparse.ArgumentParser()
    add_common_args(ap)
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--batch", type=int, default=16)
    ap.add_argument("--image", type=int, default=224)
    ap.add_argument("--bursts", type=int, default=5)
    ap.add_argument("--infers-per-burst", type=int, default=20)
    ap.add_argument("--think-s", type=float, default=2.0)
    args = ap.parse_args(argv)

    import torch

    from nvshare_amd.workloads.train_resnet import build

    dev = torch.device(args.device)
    model = build(args.model, 1000).to(dev).eval()
    x = torch.randn(args.batch, 3, args.image, args.image, device=dev)

    infers = 0
    with Timer() as t:
        with torch.no_grad():
            for _ in range(args.bursts):
                for _ in range(args.infers_per_burst):
                    model(x)
                    infers += 1
                sync(args.device)
                time.sleep(args.think_s)
    emit({
        "workload": "infer_burst", "label": args.label,
        "seconds": t.seconds, "bursts": args.bursts, "infers": infers,
        "batch": args.batch, "device": args.device,
        "infers_per_s": infers / t.seconds,
    })


if __name__ == "__main__":
    main()
