#!/usr/bin/env python3
"""Per-step NaN localization for the flagship ResNet training loop.

Round-1 fresh-box smoke/bench reported NaN loss (GPUTEST_r01.json)
while earlier lease runs were finite; the timeline implicates either
MIOpen algo choice (cold full-find vs warm db vs FIND_MODE=FAST) or
the managed-memory fast path (coarse-grain advise / eager prefetch,
commits b240c74..fcff6d0).  This tool runs the exact bench step loop,
checks the loss EVERY step, and on the first non-finite value dumps
which tensors went bad (input, logits, per-layer grads/params) so the
failure can be attributed to divergence (late, loss grows first) vs
corruption (abrupt, a single layer's buffers).

Run under the arm matrix of tools/nanhunt.sh.
"""

from __future__ import annotations

import argparse
import json
import math
import sys


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--image", type=int, default=224)
    ap.add_argument("--steps", type=int, default=40)
    ap.add_argument("--lr", type=float, default=0.02)
    ap.add_argument("--momentum", type=float, default=0.9)
    ap.add_argument("--dtype", default="bfloat16")
    ap.add_argument("--label", default="arm")
    ap.add_argument("--seed", type=int, default=1234)
    args = ap.parse_args()

    import torch

    sys.path.insert(0, ".")
    from nvshare_amd.workloads.train_resnet import build

    dev = torch.device("cuda:0")
    amp = getattr(torch, args.dtype) if args.dtype != "float32" else None
    torch.manual_seed(args.seed)
    torch.backends.cudnn.benchmark = True
    model = build(args.model, 1000).to(dev,
                                       memory_format=torch.channels_last)
    opt = torch.optim.SGD(model.parameters(), lr=args.lr,
                          momentum=args.momentum)
    lossf = torch.nn.CrossEntropyLoss()
    x = torch.randn(args.batch, 3, args.image, args.image,
                    device=dev).to(memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (args.batch,), device=dev)

    losses = []
    for i in range(args.steps):
        opt.zero_grad(set_to_none=True)
        if amp is not None:
            with torch.autocast("cuda", dtype=amp):
                logits = model(x)
                loss = lossf(logits, y)
        else:
            logits = model(x)
            loss = lossf(logits, y)
        loss.backward()
        lv = float(loss.detach().float().cpu())
        losses.append(round(lv, 4))
        if not math.isfinite(lv):
            torch.cuda.synchronize()
            print(f"[{args.label}] NONFINITE at step {i}: loss={lv}",
                  flush=True)
            bad = {}
            bad["x"] = bool(torch.isnan(x).any() or torch.isinf(x).any())
            bad["logits"] = bool(torch.isnan(logits).any().item()
                                 or torch.isinf(logits).any().item())
            for name, p in model.named_parameters():
                pn = bool(torch.isnan(p).any().item()
                          or torch.isinf(p).any().item())
                gn = (bool(torch.isnan(p.grad).any().item()
                           or torch.isinf(p.grad).any().item())
                      if p.grad is not None else None)
                if pn or gn:
                    bad[name] = {"param": pn, "grad": gn}
            for name, b in model.named_buffers():
                bn = bool(torch.isnan(b).any().item()
                          or torch.isinf(b).any().item())
                if bn:
                    bad["buf:" + name] = True
            print(f"[{args.label}] bad tensors: "
                  f"{json.dumps(bad, default=str)[:4000]}", flush=True)
            print(f"[{args.label}] losses: {losses}", flush=True)
            return 9
        opt.step()
    torch.cuda.synchronize()
    print(f"[{args.label}] FINITE {args.steps} steps; losses: {losses}",
          flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
