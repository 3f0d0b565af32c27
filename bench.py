#!/usr/bin/env python3
"""nvshare-amd flagship benchmark: N co-located training jobs on 1 MI355X.

Measures the headline metric of BASELINE.json: aggregate throughput /
makespan of N co-located PyTorch ResNet-50 training jobs sharing ONE
MI355X through the nvshare-amd stack (LD_PRELOAD interposer: hipMalloc
-> hipMallocManaged + scheduler lock; FCFS/TQ scheduler daemon).

Driver contract:
    python bench.py --gpus N --steps K --warmup W
N>1 is launched via torch.distributed.run (one rank per "GPU" slot);
every rank is one nvshare CLIENT pinned to physical GPU 0, so the
scaling curve is the 1/2/4/8-co-located-clients curve BASELINE.json
names (NOT multi-GPU data parallelism — the reference system is a
single-GPU sharing layer; rank-per-GPU would measure nothing of it).
Rank 0 prints exactly one JSON line.

The expected shape: aggregate samples/s stays ~flat as N grows (the
scheduler time-slices one GPU); per-N values vs N=1 give the sharing
efficiency; with working sets forced beyond HBM (--oversub-fake-mib)
the anti-thrashing TQ keeps the curve from collapsing.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

PRELOAD_GUARD = "NVSHARE_BENCH_CHILD"


def parse_args(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1,
                    help="number of co-located client ranks")
    ap.add_argument("--steps", type=int, default=32)
    ap.add_argument("--warmup", type=int, default=8)
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--image", type=int, default=224)
    ap.add_argument("--dtype", default="bfloat16")
    ap.add_argument("--tq", type=int,
                    default=int(os.environ.get("NVSHARE_BENCH_TQ", "2")),
                    help="scheduler time quantum for the run (s)")
    ap.add_argument("--device", default="auto",
                    help="auto|cuda|cpu (cpu = tiny CI mode)")
    ap.add_argument("--stock", action="store_true",
                    help="run WITHOUT the interposer (A/B baseline)")
    ap.add_argument("--oversub-fake-mib", type=int, default=0,
                    help="advertise this fake total to force "
                         "oversubscription behavior at small scale")
    return ap.parse_args(argv)


def pick_device(args) -> str:
    if args.device != "auto":
        return args.device
    # Decide WITHOUT importing torch in the parent (pre-exec) process:
    # /dev/kfd exists only on boxes with an AMD GPU.
    return "cuda" if os.path.exists("/dev/kfd") else "cpu"


def reexec_under_preload(args, device: str) -> None:
    """Re-exec this process as an nvshare client (before torch loads)."""
    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from nvshare_amd.env import client_env
    from nvshare_amd.scheduler import SchedulerDaemon
    from nvshare_amd import proto

    rank = int(os.environ.get("RANK", "0"))
    sock_dir = os.environ.get("NVSHARE_BENCH_SOCK_DIR",
                              "/tmp/nvshare-bench")
    spath = proto.scheduler_path(sock_dir)
    if rank == 0 and os.path.exists(spath):
        # Stale socket from a dead daemon? Probe it.
        try:
            from nvshare_amd import ctl
            ctl.status(sock_dir, timeout=2.0)
        except Exception:
            os.unlink(spath)
    if rank == 0 and not os.path.exists(proto.scheduler_path(sock_dir)):
        # Daemonize the scheduler for the duration of the bench; it is
        # torn down by the last rank (best effort) or just left idle.
        daemon = SchedulerDaemon(sock_dir=sock_dir, tq=args.tq,
                                 debug=False)
        daemon.start()
        # Detach so it survives this exec.
        daemon.proc = None
    else:
        deadline = time.monotonic() + 60
        while not os.path.exists(proto.scheduler_path(sock_dir)):
            if time.monotonic() > deadline:
                print("bench: scheduler socket never appeared",
                      file=sys.stderr)
                sys.exit(1)
            time.sleep(0.1)

    env = client_env(
        sock_dir=sock_dir,
        oversubscribe=args.oversub_fake_mib > 0,
        fake_total_mib=args.oversub_fake_mib or None,
        extra={
            PRELOAD_GUARD: "1",
            # deterministic MIOpen warmup on fresh boxes
            "MIOPEN_FIND_MODE": os.environ.get("MIOPEN_FIND_MODE",
                                               "FAST"),
            # all ranks co-locate on physical GPU 0
            "HIP_VISIBLE_DEVICES": os.environ.get(
                "NVSHARE_BENCH_GPU", "0"),
            "NVSHARE_RELEASE_INTERVAL_MS": "20",
            "NVSHARE_POD_NAME": f"bench-rank{rank}",
        },
    )
    os.execve(sys.executable, [sys.executable, os.path.abspath(__file__),
                               *sys.argv[1:]], env)


def main(argv=None):
    args = parse_args(argv)
    device = pick_device(args)

    if (device == "cuda" and not args.stock
            and os.environ.get(PRELOAD_GUARD) != "1"):
        reexec_under_preload(args, device)  # no return

    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    import torch
    import torch.distributed as dist

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29511")
        # gloo: coordination only — ranks share ONE GPU, there is no
        # inter-GPU collective in a GPU-sharing layer.
        dist.init_process_group("gloo", rank=rank, world_size=world)

    if device == "cuda":
        assert torch.cuda.is_available(), "no GPU visible"
        torch.cuda.set_device(0)
        model_name, batch, image = args.model, args.batch, args.image
        dtype = args.dtype
    else:
        model_name, batch, image, dtype = "tiny", 4, 32, "float32"

    from nvshare_amd.workloads.train_resnet import build

    dev = torch.device("cuda:0" if device == "cuda" else "cpu")
    amp_dtype = getattr(torch, dtype) if dtype != "float32" else None
    torch.manual_seed(1234 + rank)
    model = build(model_name, 1000 if device == "cuda" else 10).to(dev)
    if device == "cuda":
        torch.backends.cudnn.benchmark = True  # MIOpen autotune
        model = model.to(memory_format=torch.channels_last)
    # lr low enough that bf16 training on random labels stays finite
    # over any K the driver picks.
    opt = torch.optim.SGD(model.parameters(), lr=0.02, momentum=0.9)
    lossf = torch.nn.CrossEntropyLoss()
    x = torch.randn(batch, 3, image, image, device=dev)
    if device == "cuda":
        x = x.to(memory_format=torch.channels_last)
    y = torch.randint(0, model.fc.out_features, (batch,), device=dev)

    def step():
        opt.zero_grad(set_to_none=True)
        if amp_dtype is not None and device == "cuda":
            with torch.autocast("cuda", dtype=amp_dtype):
                loss = lossf(model(x), y)
        else:
            loss = lossf(model(x), y)
        loss.backward()
        opt.step()
        return loss

    for _ in range(args.warmup):
        step()
    if device == "cuda":
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()

    # Solo reference phase: rank 0 alone times a few steps while the
    # other ranks idle (their clients early-release the lock), giving
    # the denominator for the per-job-slowdown% half of the headline
    # metric (BASELINE.json: "makespan + per-job slowdown%").
    solo_ms = None
    k_solo = max(3, args.steps // 4)
    if rank == 0:
        if device == "cuda":
            time.sleep(0.5)  # let idle ranks hand the lock over
        ts = time.monotonic()
        for _ in range(k_solo):
            step()
        if device == "cuda":
            torch.cuda.synchronize()
        solo_ms = (time.monotonic() - ts) * 1000.0 / k_solo
    if world > 1:
        solo_l = [None]
        dist.broadcast_object_list(solo_l, src=0)
        solo_ms = solo_l[0] if rank != 0 else solo_ms
        dist.barrier()

    t0 = time.monotonic()
    for _ in range(args.steps):
        loss = step()
    if device == "cuda":
        torch.cuda.synchronize()
    my_time = time.monotonic() - t0
    if world > 1:
        dist.barrier()
    makespan = time.monotonic() - t0

    final_loss = float(loss.detach().float().cpu())
    if world > 1:
        times = [None] * world
        dist.all_gather_object(times, my_time)
        tmax = max(times)
    else:
        times = [my_time]
        tmax = my_time

    if rank == 0:
        # Whole-job aggregate: all ranks' samples over the makespan
        # (max rank time; barriers bracket the region).
        total_samples = args.steps * batch * world
        value = total_samples / tmax
        out = {
            "metric": "colocated_train_samples_per_s_total",
            "value": round(value, 2),
            "unit": "samples/s",
            "n_gpus": args.gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(tmax * 1000.0 / args.steps, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": dtype,
            "data": "synthetic",
            "config": {
                "model": model_name,
                "global_batch": batch * world,
                "image": image,
                "parallelism":
                    f"{world} co-located client(s) time-sharing 1 GPU "
                    f"via nvshare (tq={args.tq}s)"
                    + (", stock (no interposer)" if args.stock else ""),
                "per_rank_seconds": [round(t, 3) for t in times],
                "makespan_s": round(makespan, 3),
                # Per-job slowdown vs the ideal 1/N time share:
                # co-located per-rank time / (N x solo time) - 1.
                "solo_ms_per_step": round(solo_ms, 3),
                "per_job_slowdown_pct": [
                    round((t * 1000.0 / args.steps)
                          / (world * solo_ms) * 100.0 - 100.0, 1)
                    for t in times],
                "sharing_efficiency": round(
                    world * solo_ms
                    / (tmax * 1000.0 / args.steps), 4),
                "loss": round(final_loss, 4),
                "oversub_fake_mib": args.oversub_fake_mib,
            },
        }
        print(json.dumps(out), flush=True)

    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
