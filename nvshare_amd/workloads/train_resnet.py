"""ResNet training job on synthetic data (co-location benchmark unit).

One "step" = forward + loss + backward + SGD update on a synthetic
batch.  Emits steps/s and samples/s.  This is the flagship workload of
bench.py (BASELINE.json config #4: co-located ResNet-50 training).
"""

from __future__ import annotations

import argparse

from nvshare_amd.workloads.common import Timer, add_common_args, emit, sync


def build(model_name: str, num_classes: int):
    from nvshare_amd.workloads import resnet

    if model_name == "resnet50":
        return resnet.resnet50(num_classes)
    if model_name == "resnet152":
        return resnet.resnet152(num_classes)
    if model_name == "tiny":
        return resnet.tiny_resnet(num_classes)
    raise ValueError(f"unknown model {model_name}")


def run_training(model_name: str = "resnet50", device: str = "cuda",
                 batch: int = 64, image: int = 224, steps: int = 50,
                 warmup: int = 5, dtype: str = "bfloat16",
                 num_classes: int = 1000, lr: float = 0.02) -> dict:
    import torch

    dev = torch.device(device)
    amp_dtype = getattr(torch, dtype) if dtype != "float32" else None
    torch.manual_seed(0)
    if device.startswith("cuda"):
        # MIOpen autotune + NHWC: the fast conv path on CDNA.
        torch.backends.cudnn.benchmark = True
        model = build(model_name, num_classes).to(
            dev, memory_format=torch.channels_last)
    else:
        model = build(model_name, num_classes).to(dev)
    # lr 0.02 descends cleanly for 40+ bf16 steps on random labels
    # (profiles/nanhunt.log stock arm: 7.17 -> 0.003); 0.1 risks
    # early-step blowup on some seeds.
    opt = torch.optim.SGD(model.parameters(), lr=lr, momentum=0.9)
    lossf = torch.nn.CrossEntropyLoss()
    x = torch.randn(batch, 3, image, image, device=dev)
    if device.startswith("cuda"):
        x = x.to(memory_format=torch.channels_last)
    y = torch.randint(0, num_classes, (batch,), device=dev)

    def step():
        opt.zero_grad(set_to_none=True)
        if amp_dtype is not None and device.startswith("cuda"):
            with torch.autocast("cuda", dtype=amp_dtype):
                loss = lossf(model(x), y)
        else:
            loss = lossf(model(x), y)
        loss.backward()
        opt.step()
        return loss

    # First loss before any update: a healthy random-init CE loss is
    # ~ln(num_classes); the round-1 alloc-prefetch corruption showed up
    # here as ~930 (profiles/nanhunt.log).  Recorded so callers can
    # assert sanity, not just finiteness.
    first_loss = None
    for i in range(warmup):
        loss = step()
        if i == 0:
            first_loss = float(loss.detach().float().cpu())
    sync(device)
    with Timer() as t:
        for _ in range(steps):
            loss = step()
        sync(device)
    final_loss = float(loss.detach().float().cpu())
    if first_loss is None:
        first_loss = final_loss
    return {
        "workload": "train_resnet", "model": model_name,
        "seconds": t.seconds, "steps": steps, "batch": batch,
        "image": image, "device": device, "dtype": dtype,
        "steps_per_s": steps / t.seconds,
        "samples_per_s": steps * batch / t.seconds,
        "loss": final_loss, "loss_first": first_loss,
    }


def main(argv: list[str] | None = None) -> None:
    ap = argparse.ArgumentParser()
    add_common_args(ap)
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--image", type=int, default=224)
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--dtype", default="bfloat16")
    ap.add_argument("--num-classes", type=int, default=1000)
    ap.add_argument("--lr", type=float, default=0.02)
    args = ap.parse_args(argv)
    res = run_training(args.model, args.device, args.batch, args.image,
                       args.steps, args.warmup, args.dtype,
                       args.num_classes, args.lr)
    res["label"] = args.label
    emit(res)


if __name__ == "__main__":
    main()
