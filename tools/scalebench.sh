#!/bin/bash
# Round-2 scaling + thrashing batch (run via gpurun from the repo root).
#  1. N-client co-location curve on ONE MI355X: bench.py at N=1,2,4,8
#     exactly as the driver launches it (torchrun, one rank per client,
#     all sharing GPU 0).
#  2. Thrashing-parity table (BASELINE.md §3 / thesis Tables 11.7-11.8):
#     ResNet batch sweep, solo vs 2x parallel, squatter-pressured so 2x
#     oversubscribes; peak VRAM + time per cell.
set -x
mkdir -p gpurun_out
export HSA_XNACK=1 MIOPEN_FIND_MODE=FAST

# 0. XNACK-mode arm: is the steady ~1.35x managed tax the XNACK
#    retry-fault servicing itself?  HSA_XNACK=0 managed memory uses
#    legacy whole-buffer residency (no retry faults).
TRAIN='
import sys; sys.path.insert(0, ".")
from nvshare_amd.workloads.train_resnet import run_training
import math
r = run_training("resnet50", "cuda", batch=32, image=224, steps=60,
                 warmup=10)
assert math.isfinite(r["loss"]), r
print("ARM", round(r["samples_per_s"], 1), "loss", round(r["loss"], 3))
'
timeout 300 env -u LD_PRELOAD python -c "
import sys; sys.path.insert(0, '.')
from nvshare_amd.workloads.train_resnet import run_training
run_training('resnet50', 'cuda', batch=32, image=224, steps=3, warmup=3)
print('WARMED')" > gpurun_out/sc_warm.log 2>&1
: > gpurun_out/xnack.log
for arm in "stock_x1 env -u LD_PRELOAD" "hooked_x1 python -m nvshare_amd.run --standalone --"; do
  set -- $arm; label=$1; shift
  timeout 200 env HSA_XNACK=1 "$@" python -c "$TRAIN" 2>&1 | sed "s/^ARM/$label/" >> gpurun_out/xnack.log
done
for arm in "stock_x0 env -u LD_PRELOAD" "hooked_x0 python -m nvshare_amd.run --standalone --" "hooked_x0b python -m nvshare_amd.run --standalone --"; do
  set -- $arm; label=$1; shift
  timeout 200 env HSA_XNACK=0 "$@" python -c "$TRAIN" 2>&1 | sed "s/^ARM/$label/" >> gpurun_out/xnack.log
done
grep -E "^stock|^hooked" gpurun_out/xnack.log

# 1. N-client curve (K scales with N so per-rank work stays constant).
for N in 1 2 4 8; do
  timeout 420 python -m torch.distributed.run --nnodes=1 \
    --nproc-per-node $N --master-addr 127.0.0.1 --master-port 29512 \
    bench.py --gpus $N --steps 24 --warmup 6 \
    > gpurun_out/bn${N}_r2.json 2> gpurun_out/bn${N}_r2.err
  echo "bn$N rc=$?"
  grep -o '{.*}' gpurun_out/bn${N}_r2.json | tail -1
done

# 2. Thrashing table: squatter leaves ~12 GB free (the reference's
#    P100 had 16 GB); ResNet-50 batch sweep so solo fits and 2x
#    oversubscribes, mirroring thesis Tables 11.7-11.8's shape.
timeout 1500 python tools/thrashbench.py \
  --batches 32,64,96 --steps 40 --tq 10 --include-off \
  --off-timeout 300 --squat-leave-gb 12 \
  --out gpurun_out/thrashbench.json > gpurun_out/thrash.log 2>&1
echo "thrash rc=$?"
tail -8 gpurun_out/thrash.log
