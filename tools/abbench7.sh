#!/bin/bash
# Attribution round 7: do host-VM knobs (automatic NUMA balancing, THP
# defrag/khugepaged) cause the continuous HMM refault storms?  Any
# host-side PTE invalidation on a managed VMA tears down the device
# mapping via mmu-notifiers and forces an XNACK refault — autonuma's
# periodic prot_none scanning would do exactly that, continuously.
set -x
mkdir -p gpurun_out
export HSA_XNACK=1 MIOPEN_FIND_MODE=FAST

echo "numa_balancing=$(cat /proc/sys/kernel/numa_balancing)" \
  | tee gpurun_out/ab7_knobs.log
cat /sys/kernel/mm/transparent_hugepage/enabled >> gpurun_out/ab7_knobs.log
cat /sys/kernel/mm/transparent_hugepage/defrag >> gpurun_out/ab7_knobs.log

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
print('WARMED')" > gpurun_out/ab7_warm.log 2>&1

: > gpurun_out/ab7.log
run_arm () {
  local label="$1" pre="$2"
  timeout 200 env $pre python -c "$TRAIN" 2>&1 \
    | sed "s/^ARM/$label/" >> gpurun_out/ab7.log
  echo "$label rc=${PIPESTATUS[0]}" >> gpurun_out/ab7.log
}
STOCK="env -u LD_PRELOAD"
HOOK="python -m nvshare_amd.run --standalone --"

run_arm stock_b   "$STOCK"
run_arm hooked_b1 "$HOOK"
run_arm hooked_b2 "$HOOK"

# Flip the knobs.
echo 0 > /proc/sys/kernel/numa_balancing || true
echo never > /sys/kernel/mm/transparent_hugepage/defrag || true
echo "after: numa_balancing=$(cat /proc/sys/kernel/numa_balancing)" \
  >> gpurun_out/ab7_knobs.log

run_arm hooked_a1 "$HOOK"
run_arm hooked_a2 "$HOOK"
run_arm stock_a   "$STOCK"

# Isolate which knob: re-enable THP defrag, keep autonuma off.
echo madvise > /sys/kernel/mm/transparent_hugepage/defrag || true
run_arm hooked_thpdef "$HOOK"
# autonuma back on, defrag off.
echo 1 > /proc/sys/kernel/numa_balancing || true
echo never > /sys/kernel/mm/transparent_hugepage/defrag || true
run_arm hooked_numaon "$HOOK"

grep -E "rc=|loss" gpurun_out/ab7.log
cat gpurun_out/ab7_knobs.log
