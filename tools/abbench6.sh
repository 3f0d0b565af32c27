#!/bin/bash
# Attribution round 6: does the managed free cache close the step gap?
set -x
mkdir -p gpurun_out
export HSA_XNACK=1 MIOPEN_FIND_MODE=FAST

TRAIN='
import sys; sys.path.insert(0, ".")
from nvshare_amd.workloads.train_resnet import run_training
import math
r = run_training("resnet50", "cuda", batch=32, image=224, steps=60,
                 warmup=10)
assert math.isfinite(r["loss"]), r
assert r["loss_first"] < 20, r
print("ARM", round(r["samples_per_s"], 1), "loss", round(r["loss"], 3))
'
timeout 300 env -u LD_PRELOAD python -c "
import sys; sys.path.insert(0, '.')
from nvshare_amd.workloads.train_resnet import run_training
run_training('resnet50', 'cuda', batch=32, image=224, steps=3, warmup=3)
print('WARMED')" > gpurun_out/ab6_warm.log 2>&1

: > gpurun_out/ab6.log
run_arm () {
  local label="$1" pre="$2" envs="$3"
  timeout 200 env $envs $pre python -c "$TRAIN" 2>&1 \
    | sed "s/^ARM/$label/" >> gpurun_out/ab6.log
  echo "$label rc=${PIPESTATUS[0]}" >> gpurun_out/ab6.log
}
STOCK="env -u LD_PRELOAD"
HOOK="python -m nvshare_amd.run --standalone --"

run_arm stock1    "$STOCK" "IGNORE="
run_arm hooked1   "$HOOK"  "IGNORE="
run_arm hk_nocache "$HOOK" "NVSHARE_FREE_CACHE_MIB=0"
run_arm hooked2   "$HOOK"  "IGNORE="
run_arm stock2    "$STOCK" "IGNORE="
run_arm hooked3   "$HOOK"  "IGNORE="
grep -E "rc=|loss" gpurun_out/ab6.log

# Cache behavior census (how many recycles vs real allocations).
timeout 200 env NVSHARE_DEBUG=1 python -m nvshare_amd.run --standalone -- \
  python -c "$TRAIN" > gpurun_out/ab6_dbg.log 2>&1
grep -c "cached free" gpurun_out/ab6_dbg.log
grep "hook call counts" gpurun_out/ab6_dbg.log
