#!/bin/bash
# Round-2 verification + overhead attribution batch (run via gpurun).
#  1. Ordered ResNet-50 A/B: stock vs hooked (fixed defaults) with
#     passthrough-threshold and preferred-location ablations, warm
#     MIOpen db shared by every arm, stock repeated last (order ctrl).
#  2. rocprofv3 interposition-coverage crosscheck (now hard-fails on
#     empty profiler data).
#  3. Full GPU test suite + driver-style smoke.
set -x
mkdir -p gpurun_out
export HSA_XNACK=1 MIOPEN_FIND_MODE=FAST

TRAIN='
import sys; sys.path.insert(0, ".")
from nvshare_amd.workloads.train_resnet import run_training
r = run_training("resnet50", "cuda", batch=32, image=224, steps=60,
                 warmup=10)
import math
assert math.isfinite(r["loss"]), r
print("ARM", round(r["samples_per_s"], 1), "loss", round(r["loss"], 3))
'

# Shared warm MIOpen db for all arms.
timeout 300 env -u LD_PRELOAD python -c "
import sys; sys.path.insert(0, '.')
from nvshare_amd.workloads.train_resnet import run_training
run_training('resnet50', 'cuda', batch=32, image=224, steps=3, warmup=3)
print('WARMED')" > gpurun_out/ab_warm.log 2>&1

: > gpurun_out/ab.log
run_arm () {
  local label="$1" pre="$2" envs="$3"
  timeout 200 env $envs $pre python -c "$TRAIN" 2>&1 \
    | sed "s/^ARM/$label/" >> gpurun_out/ab.log
  echo "$label rc=${PIPESTATUS[0]}" >> gpurun_out/ab.log
}
STOCK="env -u LD_PRELOAD"
HOOK="python -m nvshare_amd.run --standalone --"

run_arm stock1        "$STOCK" "IGNORE="
run_arm hooked1       "$HOOK"  "IGNORE="
run_arm hk_pt4096     "$HOOK"  "NVSHARE_PASSTHROUGH_MIB=4096"
run_arm hk_noprefloc  "$HOOK"  "NVSHARE_PREFERRED_LOC=0"
run_arm stock2        "$STOCK" "IGNORE="
run_arm hooked2       "$HOOK"  "IGNORE="
run_arm hk_prof "$HOOK" "NVSHARE_DEBUG=1 NVSHARE_PROFILE_HOOKS=1"
grep -E "^(stock|hk|hooked)|wall time" gpurun_out/ab.log

# 2. Coverage crosscheck (writes gpurun_out/coverage.json).
cd /tmp && export TMPDIR=/tmp
timeout 600 python /root/repo/tools/coverage_check.py \
  --out /root/repo/gpurun_out/coverage.json \
  > /root/repo/gpurun_out/coverage.log 2>&1
echo "coverage rc=$?"
cd /root/repo

# 3. GPU test suite + smoke (driver-style).
timeout 900 python -m pytest tests/test_gpu.py -q -m gpu \
  > gpurun_out/pytest_gpu.log 2>&1
echo "pytest rc=$?"
tail -5 gpurun_out/pytest_gpu.log
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" \
  > gpurun_out/smoke.log 2>&1
echo "smoke rc=$?"
tail -3 gpurun_out/smoke.log
