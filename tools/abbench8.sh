#!/bin/bash
# Attribution round 8: residency.  (a) Sample VRAM/GTT during stock vs
# hooked runs — if the hooked working set stays host/GTT-resident the
# PCIe path explains the ~1.4x step cost.  (b) Test device-side
# zero-fill population at alloc (NVSHARE_ALLOC_MEMSET=1): pages born
# in HBM, synced before the pointer escapes — prefetch-free.
set -x
mkdir -p gpurun_out
export HSA_XNACK=1 MIOPEN_FIND_MODE=FAST

sample () { # $1 = tag; samples until killed
  while :; do
    v=$(cat /sys/class/drm/card*/device/mem_info_vram_used 2>/dev/null | awk '{s+=$1} END{print s+0}')
    g=$(cat /sys/class/drm/card*/device/mem_info_gtt_used 2>/dev/null | awk '{s+=$1} END{print s+0}')
    echo "$1 $(date +%s.%N) vram=$v gtt=$g" >> gpurun_out/ab8_res.log
    sleep 0.3
  done
}

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
print('WARMED')" > gpurun_out/ab8_warm.log 2>&1

: > gpurun_out/ab8.log
: > gpurun_out/ab8_res.log
run_arm () {
  local label="$1" pre="$2" envs="$3"
  sample "$label" & local sp=$!
  timeout 200 env $envs $pre python -c "$TRAIN" 2>&1 \
    | sed "s/^ARM/$label/" >> gpurun_out/ab8.log
  echo "$label rc=${PIPESTATUS[0]}" >> gpurun_out/ab8.log
  kill $sp 2>/dev/null
}
STOCK="env -u LD_PRELOAD"
HOOK="python -m nvshare_amd.run --standalone --"

run_arm stock1    "$STOCK" "IGNORE="
run_arm hooked1   "$HOOK"  "IGNORE="
run_arm hk_memset "$HOOK"  "NVSHARE_ALLOC_MEMSET=1"
run_arm hk_memset2 "$HOOK" "NVSHARE_ALLOC_MEMSET=1"
run_arm stock2    "$STOCK" "IGNORE="
run_arm hooked2   "$HOOK"  "IGNORE="
grep -E "rc=|loss" gpurun_out/ab8.log

# peak vram/gtt per arm
for t in stock1 hooked1 hk_memset hk_memset2 stock2 hooked2; do
  echo "$t peak: $(grep "^$t " gpurun_out/ab8_res.log | awk '{print $3, $4}' | \
    sed 's/[a-z=]*//g' | awk '{if($1>v)v=$1; if($2>g)g=$2} END{printf "vram=%.1fGB gtt=%.1fGB\n", v/1e9, g/1e9}')"
done
