#!/bin/bash
# Overhead attribution round 3 (run via gpurun): copy-prefetch v2
# (quiesce+sync, once per range), advise/THP ablations, and a working
# rocprofv3 kernel-stats diff stock-vs-hooked.
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
print("ARM", round(r["samples_per_s"], 1), "loss", round(r["loss"], 3),
      "first", round(r["loss_first"], 2))
'

timeout 300 env -u LD_PRELOAD python -c "
import sys; sys.path.insert(0, '.')
from nvshare_amd.workloads.train_resnet import run_training
run_training('resnet50', 'cuda', batch=32, image=224, steps=3, warmup=3)
print('WARMED')" > gpurun_out/ab3_warm.log 2>&1

: > gpurun_out/ab3.log
run_arm () {
  local label="$1" pre="$2" envs="$3"
  local t0=$(date +%s.%N)
  timeout 200 env $envs $pre python -c "$TRAIN" 2>&1 \
    | sed "s/^ARM/$label/" >> gpurun_out/ab3.log
  local rc=${PIPESTATUS[0]}
  echo "$label rc=$rc wall=$(awk "BEGIN{printf \"%.1f\", $(date +%s.%N)-$t0}")" \
    >> gpurun_out/ab3.log
}
STOCK="env -u LD_PRELOAD"
HOOK="python -m nvshare_amd.run --standalone --"

run_arm stock1     "$STOCK" "IGNORE="
run_arm hooked1    "$HOOK"  "IGNORE="
run_arm hk_thp     "$HOOK"  "NVSHARE_THP=1"
run_arm hk_nocg    "$HOOK"  "NVSHARE_COARSE_GRAIN=0"
run_arm hk_plain   "$HOOK"  "NVSHARE_COARSE_GRAIN=0 NVSHARE_PREFERRED_LOC=0"
run_arm hk_nocopy  "$HOOK"  "NVSHARE_COPY_PREFETCH=0"
run_arm stock2     "$STOCK" "IGNORE="
run_arm hooked2    "$HOOK"  "IGNORE="
grep -E "rc=|loss" gpurun_out/ab3.log

# rocprofv3 kernel stats, stock vs hooked (20 timed steps each).
cd /tmp && export TMPDIR=/tmp
export PYTHONPATH=/root/repo
SHORT='
import sys
from nvshare_amd.workloads.train_resnet import run_training
r = run_training("resnet50", "cuda", batch=32, image=224, steps=20,
                 warmup=5)
print("PROF", round(r["samples_per_s"], 1))
'
timeout 300 env -u LD_PRELOAD rocprofv3 --kernel-trace --stats \
  -d /tmp/prof_stock -o st -- python -c "$SHORT" \
  > /root/repo/gpurun_out/prof_stock.log 2>&1
echo "prof_stock rc=$?"
timeout 300 python -m nvshare_amd.run --standalone -- \
  rocprofv3 --kernel-trace --stats -d /tmp/prof_hooked -o hk -- \
  python -c "$SHORT" > /root/repo/gpurun_out/prof_hooked.log 2>&1
echo "prof_hooked rc=$?"
find /tmp/prof_stock /tmp/prof_hooked -name "*.csv" > /root/repo/gpurun_out/prof_files.txt
cp $(grep kernel_stats /root/repo/gpurun_out/prof_files.txt | grep stock | head -1) \
  /root/repo/gpurun_out/kstats_stock.csv
cp $(grep kernel_stats /root/repo/gpurun_out/prof_files.txt | grep hooked | head -1) \
  /root/repo/gpurun_out/kstats_hooked.csv
grep PROF /root/repo/gpurun_out/prof_stock.log \
  /root/repo/gpurun_out/prof_hooked.log
