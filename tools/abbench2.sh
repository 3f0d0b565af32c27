#!/bin/bash
# Overhead attribution round 2 (run via gpurun): copy-prefetch fix in,
# advise/THP ablations, wall-clock per arm (init cost now counts), and
# a rocprofv3 kernel-stats diff stock-vs-hooked.
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
print("ARM", round(r["samples_per_s"], 1), "loss", round(r["loss"], 3))
'

timeout 300 env -u LD_PRELOAD python -c "
import sys; sys.path.insert(0, '.')
from nvshare_amd.workloads.train_resnet import run_training
run_training('resnet50', 'cuda', batch=32, image=224, steps=3, warmup=3)
print('WARMED')" > gpurun_out/ab2_warm.log 2>&1

: > gpurun_out/ab2.log
run_arm () {
  local label="$1" pre="$2" envs="$3"
  local t0=$(date +%s.%N)
  timeout 200 env $envs $pre python -c "$TRAIN" 2>&1 \
    | sed "s/^ARM/$label/" >> gpurun_out/ab2.log
  local rc=${PIPESTATUS[0]}
  echo "$label rc=$rc wall=$(awk "BEGIN{printf \"%.1f\", $(date +%s.%N)-$t0}")" \
    >> gpurun_out/ab2.log
}
STOCK="env -u LD_PRELOAD"
HOOK="python -m nvshare_amd.run --standalone --"

run_arm stock1     "$STOCK" "IGNORE="
run_arm hooked1    "$HOOK"  "IGNORE="
run_arm hk_thp     "$HOOK"  "NVSHARE_THP=1"
run_arm hk_nocg    "$HOOK"  "NVSHARE_COARSE_GRAIN=0"
run_arm hk_plain   "$HOOK"  "NVSHARE_COARSE_GRAIN=0 NVSHARE_PREFERRED_LOC=0"
run_arm stock2     "$STOCK" "IGNORE="
run_arm hooked2    "$HOOK"  "IGNORE="
run_arm hk_nocopy  "$HOOK"  "NVSHARE_COPY_PREFETCH=0"
grep -E "rc=|loss" gpurun_out/ab2.log

# rocprofv3 kernel stats: stock vs hooked short runs (cold-safe: db
# already warm).  Counters forbidden with trace flags; --stats only.
cd /tmp && export TMPDIR=/tmp
SHORT='
import sys; sys.path.insert(0, "/root/repo")
from nvshare_amd.workloads.train_resnet import run_training
r = run_training("resnet50", "cuda", batch=32, image=224, steps=20,
                 warmup=5)
print("PROF", round(r["samples_per_s"], 1))
'
timeout 300 env -u LD_PRELOAD rocprofv3 --stats -d /tmp/prof_stock -- \
  python -c "$SHORT" > /root/repo/gpurun_out/prof_stock.log 2>&1
echo "prof_stock rc=$?"
timeout 300 python -m nvshare_amd.run --standalone -- \
  rocprofv3 --stats -d /tmp/prof_hooked -- \
  python -c "$SHORT" > /root/repo/gpurun_out/prof_hooked.log 2>&1
echo "prof_hooked rc=$?"
cp $(find /tmp/prof_stock -name "*kernel_stats*.csv" | head -1) \
  /root/repo/gpurun_out/kstats_stock.csv 2>/dev/null
cp $(find /tmp/prof_hooked -name "*kernel_stats*.csv" | head -1) \
  /root/repo/gpurun_out/kstats_hooked.csv 2>/dev/null
find /tmp/prof_stock /tmp/prof_hooked -name "*.csv" | head
grep PROF /root/repo/gpurun_out/prof_stock.log \
  /root/repo/gpurun_out/prof_hooked.log
