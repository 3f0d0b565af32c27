#!/bin/bash
# Attribution round 4: where does the ~1.4x managed-memory step cost
# live?  Kernel-stats CSV diff (stock vs hooked), plus THP and
# disable-UM controls.  Run via gpurun.
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
print('WARMED')" > gpurun_out/ab4_warm.log 2>&1

: > gpurun_out/ab4.log
run_arm () {
  local label="$1" pre="$2" envs="$3"
  timeout 200 env $envs $pre python -c "$TRAIN" 2>&1 \
    | sed "s/^ARM/$label/" >> gpurun_out/ab4.log
  echo "$label rc=${PIPESTATUS[0]}" >> gpurun_out/ab4.log
}
STOCK="env -u LD_PRELOAD"
HOOK="python -m nvshare_amd.run --standalone --"

run_arm stock1   "$STOCK" "IGNORE="
run_arm hooked1  "$HOOK"  "IGNORE="
run_arm hk_thp   "$HOOK"  "NVSHARE_THP=1"
run_arm hk_noum  "$HOOK"  "NVSHARE_DISABLE_UM=1"
run_arm stock2   "$STOCK" "IGNORE="
run_arm hooked2  "$HOOK"  "IGNORE="
grep -E "rc=|loss" gpurun_out/ab4.log

# Kernel-stats CSVs (csv format explicitly: ROCm 7.2 defaults to rocpd).
cd /tmp && export TMPDIR=/tmp PYTHONPATH=/root/repo
SHORT='
from nvshare_amd.workloads.train_resnet import run_training
r = run_training("resnet50", "cuda", batch=32, image=224, steps=20,
                 warmup=5)
print("PROF", round(r["samples_per_s"], 1))
'
timeout 300 env -u LD_PRELOAD rocprofv3 --kernel-trace --stats \
  --output-format csv -d /tmp/prof_stock -o st -- python -c "$SHORT" \
  > /root/repo/gpurun_out/prof_stock.log 2>&1
echo "prof_stock rc=$?"
timeout 300 python -m nvshare_amd.run --standalone -- \
  rocprofv3 --kernel-trace --stats --output-format csv \
  -d /tmp/prof_hooked -o hk -- python -c "$SHORT" \
  > /root/repo/gpurun_out/prof_hooked.log 2>&1
echo "prof_hooked rc=$?"
find /tmp/prof_stock /tmp/prof_hooked -type f | tee /root/repo/gpurun_out/prof_files.txt
for f in $(grep -i "kernel_stats.*csv" /root/repo/gpurun_out/prof_files.txt); do
  case "$f" in
    *stock*) cp "$f" /root/repo/gpurun_out/kstats_stock.csv ;;
    *hooked*) cp "$f" /root/repo/gpurun_out/kstats_hooked.csv ;;
  esac
done
grep PROF /root/repo/gpurun_out/prof_stock.log /root/repo/gpurun_out/prof_hooked.log
