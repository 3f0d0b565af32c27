#!/bin/bash
# Round-2 opening GPU batch (run via gpurun from the repo root).
# Targets the open items in docs/roadmap.md with one ~15-minute call.
set -x
mkdir -p gpurun_out
export HSA_XNACK=1 MIOPEN_FIND_MODE=FAST

# 1. Order-controlled ResNet A/B (roadmap #1): warm MIOpen once, then
#    alternate stock/hooked twice each on the same box.
timeout 120 python -c "
import sys; sys.path.insert(0, '.')
from nvshare_amd.workloads.train_resnet import run_training
run_training('resnet50', 'cuda', batch=32, image=224, steps=5, warmup=5)
print('WARMED')" > gpurun_out/r2_warm.log 2>&1

for arm in stock hooked stock2 hooked2; do
  case $arm in
    stock*) PRE="env -u LD_PRELOAD" ;;
    hooked*) PRE="python -m nvshare_amd.run --standalone --" ;;
  esac
  timeout 150 $PRE python -c "
import sys; sys.path.insert(0, '.')
from nvshare_amd.workloads.train_resnet import run_training
r = run_training('resnet50', 'cuda', batch=32, image=224, steps=60, warmup=10)
print('$arm SPS', round(r['samples_per_s'], 1))" >> gpurun_out/r2_ab.log 2>&1
done
grep SPS gpurun_out/r2_ab.log

# 2. Same A/B with per-hook wall-time attribution on the hooked arm.
timeout 150 env NVSHARE_DEBUG=1 NVSHARE_PROFILE_HOOKS=1 \
  python -m nvshare_amd.run --standalone -- python -c "
import sys; sys.path.insert(0, '.')
from nvshare_amd.workloads.train_resnet import run_training
r = run_training('resnet50', 'cuda', batch=32, image=224, steps=60, warmup=10)
print('prof SPS', round(r['samples_per_s'], 1))" > gpurun_out/r2_prof.log 2>&1
grep -E "SPS|wall time" gpurun_out/r2_prof.log

# 3. hipMemGetInfo-accounting hypothesis: report REAL free (disable the
#    advertised shrinkage) via a fake total equal to the device total.
timeout 150 env NVSHARE_RESERVE_MIB=0 \
  python -m nvshare_amd.run --standalone -- python -c "
import sys; sys.path.insert(0, '.')
from nvshare_amd.workloads.train_resnet import run_training
r = run_training('resnet50', 'cuda', batch=32, image=224, steps=60, warmup=10)
print('reserve0 SPS', round(r['samples_per_s'], 1))" > gpurun_out/r2_rsv0.log 2>&1
grep SPS gpurun_out/r2_rsv0.log

# 4. Coarse-grain / alloc-prefetch ablation.
for knobs in "NVSHARE_COARSE_GRAIN=0" "NVSHARE_ALLOC_PREFETCH=0" \
             "NVSHARE_COARSE_GRAIN=0 NVSHARE_ALLOC_PREFETCH=0"; do
  timeout 150 env $knobs python -m nvshare_amd.run --standalone -- python -c "
import sys; sys.path.insert(0, '.')
from nvshare_amd.workloads.train_resnet import run_training
r = run_training('resnet50', 'cuda', batch=32, image=224, steps=60, warmup=10)
print('$knobs SPS', round(r['samples_per_s'], 1))" >> gpurun_out/r2_ablate.log 2>&1
done
grep SPS gpurun_out/r2_ablate.log
