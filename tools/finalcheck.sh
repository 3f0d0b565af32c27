#!/bin/bash
# Final round-2 GPU check: proper XNACK=0 hooked arm (client_env now
# respects the override) + HEAD validation (smoke + key GPU tests).
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
: > gpurun_out/final.log
timeout 200 env -u LD_PRELOAD python -c "$TRAIN" 2>&1 \
  | sed "s/^ARM/stock/" >> gpurun_out/final.log
timeout 200 env HSA_XNACK=0 python -m nvshare_amd.run --standalone -- \
  python -c "$TRAIN" 2>&1 | sed "s/^ARM/hooked_xnack0/" >> gpurun_out/final.log
timeout 200 env HSA_XNACK=0 python -m nvshare_amd.run --standalone -- \
  python -c "$TRAIN" 2>&1 | sed "s/^ARM/hooked_xnack0b/" >> gpurun_out/final.log
timeout 200 python -m nvshare_amd.run --standalone -- \
  python -c "$TRAIN" 2>&1 | sed "s/^ARM/hooked_xnack1/" >> gpurun_out/final.log
grep -E "^stock|^hooked" gpurun_out/final.log

timeout 300 python -c "import __graft_entry__ as g; g.smoke()" \
  > gpurun_out/final_smoke.log 2>&1
echo "smoke rc=$?"
tail -2 gpurun_out/final_smoke.log

timeout 600 python -m pytest tests/test_gpu.py -q -m gpu \
  -k "colocated or numerics or train or oversub" \
  > gpurun_out/final_pytest.log 2>&1
echo "pytest rc=$?"
tail -4 gpurun_out/final_pytest.log
