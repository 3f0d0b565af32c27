#!/bin/bash
# Attribution round 9: (a) long-run amortization (steps=300) — do the
# episodic stalls persist at steady state or are they a warmup tail?
# (b) kernel-trace timeline of a hooked run for local analysis.
set -x
mkdir -p gpurun_out
export HSA_XNACK=1 MIOPEN_FIND_MODE=FAST

LONG='
import sys; sys.path.insert(0, ".")
from nvshare_amd.workloads.train_resnet import run_training
import math
r = run_training("resnet50", "cuda", batch=32, image=224, steps=300,
                 warmup=20)
assert math.isfinite(r["loss"]), r
print("ARM", round(r["samples_per_s"], 1))
'
timeout 300 env -u LD_PRELOAD python -c "
import sys; sys.path.insert(0, '.')
from nvshare_amd.workloads.train_resnet import run_training
run_training('resnet50', 'cuda', batch=32, image=224, steps=3, warmup=3)
print('WARMED')" > gpurun_out/ab9_warm.log 2>&1

: > gpurun_out/ab9.log
run_arm () {
  local label="$1" pre="$2"
  timeout 300 env $pre python -c "$LONG" 2>&1 \
    | sed "s/^ARM/$label/" >> gpurun_out/ab9.log
  echo "$label rc=${PIPESTATUS[0]}" >> gpurun_out/ab9.log
}
STOCK="env -u LD_PRELOAD"
HOOK="python -m nvshare_amd.run --standalone --"

run_arm stock_long  "$STOCK"
run_arm hooked_long "$HOOK"
run_arm hooked_long2 "$HOOK"
run_arm stock_long2 "$STOCK"
grep -E "rc=|ARM|long" gpurun_out/ab9.log

# Kernel-trace timeline (hooked, 40 steps) for local analysis.
cd /tmp && export TMPDIR=/tmp PYTHONPATH=/root/repo
SHORT='
from nvshare_amd.workloads.train_resnet import run_training
r = run_training("resnet50", "cuda", batch=32, image=224, steps=40,
                 warmup=10)
print("PROF", round(r["samples_per_s"], 1))
'
timeout 300 python -m nvshare_amd.run --standalone -- \
  rocprofv3 --kernel-trace --stats --output-format csv \
  -d /tmp/prof_hk -o hk -- python -c "$SHORT" \
  > /root/repo/gpurun_out/prof_hk9.log 2>&1
echo "prof rc=$?"
KT=$(find /tmp/prof_hk -name "*kernel_trace*.csv" | head -1)
# keep it small: only rows > 1 ms plus a 1-in-50 sample of the rest
python3 - "$KT" <<'EOF'
import csv, sys
src = sys.argv[1]
with open(src) as f, open("/root/repo/gpurun_out/ktrace_hooked.csv", "w") as o:
    r = csv.reader(f); w = csv.writer(o)
    hdr = next(r); w.writerow(hdr)
    i = {c: n for n, c in enumerate(hdr)}
    k = 0
    for row in r:
        try:
            dur = int(row[i.get("End_Timestamp", i.get("END_TS", 0))]) - \
                  int(row[i.get("Start_Timestamp", i.get("START_TS", 0))])
        except Exception:
            dur = 0
        k += 1
        if dur > 1_000_000 or k % 50 == 0:
            w.writerow(row)
EOF
echo "trace rows kept: $(wc -l < /root/repo/gpurun_out/ktrace_hooked.csv)"
head -2 "$KT"
