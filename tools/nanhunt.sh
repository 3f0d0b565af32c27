#!/bin/bash
# NaN attribution matrix for the flagship training loop (run via gpurun
# from the repo root).  Every arm gets a COLD per-arm MIOpen user db
# (the fresh-box condition that failed in round 1), per-step loss
# checks, and a bad-tensor dump on the first non-finite value.
set -x
mkdir -p gpurun_out
export HSA_XNACK=1

STOCK="env -u LD_PRELOAD"
HOOK="python -m nvshare_amd.run --standalone --"

run_arm () {
  local label="$1" pre="$2" envs="$3"; shift 3
  export MIOPEN_USER_DB_PATH=/tmp/miopen_${label}
  mkdir -p "$MIOPEN_USER_DB_PATH"
  timeout 200 env $envs $pre python tools/nanhunt.py --label "$label" \
    --steps 40 "$@" >> gpurun_out/nanhunt.log 2>&1
  echo "$label rc=$?" >> gpurun_out/nanhunt.log
}

: > gpurun_out/nanhunt.log
run_arm stock         "$STOCK" "IGNORE="
run_arm stock_fp32    "$STOCK" "IGNORE=" --dtype float32
run_arm hooked        "$HOOK"  "IGNORE="
run_arm hk_nocoarse   "$HOOK"  "NVSHARE_COARSE_GRAIN=0"
run_arm hk_noprefetch "$HOOK"  "NVSHARE_ALLOC_PREFETCH=0"
run_arm hk_nocp       "$HOOK"  "NVSHARE_COARSE_GRAIN=0 NVSHARE_ALLOC_PREFETCH=0"
run_arm hk_nopassth   "$HOOK"  "NVSHARE_PASSTHROUGH_MIB=0"
run_arm hk_noum       "$HOOK"  "NVSHARE_DISABLE_UM=1"

grep -E "FINITE|NONFINITE|rc=" gpurun_out/nanhunt.log
