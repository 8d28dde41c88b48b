#!/bin/bash
# Round-2 GPU call 3: validate ACA health parsing + temp fallback on real
# sysfs, fresh golden smi table, gpu tests, kernel-trace profile of bench
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

./native/bin/mi355x-smi > gpurun_out/r2_smi_table2.txt 2>&1
./native/bin/mi355x-smi --json > gpurun_out/r2_smi2.json 2>&1

timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/r2_gputests3.txt 2>&1
echo "pytest rc=$?" >> gpurun_out/r2_gputests3.txt

# fresh steady-state bench + kernel-trace evidence for r02
cd /tmp && export TMPDIR=/tmp
(timeout 420 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT"/gpurun_out/r2_trace -o r2trace -- \
  python "$GRAFT_REPO_ROOT"/bench.py --steps 200 --warmup 50) > "$GRAFT_REPO_ROOT"/gpurun_out/r2_bench200.log 2>&1
echo "bench-trace rc=$?" >> "$GRAFT_REPO_ROOT"/gpurun_out/r2_bench200.log
cd "$GRAFT_REPO_ROOT"
tail -4 gpurun_out/r2_gputests3.txt; grep -E '^\{' gpurun_out/r2_bench200.log | tail -1
