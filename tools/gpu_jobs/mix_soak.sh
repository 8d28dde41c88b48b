#!/bin/bash
# Production-mix soak: alternate the stack's real payloads on one GPU for
# ~10 minutes (bench steps, allreduce sweeps, smi polls, oracle runs) —
# the pod-churn pattern a time-sliced node sees, not a single kernel.
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
end=$((SECONDS + 570))
i=0
: > gpurun_out/r2_mix_soak.txt
while [ $SECONDS -lt $end ]; do
  i=$((i+1))
  timeout 120 python bench.py --steps 100 --warmup 10 2>/dev/null | \
    grep -o '"value": [0-9.]*' >> gpurun_out/r2_mix_soak.txt
  timeout 120 ./native/bin/mi-allreduce --ngpus 1 --max-mib 64 2>/dev/null | \
    grep -o '"verify_ok": [a-z]*' >> gpurun_out/r2_mix_soak.txt
  timeout 60 ./native/bin/mi-stream --mib 256 --iters 3 2>/dev/null | \
    grep -oE "numerics: .*" >> gpurun_out/r2_mix_soak.txt
  ./native/bin/mi355x-smi --json 2>/dev/null | \
    grep -o '"ras_uncorrectable":[ 0-9-]*' | head -1 >> gpurun_out/r2_mix_soak.txt
done
echo "cycles=$i" >> gpurun_out/r2_mix_soak.txt
sort gpurun_out/r2_mix_soak.txt | uniq -c | sort -rn | head -12
