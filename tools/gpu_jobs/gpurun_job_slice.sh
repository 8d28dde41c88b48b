#!/bin/bash
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
# N identical long-running triad processes share one GPU; each reports its
# inner-timed rate over a multi-second region. Regions overlap almost
# fully (equal work, simultaneous start), so the SUM of reported rates is
# the aggregate delivered bandwidth under time-slice sharing.
: > gpurun_out/r2_slice_scaling.txt
for n in 1 2 4 8; do
  pids=""
  for i in $(seq 1 $n); do
    (timeout 280 python bench.py --buffer-mib 256 --steps 20000 --warmup 100 --variant nt \
       > gpurun_out/r2_ts_${n}_${i}.json 2>/dev/null) &
    pids="$pids $!"
  done
  wait $pids
  vals=$(grep -ho '"value": [0-9.]*' gpurun_out/r2_ts_${n}_*.json | awk '{printf "%s%.0f", sep, $2; sep=","}')
  sum=$(grep -ho '"value": [0-9.]*' gpurun_out/r2_ts_${n}_*.json | awk '{s+=$2} END {printf "%.1f", s}')
  echo "n=$n per-proc=[${vals}] aggregate=${sum} GB/s" >> gpurun_out/r2_slice_scaling.txt
  rm -f gpurun_out/r2_ts_${n}_*.json
done
cat gpurun_out/r2_slice_scaling.txt
