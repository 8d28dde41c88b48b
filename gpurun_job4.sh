#!/bin/bash
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
# buffer-size sweep: does triad bandwidth hold from 64 MiB to 64 GiB
# buffers (192 GiB resident = 2/3 of HBM)?
: > gpurun_out/r2_buffer_sweep.jsonl
for mib in 64 256 1024 4096 16384 65536; do
  steps=200; [ $mib -ge 4096 ] && steps=50; [ $mib -ge 16384 ] && steps=20
  timeout 420 python bench.py --buffer-mib $mib --steps $steps --warmup 5 \
    >> gpurun_out/r2_buffer_sweep.jsonl 2>> gpurun_out/r2_buffer_sweep.err
  echo "mib=$mib rc=$?" >> gpurun_out/r2_buffer_sweep.err
done
# launch-bound regime: 16 MiB buffers, eager vs hipGraph
timeout 300 python bench.py --buffer-mib 16 --steps 2000 --warmup 100 --graph off \
  > gpurun_out/r2_graph_off16.json 2>&1
timeout 300 python bench.py --buffer-mib 16 --steps 2000 --warmup 100 --graph on \
  > gpurun_out/r2_graph_on16.json 2>&1
timeout 300 python bench.py --buffer-mib 4 --steps 4000 --warmup 100 --graph off \
  > gpurun_out/r2_graph_off4.json 2>&1
timeout 300 python bench.py --buffer-mib 4 --steps 4000 --warmup 100 --graph on \
  > gpurun_out/r2_graph_on4.json 2>&1
./native/bin/mi355x-smi --topo > gpurun_out/r2_topo_real.txt 2>&1
cat gpurun_out/r2_buffer_sweep.jsonl
tail -1 gpurun_out/r2_graph_off16.json; tail -1 gpurun_out/r2_graph_on16.json
