#!/bin/bash
# Round-2 closing validation: the full GPU surface in one run.
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
{
  echo "== pytest -m gpu =="
  timeout 700 python -m pytest tests -m gpu -q 2>&1 | tail -3
  echo "== smoke() =="
  timeout 300 python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | tail -2
  echo "== bench default =="
  timeout 420 python bench.py 2>/dev/null | tail -1
  echo "== mi-stream (oracle + ladder) =="
  timeout 300 ./native/bin/mi-stream --mib 1024 --iters 10 2>&1 | grep -E "numerics|MFMA|triad_gbps"
  echo "== mi-allreduce N=1 =="
  timeout 300 ./native/bin/mi-allreduce --ngpus 1 --max-mib 64 2>&1 | tail -2
  echo "== mi355x-smi =="
  ./native/bin/mi355x-smi
  ./native/bin/mi355x-smi --topo
  echo "== node-doctor (bare GPU box: k8s FAILs expected) =="
  sh deploy/scripts/node-doctor.sh; echo "doctor rc=$?"
} > gpurun_out/r2_final_validation.txt 2>&1
tail -60 gpurun_out/r2_final_validation.txt
