#!/bin/bash
# Round-2 GPU call 2: health-file formats on real sysfs + MFMA PMC + gpu tests
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

# 1. real amdgpu health sysfs formats (validates gpu_health.cc's contract)
{
  echo "== ras dir =="
  ls -l /sys/class/drm/card0/device/ras/ 2>&1
  for f in /sys/class/drm/card0/device/ras/*_err_count; do
    echo "-- $f"; cat "$f" 2>&1
  done
  echo "== pcie_replay_count =="; cat /sys/class/drm/card0/device/pcie_replay_count 2>&1
  echo "== reset_count-like files =="
  ls /sys/class/drm/card0/device/ | grep -iE "reset|recover" 2>&1
  cat /sys/class/drm/card0/device/reset_count 2>&1
  echo "== unique_id =="; cat /sys/class/drm/card0/device/unique_id 2>&1
} > gpurun_out/r2_health_sysfs.txt 2>&1

# 2. mi355x-smi with new ECC column on real sysfs
./native/bin/mi355x-smi > gpurun_out/r2_smi_table.txt 2>&1
./native/bin/mi355x-smi --json > gpurun_out/r2_smi.json 2>&1

# 3. full GPU test tier
timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/r2_gputests2.txt 2>&1
echo "pytest rc=$?" >> gpurun_out/r2_gputests2.txt

# 4. MFMA PMC counters on the MFMA ladder (PMC alone, no trace domains)
cd /tmp && export TMPDIR=/tmp
(timeout 240 rocprofv3 --pmc SQ_INSTS_MFMA SQ_VALU_MFMA_BUSY_CYCLES -d "$GRAFT_REPO_ROOT"/gpurun_out/r2_pmc_mfma -o mfma -- \
   "$GRAFT_REPO_ROOT"/native/bin/mi-stream --mib 256 --iters 5) > "$GRAFT_REPO_ROOT"/gpurun_out/r2_pmc_mfma.log 2>&1
echo "mfma pmc rc=$?" >> "$GRAFT_REPO_ROOT"/gpurun_out/r2_pmc_mfma.log
cd "$GRAFT_REPO_ROOT"

# 5. mi-allreduce with the new max-over-devices timing (N=1 sanity)
timeout 300 ./native/bin/mi-allreduce --ngpus 1 --max-mib 256 > gpurun_out/r2_allreduce1.txt 2>&1
echo "allreduce rc=$?" >> gpurun_out/r2_allreduce1.txt
tail -3 gpurun_out/r2_gputests2.txt
