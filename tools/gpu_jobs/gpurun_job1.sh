#!/bin/bash
# Round-2 GPU call 1: validate rebuilt natives + PMC retry + media probe
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

# 1. GPU test suite (includes new default-deny ocihook test on real sysfs)
timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/r2_gputests.txt 2>&1
echo "pytest rc=$?" >> gpurun_out/r2_gputests.txt

# 2. Media / VCN capability probe
{
  echo "== ip_discovery blocks =="
  ls /sys/class/drm/card0/device/ip_discovery/die/0/ 2>&1
  for b in VCN UVD VCE JPEG; do
    d=/sys/class/drm/card0/device/ip_discovery/die/0/$b
    if [ -d "$d" ]; then
      echo "-- $b present:"; ls "$d"; cat "$d"/*/major "$d"/*/minor "$d"/*/revision 2>/dev/null
    fi
  done
  echo "== vcn sysfs =="
  ls /sys/class/drm/card0/device/ | grep -i -E 'vcn|video' 
  cat /sys/class/drm/card0/device/vcn_busy_percent 2>&1
  echo "== userspace media stack present? =="
  ls /usr/lib/x86_64-linux-gnu/libva* /usr/lib/x86_64-linux-gnu/dri 2>&1
  which ffmpeg vainfo mpv 2>&1
  echo "== amdgpu_firmware_info (VCN fw?) =="
  grep -i -E 'vcn|uvd|vce' /sys/kernel/debug/dri/0/amdgpu_firmware_info 2>&1 | head
  echo "== rocm-smi fw version table =="
  rocm-smi --showfwinfo 2>&1 | grep -i -E 'vcn|uvd|vce|VCLK' | head
  echo "== amd-smi static asic =="
  amd-smi static -a 2>&1 | head -30
} > gpurun_out/r2_media_probe.txt 2>&1

# 3. PMC retry: counters WITHOUT any trace domain
cd /tmp && export TMPDIR=/tmp
timeout 120 rocprofv3 --list-avail > "$GRAFT_REPO_ROOT"/gpurun_out/r2_counters_avail.txt 2>&1
cd "$GRAFT_REPO_ROOT"
for cset in "FETCH_SIZE WRITE_SIZE" "TCC_EA0_RDREQ_sum TCC_EA0_WRREQ_sum" "SQ_INSTS_VALU"; do
  name=$(echo $cset | tr ' ' '_')
  (cd /tmp && timeout 240 rocprofv3 --pmc $cset -d "$GRAFT_REPO_ROOT"/gpurun_out/r2_pmc_$name -o pmc_$name -- \
     "$GRAFT_REPO_ROOT"/native/bin/mi-stream --mib 1024 --iters 20 --no-mfma) \
     > gpurun_out/r2_pmc_$name.log 2>&1
  echo "pmc[$cset] rc=$?" >> gpurun_out/r2_pmc_summary.txt
done

# 4. quick bench sanity (native path unchanged, but confirm)
timeout 300 python bench.py --steps 50 --warmup 10 > gpurun_out/r2_bench_check.json 2> gpurun_out/r2_bench_check.err
echo "bench rc=$?" >> gpurun_out/r2_bench_check.err
tail -2 gpurun_out/r2_gputests.txt
