#!/bin/sh
# k3samd node doctor — one-shot diagnosis of everything node-side that the
# stack needs, in dependency order. The scripted version of the
# reference's "is the node ready" manual checks
# (/root/reference/README.md:21-95), extended with the k3samd pieces.
#
#   node-doctor.sh [--containerd-config /var/lib/rancher/k3s/agent/etc/containerd/config.toml]
#
# Exit 0 = node ready; non-zero = first failing layer printed.
set -u

containerd_cfg="${K3SAMD_CONTAINERD_CONFIG:-/var/lib/rancher/k3s/agent/etc/containerd/config.toml}"
kubelet_dir="${K3SAMD_KUBELET_DIR:-/var/lib/kubelet/device-plugins}"
dev_root="${K3SAMD_DEV_ROOT:-/dev}"
while [ $# -gt 0 ]; do
  case "$1" in
    --containerd-config) containerd_cfg="$2"; shift 2 ;;
    *) echo "usage: node-doctor.sh [--containerd-config F]"; exit 2 ;;
  esac
done

fail=0
pass() { printf 'ok    %s\n' "$*"; }
warn() { printf 'warn  %s\n' "$*"; }
bad()  { printf 'FAIL  %s\n' "$*"; fail=1; }

echo "== k3samd node doctor =="

# L0: kernel driver interface
if [ -e "$dev_root/kfd" ]; then
  pass "/dev/kfd present (amdgpu KFD loaded)"
else
  bad "/dev/kfd missing - amdgpu/KFD not loaded (see README 'Node OS provisioning')"
fi
renders=$(ls "$dev_root"/dri/renderD* 2>/dev/null | wc -l)
if [ "$renders" -ge 1 ]; then
  pass "$renders render node(s) under $dev_root/dri"
else
  bad "no $dev_root/dri/renderD* nodes"
fi

# GPU enumeration (our topology lib view)
smi=$(command -v mi355x-smi || echo ./native/bin/mi355x-smi)
if [ -x "$smi" ]; then
  gpus=$("$smi" --json 2>/dev/null | sed -n 's/.*"gpu_count": *\([0-9]*\).*/\1/p')
  if [ "${gpus:-0}" -ge 1 ]; then
    pass "mi355x-smi enumerates $gpus GPU(s)"
  else
    bad "mi355x-smi sees 0 GPUs (KFD topology empty?)"
  fi
else
  warn "mi355x-smi not installed; skipping enumeration check"
fi

# L1: OCI runtime binary + containerd registration
if command -v k3samd-oci-runtime >/dev/null 2>&1 ||
   [ -x /usr/local/bin/k3samd-oci-runtime ]; then
  pass "k3samd-oci-runtime installed"
else
  bad "k3samd-oci-runtime not on PATH (/usr/local/bin)"
fi
if [ -r "$containerd_cfg" ]; then
  if grep -q 'runtimes.amd' "$containerd_cfg" 2>/dev/null &&
     grep -q 'k3samd-oci-runtime' "$containerd_cfg" 2>/dev/null; then
    pass "containerd RuntimeClass 'amd' registered in $containerd_cfg"
  else
    bad "containerd config $containerd_cfg lacks the 'amd' runtime entry (deploy/containerd-runtime.md)"
  fi
else
  warn "containerd config $containerd_cfg unreadable; cannot verify RuntimeClass"
fi

# L3: kubelet device-plugin registry + our socket
if [ -d "$kubelet_dir" ]; then
  pass "kubelet device-plugin dir $kubelet_dir exists"
  if [ -S "$kubelet_dir/kubelet.sock" ]; then
    pass "kubelet registration socket present"
  else
    warn "kubelet.sock not present (kubelet down, or non-standard dir)"
  fi
  if [ -S "$kubelet_dir/amd-gpu.sock" ]; then
    pass "k3samd plugin socket present (DaemonSet running)"
  else
    warn "amd-gpu.sock not present (device-plugin DaemonSet not running yet)"
  fi
else
  bad "$kubelet_dir missing - kubelet not installed/running?"
fi

# Health: RAS surface
card0=/sys/class/drm/card0/device
if [ -d "$card0/ras" ]; then
  pass "amdgpu RAS sysfs present (ECC health monitoring active)"
else
  warn "no RAS sysfs on card0 (driver build without RAS; health model degrades to existence checks)"
fi

if [ "$fail" -eq 0 ]; then
  echo "node ready."
else
  echo "node NOT ready - fix the FAIL lines above."
fi
exit "$fail"
