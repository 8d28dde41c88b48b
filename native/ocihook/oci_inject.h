// OCI config.json GPU injection — the MI355X-native analog of
// nvidia-container-runtime.
//
// The reference registers an OCI runtime with containerd that
// "automatically cop[ies] everything needed for your pod to use the GPU"
// when a pod sets `runtimeClassName: nvidia`
// (/root/reference/README.md:57-69, :164). This module implements that
// behavior for AMD: given a container's OCI spec, inject /dev/kfd, the
// selected /dev/dri render+card nodes, matching device-cgroup allow rules,
// and (optionally) the host ROCm userspace as read-only bind mounts.
//
// Device selection mirrors the NVIDIA env-var contract:
//   K3SAMD_VISIBLE_DEVICES unset  -> all GPUs (RuntimeClass alone suffices)
//   K3SAMD_VISIBLE_DEVICES=none   -> nothing injected
//   K3SAMD_VISIBLE_DEVICES=<ids>  -> the listed stable ids (set by the
//                                    device plugin's Allocate response)

#pragma once

#include <string>
#include <vector>

#include "../common/minijson.h"
#include "../topology/kfd_topology.h"

namespace k3samd {

struct InjectOptions {
  std::string sysfs_root = "/sys";
  std::string dev_root = "/dev";
  std::string rocm_root = "/opt/rocm";  // host path bind-mounted when asked
  bool inject_rocm_default = false;     // K3SAMD_INJECT_ROCM=1 overrides
};

struct InjectReport {
  std::vector<std::string> devices_added;
  std::vector<std::string> mounts_added;
  bool skipped = false;  // K3SAMD_VISIBLE_DEVICES=none/void
};

// Mutates `config` (an OCI runtime spec DOM) in place.
// Returns false on malformed spec.
bool oci_inject_gpus(const JPtr& config, const Topology& topo,
                     const InjectOptions& opts, InjectReport* report);

}  // namespace k3samd
