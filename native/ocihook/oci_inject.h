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
// Device selection is default-deny (unlike NVIDIA's implicit-all, which
// lets any runtimeClassName pod bypass kubelet device accounting):
//   K3SAMD_VISIBLE_DEVICES unset  -> NOTHING injected (no allocation ⇒ no
//                                    GPUs), unless the operator set the
//                                    runtime-level allow_all_default flag
//                                    (K3SAMD_ALLOW_ALL=1, debug only)
//   K3SAMD_VISIBLE_DEVICES=all    -> all GPUs (explicit opt-in sentinel)
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
  bool allow_all_default = false;       // K3SAMD_ALLOW_ALL=1 (runtime env,
                                        // debug): no-allocation ⇒ all GPUs
};

struct InjectReport {
  std::vector<std::string> devices_added;
  std::vector<std::string> mounts_added;
  bool skipped = false;        // nothing injected
  bool no_allocation = false;  // skipped because no env/annotation present
};

// Mutates `config` (an OCI runtime spec DOM) in place.
// Returns false on malformed spec.
bool oci_inject_gpus(const JPtr& config, const Topology& topo,
                     const InjectOptions& opts, InjectReport* report);

}  // namespace k3samd
