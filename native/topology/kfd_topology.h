// kfd_topology — MI355X GPU enumeration from the KFD sysfs topology.
//
// The foundation of the stack: discovers AMD GPUs the way the kernel
// exposes them (/sys/class/kfd/kfd/topology/nodes/*), with no dependency on
// ROCm userspace. This is the MI355X-native replacement for the driver/NVML
// enumeration the reference stack gets from the NVIDIA driver + device
// plugin (/root/reference/README.md:47, values.yaml:6-18).
//
// Roots are injectable so CPU-only unit tests can run against fixture trees
// (including the "CPU-only node => 0 GPUs" case, BASELINE.json config #1).

#pragma once

#include <cstdint>
#include <map>
#include <optional>
#include <string>
#include <vector>

namespace k3samd {

// KFD io_link `type` values (linux/drivers/gpu/drm/amd/amdkfd: HSA IOLINK).
constexpr uint32_t kIoLinkPcie = 2;
constexpr uint32_t kIoLinkXgmi = 11;

struct GpuDevice {
  int kfd_node = -1;               // index under topology/nodes/
  std::string name;                // marketing name (topology `name` file)
  uint64_t unique_id = 0;          // KFD unique_id (0 if absent)
  uint32_t vendor_id = 0;          // 0x1002 for AMD
  uint32_t device_id = 0;
  std::string pci_bdf;             // "0000:0c:00.0" from domain+location_id
  int drm_render_minor = -1;       // /dev/dri/renderD<minor>
  int card_index = -1;             // /dev/dri/card<idx> (-1 if unresolved)
  uint64_t vram_bytes = 0;         // sum of FB heaps
  uint32_t simd_count = 0;         // CUs * simd_per_cu
  uint32_t simd_per_cu = 0;
  uint32_t gfx_target_version = 0; // e.g. 90500 => gfx950
  int xgmi_links = 0;              // count of type-11 io_links
  std::vector<int> xgmi_peer_nodes;  // KFD node ids of xGMI peers
  int numa_node = 0;               // peer CPU node via io_links (best effort)

  uint32_t compute_units() const {
    return simd_per_cu ? simd_count / simd_per_cu : 0;
  }
  // "gfx950" from gfx_target_version (major*10000 + minor*100 + step).
  std::string gfx_arch() const;
  // Stable device identifier for the device plugin.
  std::string stable_id() const;
};

struct Topology {
  std::vector<GpuDevice> gpus;
  // driver version string (sysfs module info), may be empty
  std::string driver_version;
};

// Parse one KFD `properties` file (whitespace-separated key/value lines).
std::map<std::string, uint64_t> parse_properties(const std::string& path);

// Enumerate GPUs. `sysfs_root` defaults to "/sys"; fixture trees mirror the
// layout below it:
//   <root>/class/kfd/kfd/topology/nodes/<N>/{properties,name}
//   <root>/class/kfd/kfd/topology/nodes/<N>/mem_banks/<M>/properties
//   <root>/class/kfd/kfd/topology/nodes/<N>/io_links/<L>/properties
//   <root>/module/amdgpu/version
Topology enumerate_topology(const std::string& sysfs_root = "/sys");

// Resolve the sysfs root from the environment (K3SAMD_SYSFS_ROOT) or "/sys".
std::string default_sysfs_root();

// Per-GPU runtime stats from the amdgpu DRM sysfs (all best-effort; -1 /
// 0 when a file is absent). Shared by mi355x-smi and the device plugin's
// metrics endpoint.
struct GpuRuntimeStats {
  uint64_t vram_used = 0;   // bytes
  uint64_t vram_total = 0;  // bytes (sysfs view; may differ from KFD heaps)
  long busy_percent = -1;
  long temp_mc = -1;        // millidegrees C
  long power_uw = -1;       // microwatts
};
GpuRuntimeStats read_runtime_stats(const std::string& sysfs_root,
                                   int card_index);

}  // namespace k3samd
