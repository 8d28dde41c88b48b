// gpu_health — per-GPU RAS/error-state probing from amdgpu sysfs.
//
// The reference stack's device plugin implicitly owns device health
// (/root/reference/README.md:120-126: a sick plugin pod is the operator's
// only signal). k3samd r01 marked a GPU Unhealthy only when its KFD node
// VANISHED; this module adds the sick-but-present signals so kubelet
// stops scheduling pods onto a failing MI355X:
//
//   * RAS error counts:  <card>/device/ras/<block>_err_count files, each
//     "ue: N\nce: N" (uncorrectable / correctable), summed over blocks
//     (umc = HBM ECC, gfx, sdma, mmhub, ...).
//   * PCIe replays:      <card>/device/pcie_replay_count (link health).
//   * GPU resets:        <card>/device/reset_count or amdgpu reset_count
//     (a recovered-from-hang GPU should drain before new pods land).
//
// All reads are best-effort: a missing file yields -1 ("not exposed"),
// which never trips a threshold — CPU fixtures and driver builds without
// RAS stay healthy.

#pragma once

#include <string>

namespace k3samd {

struct GpuHealthCounters {
  long ras_ue = -1;        // sum of uncorrectable errors over RAS blocks
  long ras_ce = -1;        // sum of correctable errors
  long pcie_replay = -1;   // link-level replay count
  long reset_count = -1;   // completed GPU resets
  bool ras_present = false;
};

// Thresholds; -1 disables a check. Defaults follow the NVIDIA device
// plugin's stance (any uncorrectable/double-bit error ⇒ unhealthy) plus
// a generous correctable-error budget (ECC CEs are self-healing; a storm
// of them predicts failure).
struct HealthPolicy {
  long max_uncorrectable = 0;
  long max_correctable = 10000;
  long max_pcie_replays = -1;  // disabled by default (noisy on some hosts)
  long max_resets = 0;         // any completed reset ⇒ drain
};

GpuHealthCounters read_gpu_health(const std::string& sysfs_root,
                                  int card_index);

// Empty string = healthy; otherwise a human-readable reason.
std::string health_verdict(const GpuHealthCounters& c,
                           const HealthPolicy& p);

}  // namespace k3samd
