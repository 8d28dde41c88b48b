// gpu_health — per-GPU RAS/error-state probing from amdgpu sysfs.
//
// The reference stack's device plugin implicitly owns device health
// (/root/reference/README.md:120-126: a sick plugin pod is the operator's
// only signal). k3samd r01 marked a GPU Unhealthy only when its KFD node
// VANISHED; this module adds the sick-but-present signals so kubelet
// stops scheduling pods onto a failing MI355X:
//
//   * RAS error counts — both amdgpu generations (formats verified on a
//     real MI355X, gpurun_out/r2_aca_probe.txt):
//       - ACA banks (MI300+): <card>/device/ras/aca_<block> files, each
//         "ue: N\nce: N\nde: N" (uncorrectable / correctable / deferred)
//       - legacy:             <card>/device/ras/<block>_err_count files,
//         "ue: N\nce: N"
//   * Fatal RAS events:  ras/event_state ("Fatal Error: count:N, ...") —
//     any fatal event means the GPU took (or needs) a reset.
//   * Retired VRAM pages: ras/gpu_vram_bad_pages (one row per retired
//     page; empty when clean).
//   * PCIe replays:      <card>/device/pcie_replay_count (absent on the
//     MI355X validation box — stays -1 there).
//   * GPU resets:        <card>/device/reset_count (ditto).
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
  long ras_de = -1;        // sum of deferred (poison-pending) errors, ACA
  long fatal_events = -1;  // ras/event_state "Fatal Error" count
  long bad_pages = -1;     // retired VRAM pages (rows in gpu_vram_bad_pages)
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
  long max_deferred = 0;       // deferred = poison pending consumption
  long max_fatal_events = 0;   // any fatal RAS event ⇒ drain
  long max_bad_pages = -1;     // retired pages are handled by the driver;
                               // threshold only if the operator opts in
  long max_pcie_replays = -1;  // disabled by default (noisy on some hosts)
  long max_resets = 0;         // any completed reset ⇒ drain
};

GpuHealthCounters read_gpu_health(const std::string& sysfs_root,
                                  int card_index);

// Empty string = healthy; otherwise a human-readable reason.
std::string health_verdict(const GpuHealthCounters& c,
                           const HealthPolicy& p);

}  // namespace k3samd
