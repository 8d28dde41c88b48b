// mi355x-smi — GPU status CLI for the smoke pod.
//
// The MI355X-native analog of the `nvidia-smi` payload the reference runs in
// its validation pod (/root/reference/nvidia-smi.yaml:13) and of the
// golden-output blocks the reference's docs assert on
// (/root/reference/README.md:71-93, :137-156). Reads the KFD sysfs topology
// directly (native/topology), plus per-card DRM runtime stats where present.
//
//   mi355x-smi            human-readable table
//   mi355x-smi --json     machine-readable (used by tests/fixtures)
//
// Honors K3SAMD_SYSFS_ROOT for fixture-tree testing on CPU-only boxes.

#include <cstdio>
#include <cstring>
#include <dirent.h>
#include <fstream>
#include <sstream>
#include <string>

#include "../common/json_writer.h"
#include "../common/version.h"
#include "../topology/gpu_health.h"
#include "../topology/kfd_topology.h"

namespace {

using CardStats = k3samd::GpuRuntimeStats;

CardStats card_stats(const std::string& sysfs_root, int card_index) {
  return k3samd::read_runtime_stats(sysfs_root, card_index);
}

double mib(uint64_t b) { return (double)b / (1024.0 * 1024.0); }

}  // namespace

static int print_report(bool json);

int main(int argc, char** argv) {
  if (k3samd::handle_version_flag(argc, argv, "mi355x-smi")) return 0;
  bool json = false;
  bool topo_matrix = false;
  int watch_s = 0;
  for (int i = 1; i < argc; ++i) {
    if (!std::strcmp(argv[i], "--json")) json = true;
    else if (!std::strcmp(argv[i], "--topo")) topo_matrix = true;
    else if ((!std::strcmp(argv[i], "--watch") || !std::strcmp(argv[i], "-l"))
             && i + 1 < argc)
      watch_s = std::atoi(argv[++i]);
    else if (!std::strcmp(argv[i], "--help") || !std::strcmp(argv[i], "-h")) {
      std::printf(
          "usage: mi355x-smi [--json] [--topo] [--watch SECONDS]\n");
      return 0;
    }
  }
  if (topo_matrix) {
    // the `nvidia-smi topo -m` analog: pairwise connectivity from the
    // KFD io_links. On a full MI355X node every pair is XGMI (7 direct
    // point-to-point links per GPU); PHB = through the host bridge.
    const std::string root = k3samd::default_sysfs_root();
    k3samd::Topology topo = k3samd::enumerate_topology(root);
    size_t n = topo.gpus.size();
    if (n == 0) {
      std::printf("no AMD GPUs found\n");
      return 0;
    }
    std::printf("%8s", "");
    for (size_t j = 0; j < n; ++j) std::printf(" %6s", ("GPU" + std::to_string(j)).c_str());
    std::printf("   NUMA  BDF\n");
    for (size_t i = 0; i < n; ++i) {
      std::printf("%8s", ("GPU" + std::to_string(i)).c_str());
      for (size_t j = 0; j < n; ++j) {
        const char* cell = "PHB";
        if (i == j) {
          cell = "X";
        } else {
          for (int peer : topo.gpus[i].xgmi_peer_nodes)
            if (peer == topo.gpus[j].kfd_node) {
              cell = "XGMI";
              break;
            }
        }
        std::printf(" %6s", cell);
      }
      std::printf("  %4d   %s\n", topo.gpus[i].numa_node,
                  topo.gpus[i].pci_bdf.c_str());
    }
    std::printf("\nLegend: XGMI = direct xGMI link, PHB = host bridge "
                "path, X = self\n");
    return 0;
  }
  if (watch_s > 0) {
    // refresh loop (the nvidia-smi -l analog); re-exec the table printer
    for (;;) {
      std::printf("\033[2J\033[H");
      int rc = print_report(json);
      if (rc != 0) return rc;
      std::fflush(stdout);
      struct timespec ts = {watch_s, 0};
      nanosleep(&ts, nullptr);
    }
  }
  return print_report(json);
}

static int print_report(bool json) {
  const std::string root = k3samd::default_sysfs_root();
  k3samd::Topology topo = k3samd::enumerate_topology(root);

  if (json) {
    k3samd::JsonWriter w;
    w.begin_obj();
    w.key("driver_version").value(topo.driver_version);
    w.key("gpu_count").value((uint64_t)topo.gpus.size());
    w.key("gpus").begin_arr();
    for (const auto& g : topo.gpus) {
      CardStats st = card_stats(root, g.card_index);
      w.begin_obj();
      w.key("kfd_node").value(g.kfd_node);
      w.key("id").value(g.stable_id());
      w.key("name").value(g.name);
      w.key("arch").value(g.gfx_arch());
      w.key("pci_bdf").value(g.pci_bdf);
      w.key("render_minor").value(g.drm_render_minor);
      w.key("card_index").value(g.card_index);
      w.key("vram_bytes").value(g.vram_bytes);
      w.key("vram_used_bytes").value(st.vram_used);
      w.key("compute_units").value(g.compute_units());
      w.key("xgmi_links").value(g.xgmi_links);
      w.key("numa_node").value(g.numa_node);
      w.key("busy_percent").value((int64_t)st.busy_percent);
      w.key("temp_milli_c").value((int64_t)st.temp_mc);
      w.key("power_uw").value((int64_t)st.power_uw);
      // RAS/error state (same probe the device plugin's health model
      // uses, so the operator sees exactly what kubelet is told)
      k3samd::GpuHealthCounters hc =
          k3samd::read_gpu_health(root, g.card_index);
      w.key("ras_supported").value(hc.ras_present ? "true" : "false");
      w.key("ras_uncorrectable").value((int64_t)hc.ras_ue);
      w.key("ras_correctable").value((int64_t)hc.ras_ce);
      w.key("ras_deferred").value((int64_t)hc.ras_de);
      w.key("fatal_ras_events").value((int64_t)hc.fatal_events);
      w.key("retired_vram_pages").value((int64_t)hc.bad_pages);
      w.key("pcie_replay_count").value((int64_t)hc.pcie_replay);
      w.key("reset_count").value((int64_t)hc.reset_count);
      w.end_obj();
    }
    w.end_arr();
    w.end_obj();
    std::printf("%s\n", w.str().c_str());
    return 0;
  }

  std::printf("+-------------------------------------------------------------------------------+\n");
  std::printf("| mi355x-smi            driver: %-25.25s      k3samd stack   |\n",
              topo.driver_version.empty() ? "unknown" : topo.driver_version.c_str());
  std::printf("+----+----------------------+--------+-----+-------------------+------+-----+------+-----------+\n");
  std::printf("| ## | Name                 | Arch   | CUs | VRAM used / total | xGMI | Tmp | Pwr  | ECC ue/ce |\n");
  std::printf("+----+----------------------+--------+-----+-------------------+------+-----+------+-----------+\n");
  int i = 0;
  for (const auto& g : topo.gpus) {
    CardStats st = card_stats(root, g.card_index);
    char vram[40];
    std::snprintf(vram, sizeof(vram), "%6.0f/%6.0f MiB", mib(st.vram_used),
                  mib(g.vram_bytes));
    char temp[24];
    if (st.temp_mc >= 0)
      std::snprintf(temp, sizeof(temp), "%2ldC", (long)(st.temp_mc / 1000));
    else
      std::snprintf(temp, sizeof(temp), " - ");
    char pwr[16];
    if (st.power_uw >= 0)
      std::snprintf(pwr, sizeof(pwr), "%4ldW", st.power_uw / 1000000);
    else
      std::snprintf(pwr, sizeof(pwr), "  - ");
    k3samd::GpuHealthCounters hc =
        k3samd::read_gpu_health(root, g.card_index);
    char ecc[16];
    if (hc.ras_present)
      std::snprintf(ecc, sizeof(ecc), "%ld/%ld", hc.ras_ue, hc.ras_ce);
    else
      std::snprintf(ecc, sizeof(ecc), "n/a");
    std::printf("| %2d | %-20.20s | %-6s | %3u | %-17s | %4d | %s | %s | %-9s |\n",
                i++, g.name.c_str(), g.gfx_arch().c_str(), g.compute_units(),
                vram, g.xgmi_links, temp, pwr, ecc);
  }
  if (topo.gpus.empty()) {
    std::printf("| no AMD GPUs found (no KFD topology under %s)\n",
                root.c_str());
  }
  std::printf("+----+----------------------+--------+-----+-------------------+------+-----+------+-----------+\n");
  return 0;
}
