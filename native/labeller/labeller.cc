// k3samd-node-labeller — publishes per-node GPU labels.
//
// Replaces the roles NFD + GPU Feature Discovery fill in the reference
// stack (/root/reference/README.md:97-103 installs NFD so the device-plugin
// DaemonSet lands on GPU nodes; values.yaml:1-2 enables GFD; the
// nodeSelector example uses `nvidia.com/gpu.present`,
// nvidia-smi.yaml:6-7). Instead of patching the API server directly, the
// labeller writes an NFD *local feature file* — NFD's supported extension
// point for third-party label producers — so the k3samd DaemonSet needs no
// API-server credentials:
//
//   k3samd-node-labeller
//       --features-file=/etc/kubernetes/node-feature-discovery/features.d/k3samd
//
// Labels (amd.com domain, mirrors the nvidia.com/gpu.* surface):
//   amd.com/gpu.present=true        amd.com/gpu.count=8
//   amd.com/gpu.family=CDNA4        amd.com/gpu.arch=gfx950
//   amd.com/gpu.product=AMD-Instinct-MI355X
//   amd.com/gpu.vram=294912Mi       amd.com/gpu.xgmi-links=7
//   amd.com/gpu.cu-count=256        amd.com/gpu.driver-version=<ver>
//   amd.com/gpu.ras=true|false      (ECC/RAS monitoring available)
//
// Modes: --oneshot [--json] (print and exit), daemon (rewrite every
// --interval-s, default 60, picking up hotplug/health changes).

#include <cstdio>
#include <cstring>
#include <fstream>
#include <map>
#include <string>
#include <thread>
#include <chrono>

#include "../common/json_writer.h"
#include "../common/version.h"
#include "../topology/gpu_health.h"
#include "../topology/kfd_topology.h"

namespace {

std::string sanitize(std::string s) {
  for (char& c : s) {
    if (!(std::isalnum((unsigned char)c) || c == '-' || c == '_' || c == '.'))
      c = '-';
  }
  // collapse trailing separators
  while (!s.empty() && (s.back() == '-' || s.back() == '.')) s.pop_back();
  return s;
}

std::string family_for_arch(const std::string& arch) {
  if (arch.rfind("gfx95", 0) == 0) return "CDNA4";
  if (arch.rfind("gfx94", 0) == 0) return "CDNA3";
  if (arch.rfind("gfx90a", 0) == 0) return "CDNA2";
  if (arch.rfind("gfx9", 0) == 0) return "CDNA";
  return "unknown";
}

std::map<std::string, std::string> compute_labels(
    const k3samd::Topology& topo) {
  std::map<std::string, std::string> labels;
  if (topo.gpus.empty()) return labels;
  const auto& g0 = topo.gpus[0];
  labels["amd.com/gpu.present"] = "true";
  labels["amd.com/gpu.count"] = std::to_string(topo.gpus.size());
  labels["amd.com/gpu.arch"] = g0.gfx_arch();
  labels["amd.com/gpu.family"] = family_for_arch(g0.gfx_arch());
  labels["amd.com/gpu.product"] = sanitize(g0.name);
  labels["amd.com/gpu.vram"] =
      std::to_string(g0.vram_bytes / (1024 * 1024)) + "Mi";
  labels["amd.com/gpu.xgmi-links"] = std::to_string(g0.xgmi_links);
  labels["amd.com/gpu.cu-count"] = std::to_string(g0.compute_units());
  if (!topo.driver_version.empty())
    labels["amd.com/gpu.driver-version"] = sanitize(topo.driver_version);
  // RAS capability: ECC/error monitoring available (drives the health
  // model's sick-GPU draining) — lets ECC-sensitive workloads select
  // monitored nodes
  {
    k3samd::GpuHealthCounters hc = k3samd::read_gpu_health(
        k3samd::default_sysfs_root(), g0.card_index);
    labels["amd.com/gpu.ras"] = hc.ras_present ? "true" : "false";
  }
  return labels;
}

bool write_features_file(const std::string& path,
                         const std::map<std::string, std::string>& labels) {
  // atomic replace so NFD never reads a torn file
  std::string tmp = path + ".tmp";
  {
    std::ofstream f(tmp, std::ios::trunc);
    if (!f) return false;
    for (const auto& [k, v] : labels) f << k << "=" << v << "\n";
  }
  return std::rename(tmp.c_str(), path.c_str()) == 0;
}

}  // namespace

int main(int argc, char** argv) {
  if (k3samd::handle_version_flag(argc, argv, "k3samd-node-labeller")) return 0;
  std::string features_file;
  bool oneshot = false, json = false;
  int interval_s = 60;
  for (int i = 1; i < argc; ++i) {
    if (!std::strcmp(argv[i], "--features-file") && i + 1 < argc)
      features_file = argv[++i];
    else if (!std::strcmp(argv[i], "--interval-s") && i + 1 < argc)
      interval_s = std::atoi(argv[++i]);
    else if (!std::strcmp(argv[i], "--oneshot")) oneshot = true;
    else if (!std::strcmp(argv[i], "--json")) { json = true; oneshot = true; }
    else {
      std::printf("k3samd-node-labeller [--features-file F] [--interval-s N]"
                  " [--oneshot] [--json]\n");
      return !std::strcmp(argv[i], "--help") ? 0 : 2;
    }
  }

  do {
    auto topo = k3samd::enumerate_topology(k3samd::default_sysfs_root());
    auto labels = compute_labels(topo);
    if (!features_file.empty()) {
      if (!write_features_file(features_file, labels)) {
        std::fprintf(stderr, "labeller: cannot write %s\n",
                     features_file.c_str());
        return 1;
      }
    }
    if (json) {
      k3samd::JsonWriter w;
      w.begin_obj();
      for (const auto& [k, v] : labels) w.key(k).value(v);
      w.end_obj();
      std::printf("%s\n", w.str().c_str());
    } else if (oneshot && features_file.empty()) {
      for (const auto& [k, v] : labels) std::printf("%s=%s\n", k.c_str(),
                                                    v.c_str());
    }
    if (!oneshot)
      std::this_thread::sleep_for(std::chrono::seconds(interval_s));
  } while (!oneshot);
  return 0;
}
