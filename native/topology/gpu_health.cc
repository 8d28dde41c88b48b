#include "gpu_health.h"

#include <dirent.h>

#include <cstdio>
#include <cstring>
#include <fstream>
#include <sstream>

namespace k3samd {

namespace {

long read_long_file(const std::string& path) {
  std::ifstream f(path);
  long v = -1;
  if (f >> v) return v;
  return -1;
}

// Parse one RAS err_count file: lines "ue: N" / "ce: N".
bool parse_err_count(const std::string& path, long& ue, long& ce) {
  std::ifstream f(path);
  if (!f) return false;
  std::string key;
  long val;
  bool any = false;
  while (f >> key >> val) {
    if (key == "ue:") {
      ue += val;
      any = true;
    } else if (key == "ce:") {
      ce += val;
      any = true;
    }
  }
  return any;
}

}  // namespace

GpuHealthCounters read_gpu_health(const std::string& sysfs_root,
                                  int card_index) {
  GpuHealthCounters out;
  if (card_index < 0) return out;
  std::string dev =
      sysfs_root + "/class/drm/card" + std::to_string(card_index) + "/device";

  std::string ras_dir = dev + "/ras";
  if (DIR* d = ::opendir(ras_dir.c_str())) {
    long ue = 0, ce = 0;
    bool any = false;
    while (dirent* e = ::readdir(d)) {
      const char* n = e->d_name;
      size_t len = std::strlen(n);
      if (len > 10 && !std::strcmp(n + len - 10, "_err_count"))
        any |= parse_err_count(ras_dir + "/" + n, ue, ce);
    }
    ::closedir(d);
    if (any) {
      out.ras_present = true;
      out.ras_ue = ue;
      out.ras_ce = ce;
    }
  }

  out.pcie_replay = read_long_file(dev + "/pcie_replay_count");
  out.reset_count = read_long_file(dev + "/reset_count");
  return out;
}

std::string health_verdict(const GpuHealthCounters& c,
                           const HealthPolicy& p) {
  std::ostringstream r;
  if (p.max_uncorrectable >= 0 && c.ras_ue > p.max_uncorrectable)
    r << "RAS uncorrectable errors " << c.ras_ue << " > "
      << p.max_uncorrectable << "; ";
  if (p.max_correctable >= 0 && c.ras_ce > p.max_correctable)
    r << "RAS correctable errors " << c.ras_ce << " > " << p.max_correctable
      << "; ";
  if (p.max_pcie_replays >= 0 && c.pcie_replay > p.max_pcie_replays)
    r << "PCIe replays " << c.pcie_replay << " > " << p.max_pcie_replays
      << "; ";
  if (p.max_resets >= 0 && c.reset_count > p.max_resets)
    r << "GPU resets " << c.reset_count << " > " << p.max_resets << "; ";
  std::string s = r.str();
  if (!s.empty()) s.resize(s.size() - 2);  // trim trailing "; "
  return s;
}

}  // namespace k3samd
