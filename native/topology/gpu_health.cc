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

// Parse one RAS count file — legacy "<block>_err_count" ("ue: N\nce: N")
// or ACA "aca_<block>" ("ue: N\nce: N\nde: N") — same line grammar.
bool parse_err_count(const std::string& path, long& ue, long& ce, long& de) {
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
    } else if (key == "de:") {
      de += val;
      any = true;
    }
  }
  return any;
}

// ras/event_state: "Fatal Error: count:N, last_seqno:M" (plus Poison
// Creation/Consumption rows). Returns the fatal count, -1 if unreadable.
long parse_fatal_events(const std::string& path) {
  std::ifstream f(path);
  if (!f) return -1;
  std::string line;
  while (std::getline(f, line)) {
    auto pos = line.find("Fatal Error:");
    if (pos == std::string::npos) continue;
    auto c = line.find("count:", pos);
    if (c == std::string::npos) continue;
    return std::atol(line.c_str() + c + 6);
  }
  return -1;
}

// ras/gpu_vram_bad_pages: one row per retired page; empty file = clean.
long count_bad_pages(const std::string& path) {
  std::ifstream f(path);
  if (!f) return -1;
  long n = 0;
  std::string line;
  while (std::getline(f, line))
    if (!line.empty() && line[0] == '0') ++n;  // rows start with 0x<addr>
  return n;
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
    long ue = 0, ce = 0, de = 0;
    bool any = false, any_de = false;
    while (dirent* e = ::readdir(d)) {
      const char* n = e->d_name;
      size_t len = std::strlen(n);
      bool legacy = len > 10 && !std::strcmp(n + len - 10, "_err_count");
      bool aca = !std::strncmp(n, "aca_", 4);
      if (legacy || aca) {
        if (parse_err_count(ras_dir + "/" + n, ue, ce, de)) {
          any = true;
          if (aca) any_de = true;  // ACA banks report de:; legacy doesn't
        }
      }
    }
    ::closedir(d);
    if (any) {
      out.ras_present = true;
      out.ras_ue = ue;
      out.ras_ce = ce;
      if (any_de) out.ras_de = de;
    }
    out.fatal_events = parse_fatal_events(ras_dir + "/event_state");
    out.bad_pages = count_bad_pages(ras_dir + "/gpu_vram_bad_pages");
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
  if (p.max_deferred >= 0 && c.ras_de > p.max_deferred)
    r << "RAS deferred (poison) errors " << c.ras_de << " > "
      << p.max_deferred << "; ";
  if (p.max_fatal_events >= 0 && c.fatal_events > p.max_fatal_events)
    r << "fatal RAS events " << c.fatal_events << " > " << p.max_fatal_events
      << "; ";
  if (p.max_bad_pages >= 0 && c.bad_pages > p.max_bad_pages)
    r << "retired VRAM pages " << c.bad_pages << " > " << p.max_bad_pages
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
