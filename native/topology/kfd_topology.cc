#include "kfd_topology.h"

#include <algorithm>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <dirent.h>
#include <fstream>
#include <sstream>
#include <sys/stat.h>

namespace k3samd {

namespace {

bool is_dir(const std::string& p) {
  struct stat st;
  return ::stat(p.c_str(), &st) == 0 && S_ISDIR(st.st_mode);
}

std::string read_file(const std::string& path) {
  std::ifstream f(path);
  if (!f) return {};
  std::stringstream ss;
  ss << f.rdbuf();
  return ss.str();
}

std::string trim(const std::string& s) {
  size_t b = s.find_first_not_of(" \t\r\n");
  if (b == std::string::npos) return {};
  size_t e = s.find_last_not_of(" \t\r\n");
  return s.substr(b, e - b + 1);
}

// numeric subdirectories of `dir`, sorted ascending
std::vector<int> numeric_subdirs(const std::string& dir) {
  std::vector<int> out;
  DIR* d = ::opendir(dir.c_str());
  if (!d) return out;
  while (struct dirent* e = ::readdir(d)) {
    const char* n = e->d_name;
    if (n[0] == '\0' || n[0] == '.') continue;
    char* end = nullptr;
    long v = std::strtol(n, &end, 10);
    if (end && *end == '\0') out.push_back((int)v);
  }
  ::closedir(d);
  std::sort(out.begin(), out.end());
  return out;
}

std::string bdf_string(uint64_t domain, uint64_t location_id) {
  // location_id encodes bus/device/function: bus[15:8], dev[7:3], fn[2:0]
  unsigned bus = (unsigned)((location_id >> 8) & 0xff);
  unsigned dev = (unsigned)((location_id >> 3) & 0x1f);
  unsigned fn = (unsigned)(location_id & 0x7);
  char buf[32];
  std::snprintf(buf, sizeof(buf), "%04llx:%02x:%02x.%x",
                (unsigned long long)domain, bus, dev, fn);
  return buf;
}

}  // namespace

std::string GpuDevice::gfx_arch() const {
  if (!gfx_target_version) return {};
  unsigned major = gfx_target_version / 10000;
  unsigned minor = (gfx_target_version / 100) % 100;
  unsigned step = gfx_target_version % 100;
  char buf[32];
  // gfx{major}{minor:x}{step:x} — e.g. 90500 -> gfx950, 90402 -> gfx942
  std::snprintf(buf, sizeof(buf), "gfx%u%x%x", major, minor, step);
  return buf;
}

std::string GpuDevice::stable_id() const {
  char buf[64];
  if (unique_id) {
    std::snprintf(buf, sizeof(buf), "amdgpu-%016llx",
                  (unsigned long long)unique_id);
    return buf;
  }
  // fall back to the PCI address, which is stable across reboots
  return "amdgpu-" + pci_bdf;
}

std::map<std::string, uint64_t> parse_properties(const std::string& path) {
  std::map<std::string, uint64_t> kv;
  std::ifstream f(path);
  std::string key;
  uint64_t val;
  while (f >> key >> val) kv[key] = val;
  return kv;
}

GpuRuntimeStats read_runtime_stats(const std::string& sysfs_root,
                                   int card_index) {
  GpuRuntimeStats st;
  if (card_index < 0) return st;
  const std::string dev =
      sysfs_root + "/class/drm/card" + std::to_string(card_index) + "/device";
  auto read_long = [](const std::string& p, long dflt) {
    std::string v = trim(read_file(p));
    return v.empty() ? dflt : std::strtol(v.c_str(), nullptr, 10);
  };
  st.vram_used = (uint64_t)read_long(dev + "/mem_info_vram_used", 0);
  st.vram_total = (uint64_t)read_long(dev + "/mem_info_vram_total", 0);
  st.busy_percent = read_long(dev + "/gpu_busy_percent", -1);
  // hwmon index is global; scan the card's hwmon dir
  const std::string hwdir = dev + "/hwmon";
  if (DIR* d = ::opendir(hwdir.c_str())) {
    while (struct dirent* e = ::readdir(d)) {
      if (e->d_name[0] == '.') continue;
      std::string base = hwdir + "/" + e->d_name;
      // sensor numbering varies: consumer cards expose temp1 (edge), the
      // MI355X OAM exposes temp2 (junction) first — take the first present
      for (int t = 1; t <= 3 && st.temp_mc < 0; ++t)
        st.temp_mc =
            read_long(base + "/temp" + std::to_string(t) + "_input", -1);
      if (st.power_uw < 0) {
        st.power_uw = read_long(base + "/power1_average", -1);
        if (st.power_uw < 0)
          st.power_uw = read_long(base + "/power1_input", -1);
      }
    }
    ::closedir(d);
  }
  return st;
}

std::string default_sysfs_root() {
  const char* env = std::getenv("K3SAMD_SYSFS_ROOT");
  return env && *env ? env : "/sys";
}

Topology enumerate_topology(const std::string& sysfs_root) {
  Topology topo;
  // DKMS builds publish /sys/module/amdgpu/version; the in-tree driver does
  // not — fall back to the running kernel release (the driver version IS
  // the kernel version for in-tree amdgpu).
  topo.driver_version = trim(read_file(sysfs_root + "/module/amdgpu/version"));
  if (topo.driver_version.empty() &&
      is_dir(sysfs_root + "/module/amdgpu")) {
    std::string rel = trim(read_file("/proc/sys/kernel/osrelease"));
    if (!rel.empty()) topo.driver_version = "in-tree/" + rel;
  }

  const std::string nodes_dir = sysfs_root + "/class/kfd/kfd/topology/nodes";
  if (!is_dir(nodes_dir)) return topo;  // no KFD => no GPUs (CPU-only node)

  struct NodeInfo {
    bool is_gpu = false;
    int numa_hint = -1;
  };
  std::map<int, NodeInfo> nodes;

  for (int n : numeric_subdirs(nodes_dir)) {
    const std::string ndir = nodes_dir + "/" + std::to_string(n);
    auto props = parse_properties(ndir + "/properties");
    auto get = [&](const char* k) -> uint64_t {
      auto it = props.find(k);
      return it == props.end() ? 0 : it->second;
    };

    if (get("simd_count") == 0) {
      nodes[n].is_gpu = false;  // CPU/memory-only node
      continue;
    }
    nodes[n].is_gpu = true;

    GpuDevice g;
    g.kfd_node = n;
    g.name = trim(read_file(ndir + "/name"));
    g.unique_id = get("unique_id");
    g.vendor_id = (uint32_t)get("vendor_id");
    g.device_id = (uint32_t)get("device_id");
    g.simd_count = (uint32_t)get("simd_count");
    g.simd_per_cu = (uint32_t)get("simd_per_cu");
    g.gfx_target_version = (uint32_t)get("gfx_target_version");
    g.drm_render_minor = (int)get("drm_render_minor");
    g.pci_bdf = bdf_string(get("domain"), get("location_id"));

    // VRAM: sum framebuffer heaps (heap_type 1 = FB public, 2 = FB private)
    const std::string mdir = ndir + "/mem_banks";
    for (int m : numeric_subdirs(mdir)) {
      auto mp = parse_properties(mdir + "/" + std::to_string(m) +
                                 "/properties");
      uint64_t heap = mp.count("heap_type") ? mp["heap_type"] : 0;
      if (heap == 1 || heap == 2) {
        g.vram_bytes += mp.count("size_in_bytes") ? mp["size_in_bytes"] : 0;
      }
    }

    // io_links: count xGMI peers; remember a CPU peer as the NUMA hint
    const std::string ldir = ndir + "/io_links";
    for (int l : numeric_subdirs(ldir)) {
      auto lp = parse_properties(ldir + "/" + std::to_string(l) +
                                 "/properties");
      uint64_t type = lp.count("type") ? lp["type"] : 0;
      if (type == kIoLinkXgmi) {
        g.xgmi_links++;
        if (lp.count("node_to")) g.xgmi_peer_nodes.push_back((int)lp["node_to"]);
      } else if (type == kIoLinkPcie || type == 1 /*hypertransport*/) {
        if (lp.count("node_to")) g.numa_node = (int)lp["node_to"];
      }
    }

    // resolve /dev/dri/card index through /sys/class/drm by PCI address
    if (g.drm_render_minor >= 0) {
      const std::string drm = sysfs_root + "/class/drm";
      for (int idx = 0; idx < 256; ++idx) {
        std::string link = drm + "/card" + std::to_string(idx) + "/device";
        std::string ue = read_file(link + "/uevent");
        if (ue.find(g.pci_bdf) != std::string::npos) {
          g.card_index = idx;
          // The KFD `name` file on real MI355X silicon reports the IP
          // discovery string ("ip discovery"), not a marketing name; prefer
          // the PCI product_name the amdgpu driver publishes when present.
          std::string product = trim(read_file(link + "/product_name"));
          if (!product.empty() &&
              (g.name.empty() || g.name == "ip discovery")) {
            g.name = product;
          }
          break;
        }
      }
      if (g.name.empty() || g.name == "ip discovery") {
        g.name = "AMD GPU " + g.gfx_arch();
      }
    }

    topo.gpus.push_back(std::move(g));
  }

  // KFD numa hint above gives KFD node ids; translate to "CPU node ordinal"
  // (the i-th non-GPU node) which is what kubelet topology wants.
  std::vector<int> cpu_nodes;
  for (auto& [id, info] : nodes)
    if (!info.is_gpu) cpu_nodes.push_back(id);
  for (auto& g : topo.gpus) {
    auto it = std::find(cpu_nodes.begin(), cpu_nodes.end(), g.numa_node);
    g.numa_node = it == cpu_nodes.end() ? 0 : (int)(it - cpu_nodes.begin());
  }

  return topo;
}

}  // namespace k3samd
