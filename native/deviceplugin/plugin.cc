#include "plugin.h"

#include <algorithm>
#include <chrono>
#include <cstdio>
#include <fstream>
#include <map>
#include <set>
#include <sstream>
#include <sys/stat.h>
#include <thread>
#include <tuple>

#include "../common/miniyaml.h"

namespace k3samd {

bool PluginConfig::from_yaml(const std::string& text, PluginConfig& out,
                             std::string* err) {
  try {
    YNode root = yaml_parse(text);
    if (const YNode* v = root.get("version")) out.version = v->as_str("v1");
    if (out.version != "v1") {
      if (err) *err = "unsupported config version: " + out.version;
      return false;
    }
    if (const YNode* m = root.get_path("flags.migStrategy"))
      out.mig_strategy = m->as_str("none");
    if (out.mig_strategy != "none") {
      // MI355X partitioning (SR-IOV/NPS) is not scheduling-level sharing;
      // mirror the reference's `migStrategy: none` (values.yaml:11) only.
      if (err) *err = "only migStrategy 'none' is supported";
      return false;
    }
    // additive `health:` block — RAS/ECC thresholds (gpu_health.h). A
    // value of -1 disables that check.
    if (const YNode* h = root.get("health")) {
      if (const YNode* n = h->get("maxUncorrectableErrors"))
        out.health.max_uncorrectable = n->as_int(0);
      if (const YNode* n = h->get("maxCorrectableErrors"))
        out.health.max_correctable = n->as_int(10000);
      if (const YNode* n = h->get("maxDeferredErrors"))
        out.health.max_deferred = n->as_int(0);
      if (const YNode* n = h->get("maxFatalEvents"))
        out.health.max_fatal_events = n->as_int(0);
      if (const YNode* n = h->get("maxBadPages"))
        out.health.max_bad_pages = n->as_int(-1);
      if (const YNode* n = h->get("maxPcieReplays"))
        out.health.max_pcie_replays = n->as_int(-1);
      if (const YNode* n = h->get("maxResets"))
        out.health.max_resets = n->as_int(0);
    }
    const YNode* ts = root.get_path("sharing.timeSlicing");
    if (ts) {
      if (const YNode* n = ts->get("renameByDefault"))
        out.rename_by_default = n->as_bool(false);
      if (const YNode* n = ts->get("failRequestsGreaterThanOne"))
        out.fail_requests_greater_than_one = n->as_bool(false);
      if (const YNode* rs = ts->get("resources")) {
        for (const YNode& r : rs->list) {
          std::string name =
              r.get("name") ? r.get("name")->as_str() : out.resource_name;
          if (name == out.resource_name) {
            if (const YNode* rep = r.get("replicas")) {
              out.replicas = (int)rep->as_int(1);
              if (out.replicas < 1 || out.replicas > 256) {
                if (err) *err = "replicas out of range";
                return false;
              }
            }
          }
        }
      }
    }
    return true;
  } catch (const std::exception& e) {
    if (err) *err = e.what();
    return false;
  }
}

DevicePlugin::DevicePlugin(PluginConfig cfg, std::string sysfs_root,
                           std::string dev_root)
    : cfg_(std::move(cfg)),
      sysfs_root_(std::move(sysfs_root)),
      dev_root_(std::move(dev_root)) {
  refresh_topology();
}

DevicePlugin::~DevicePlugin() { stop(); }

void DevicePlugin::refresh_topology() {
  Topology topo = enumerate_topology(sysfs_root_);
  std::lock_guard<std::recursive_mutex> lk(mu_);
  topo_ = std::move(topo);
  devices_.clear();
  for (size_t gi = 0; gi < topo_.gpus.size(); ++gi) {
    const auto& g = topo_.gpus[gi];
    if (cfg_.replicas <= 1) {
      devices_.push_back({g.stable_id(), (int)gi, true});
    } else {
      for (int r = 0; r < cfg_.replicas; ++r) {
        devices_.push_back(
            {g.stable_id() + "::" + std::to_string(r), (int)gi, true});
      }
    }
  }
  ++generation_;
}

std::vector<VirtualDevice> DevicePlugin::devices() const {
  std::lock_guard<std::recursive_mutex> lk(mu_);
  return devices_;
}

std::string DevicePlugin::advertised_resource() const {
  // NVIDIA-plugin semantics (README.md:112 behavior surface): with sharing
  // enabled and renameByDefault, the resource is advertised as
  // "<name>.shared". cfg_ can be rewritten by config hot-reload, so read
  // under the lock (recursive: callers may already hold it).
  std::lock_guard<std::recursive_mutex> lk(mu_);
  if (cfg_.replicas > 1 && cfg_.rename_by_default)
    return cfg_.resource_name + ".shared";
  return cfg_.resource_name;
}

std::string DevicePlugin::handle_options() {
  dp::DevicePluginOptions opts;
  opts.get_preferred_allocation_available = true;
  opts.pre_start_required = false;
  return opts.encode();
}

std::vector<dp::Device> DevicePlugin::current_device_list() {
  std::lock_guard<std::recursive_mutex> lk(mu_);
  std::vector<dp::Device> out;
  for (const auto& vd : devices_) {
    dp::Device d;
    d.id = vd.id;
    d.health = vd.healthy ? dp::kHealthy : dp::kUnhealthy;
    d.numa_node = topo_.gpus[vd.gpu_index].numa_node;
    out.push_back(std::move(d));
  }
  return out;
}

int DevicePlugin::gpu_for_id(const std::string& vid) const {
  for (const auto& vd : devices_)
    if (vd.id == vid) return vd.gpu_index;
  return -1;
}

GrpcStatus DevicePlugin::handle_allocate(const std::string& req,
                                         std::string& resp) {
  std::vector<std::vector<std::string>> containers;
  if (!dp::decode_allocate_request(req, containers))
    return {13, "malformed AllocateRequest"};

  std::lock_guard<std::recursive_mutex> lk(mu_);
  std::vector<dp::ContainerAllocateResponse> crs;
  for (const auto& ids : containers) {
    if (cfg_.replicas > 1 && cfg_.fail_requests_greater_than_one &&
        ids.size() > 1) {
      metrics_.allocation_errors_total.fetch_add(1);
      return {3,
              "request for more than one " + advertised_resource() +
                  " is not allowed with time-slicing "
                  "(failRequestsGreaterThanOne=true)"};
    }
    dp::ContainerAllocateResponse cr;
    std::set<int> gpus;  // dedupe physical GPUs across replica ids
    for (const auto& id : ids) {
      int gi = -1;
      for (const auto& vd : devices_)
        if (vd.id == id) {
          gi = vd.gpu_index;
          break;
        }
      if (gi < 0) {
        metrics_.allocation_errors_total.fetch_add(1);
        return {3, "unknown device id " + id};
      }
      gpus.insert(gi);
    }
    std::string visible, minors;
    if (!cfg_.use_cdi) {
      // /dev/kfd is the compute entry point, shared by all GPUs
      cr.devices.push_back({"/dev/kfd", dev_root_ + "/kfd", "rw"});
    }
    for (int gi : gpus) {
      const auto& g = topo_.gpus[gi];
      if (cfg_.use_cdi) {
        // containerd resolves these against the k3samd-cdi-gen spec
        cr.cdi_devices.push_back(cfg_.cdi_kind + "=" + g.stable_id());
      } else {
        std::string rnode =
            "/dri/renderD" + std::to_string(g.drm_render_minor);
        cr.devices.push_back({"/dev" + rnode, dev_root_ + rnode, "rw"});
        if (g.card_index >= 0) {
          std::string cnode = "/dri/card" + std::to_string(g.card_index);
          cr.devices.push_back({"/dev" + cnode, dev_root_ + cnode, "rw"});
        }
      }
      if (!visible.empty()) visible += ",";
      visible += g.stable_id();
      if (!minors.empty()) minors += ",";
      minors += std::to_string(g.drm_render_minor);
    }
    cr.envs["K3SAMD_VISIBLE_DEVICES"] = visible;
    cr.envs["K3SAMD_RENDER_MINORS"] = minors;
    cr.annotations["k3samd.ai/allocated-gpus"] = visible;
    std::fprintf(stderr,
                 "deviceplugin: Allocate %zu id(s) -> gpus [%s] minors [%s]\n",
                 ids.size(), visible.c_str(), minors.c_str());
    metrics_.allocations_total.fetch_add(1);
    metrics_.allocated_devices_total.fetch_add(gpus.size());
    crs.push_back(std::move(cr));
  }
  resp = dp::encode_allocate_response(crs);
  return GrpcStatus::Ok();
}

GrpcStatus DevicePlugin::handle_preferred(const std::string& req,
                                          std::string& resp) {
  std::vector<dp::PreferredRequest> reqs;
  if (!dp::decode_preferred_request(req, reqs))
    return {13, "malformed PreferredAllocationRequest"};

  std::lock_guard<std::recursive_mutex> lk(mu_);
  std::vector<std::vector<std::string>> out;
  for (const auto& pr : reqs) {
    std::vector<std::string> chosen(pr.must_include);
    std::set<std::string> used(chosen.begin(), chosen.end());

    // Two regimes:
    //  * time-sliced (replicas>1): pack replicas of the same physical GPU
    //    together — keeps other physical GPUs free for exclusive jobs.
    //  * exclusive multi-GPU: prefer GPUs on the SAME host NUMA node as
    //    the must-include set (or the node with the most free GPUs) —
    //    xGMI is uniform all-to-all on the MI355X node, so the only
    //    locality that differs is the host CPU/PCIe side feeding H2D.
    std::vector<std::string> avail;
    for (const auto& id : pr.available)
      if (!used.count(id)) avail.push_back(id);
    int target_numa = -1;
    if (cfg_.replicas <= 1) {
      std::map<int, int> numa_free;
      for (const auto& id : pr.must_include) {
        int g = gpu_for_id(id);
        if (g >= 0) target_numa = topo_.gpus[g].numa_node;
      }
      if (target_numa < 0) {
        for (const auto& id : avail) {
          int g = gpu_for_id(id);
          if (g >= 0) numa_free[topo_.gpus[g].numa_node]++;
        }
        int best = -1;
        for (auto& [node, cnt] : numa_free)
          if (cnt > best) {
            best = cnt;
            target_numa = node;
          }
      }
    }
    std::stable_sort(avail.begin(), avail.end(),
                     [&](const std::string& a, const std::string& b) {
                       int ga = gpu_for_id(a), gb = gpu_for_id(b);
                       if (target_numa >= 0 && ga >= 0 && gb >= 0) {
                         bool na = topo_.gpus[ga].numa_node == target_numa;
                         bool nb = topo_.gpus[gb].numa_node == target_numa;
                         if (na != nb) return na;
                       }
                       if (ga != gb) return ga < gb;
                       return a < b;
                     });
    for (const auto& id : avail) {
      if ((int)chosen.size() >= pr.size) break;
      chosen.push_back(id);
    }
    if ((int)chosen.size() > pr.size) chosen.resize(pr.size);
    out.push_back(std::move(chosen));
  }
  resp = dp::encode_preferred_response(out);
  return GrpcStatus::Ok();
}

void DevicePlugin::watch_config(const std::string& path) {
  config_path_ = path;
  struct stat st{};
  if (::stat(path.c_str(), &st) == 0) {
    config_mtime_ = st.st_mtim.tv_sec * 1000000000L + st.st_mtim.tv_nsec;
    config_size_ = (long)st.st_size;
  }
}

bool DevicePlugin::poll_config_once() {
  if (config_path_.empty()) return false;
  struct stat st{};
  if (::stat(config_path_.c_str(), &st) != 0) return false;
  long mtime_ns = st.st_mtim.tv_sec * 1000000000L + st.st_mtim.tv_nsec;
  if (mtime_ns == config_mtime_ && (long)st.st_size == config_size_)
    return false;
  config_mtime_ = mtime_ns;
  config_size_ = (long)st.st_size;

  std::ifstream f(config_path_);
  std::stringstream ss;
  ss << f.rdbuf();
  PluginConfig fresh;
  fresh.resource_name = cfg_.resource_name;
  fresh.use_cdi = cfg_.use_cdi;
  fresh.cdi_kind = cfg_.cdi_kind;
  std::string err;
  if (!PluginConfig::from_yaml(ss.str(), fresh, &err)) {
    std::fprintf(stderr, "deviceplugin: ignoring invalid config update: %s\n",
                 err.c_str());
    return false;
  }
  std::string old_resource = advertised_resource();
  {
    std::lock_guard<std::recursive_mutex> lk(mu_);
    cfg_ = fresh;
  }
  refresh_topology();  // rebuilds devices with the new replicas; bumps gen
  std::fprintf(stderr,
               "deviceplugin: config reloaded (replicas=%d, %zu devices)\n",
               fresh.replicas, devices().size());
  // a renamed resource needs a fresh registration with kubelet
  if (!kubelet_sock_.empty() && advertised_resource() != old_resource)
    register_with_kubelet();
  return true;
}

bool DevicePlugin::poll_health_once() {
  Topology topo = enumerate_topology(sysfs_root_);
  std::set<std::string> present;
  for (const auto& g : topo.gpus) present.insert(g.stable_id());

  // RAS/error-state probe outside the lock (sysfs reads). A GPU is
  // unhealthy when it VANISHED from KFD *or* its error counters trip the
  // configured thresholds (sick-but-present: HBM ECC UEs, reset events).
  std::map<int, std::string> sick;  // gpu_index -> reason
  {
    std::lock_guard<std::recursive_mutex> lk(mu_);
    for (size_t gi = 0; gi < topo_.gpus.size(); ++gi) {
      GpuHealthCounters c =
          read_gpu_health(sysfs_root_, topo_.gpus[gi].card_index);
      std::string reason = health_verdict(c, cfg_.health);
      if (!reason.empty()) sick[(int)gi] = std::move(reason);
    }
  }

  std::lock_guard<std::recursive_mutex> lk(mu_);
  bool changed = false;
  for (auto& vd : devices_) {
    const auto& g = topo_.gpus[vd.gpu_index];
    bool healthy = present.count(g.stable_id()) > 0;
    auto it = sick.find(vd.gpu_index);
    if (healthy && it != sick.end()) healthy = false;
    if (healthy != vd.healthy) {
      vd.healthy = healthy;
      metrics_.health_transitions_total.fetch_add(1);
      changed = true;
      if (!healthy)
        std::fprintf(stderr, "k3samd-device-plugin: %s -> Unhealthy (%s)\n",
                     vd.id.c_str(),
                     it != sick.end() ? it->second.c_str()
                                      : "KFD node disappeared");
      else
        std::fprintf(stderr, "k3samd-device-plugin: %s -> Healthy\n",
                     vd.id.c_str());
    }
  }
  if (changed) {
    ++generation_;
  }
  return changed;
}

GrpcStatus DevicePlugin::handle_list_and_watch(
    const std::string&, const GrpcServer::WriteFn& write) {
  uint64_t seen_gen;
  {
    std::lock_guard<std::recursive_mutex> lk(mu_);
    seen_gen = generation_;
  }
  metrics_.list_and_watch_updates_total.fetch_add(1);
  if (!write(dp::encode_list_and_watch(current_device_list())))
    return GrpcStatus::Ok();
  // Push an update whenever the device list changes (kubelet keeps this
  // stream open for the plugin's lifetime). A bounded 100 ms poll keeps the
  // concurrency trivial — health transitions are second-granularity events.
  while (!stopping_.load()) {
    std::this_thread::sleep_for(std::chrono::milliseconds(100));
    uint64_t gen;
    {
      std::lock_guard<std::recursive_mutex> lk(mu_);
      gen = generation_;
    }
    if (gen == seen_gen) continue;
    seen_gen = gen;
    metrics_.list_and_watch_updates_total.fetch_add(1);
    if (!write(dp::encode_list_and_watch(current_device_list()))) break;
  }
  return GrpcStatus::Ok();
}

std::string DevicePlugin::render_metrics() {
  size_t healthy = 0, unhealthy = 0;
  std::vector<GpuDevice> gpus;
  {
    std::lock_guard<std::recursive_mutex> lk(mu_);
    for (const auto& vd : devices_) (vd.healthy ? healthy : unhealthy)++;
    gpus = topo_.gpus;
  }
  // per-GPU runtime gauges (node-exporter style, read fresh per scrape)
  std::string per_gpu =
      "# TYPE k3samd_gpu_busy_percent gauge\n"
      "# TYPE k3samd_gpu_temp_celsius gauge\n"
      "# TYPE k3samd_gpu_power_watts gauge\n"
      "# TYPE k3samd_gpu_vram_used_bytes gauge\n"
      "# TYPE k3samd_gpu_ras_uncorrectable_errors gauge\n"
      "# TYPE k3samd_gpu_ras_correctable_errors gauge\n"
      "# TYPE k3samd_gpu_reset_count gauge\n";
  for (const auto& g : gpus) {
    GpuRuntimeStats st = read_runtime_stats(sysfs_root_, g.card_index);
    char line[512];
    std::snprintf(line, sizeof(line),
                  "k3samd_gpu_busy_percent{gpu=\"%s\"} %ld\n"
                  "k3samd_gpu_temp_celsius{gpu=\"%s\"} %.1f\n"
                  "k3samd_gpu_power_watts{gpu=\"%s\"} %.1f\n"
                  "k3samd_gpu_vram_used_bytes{gpu=\"%s\"} %llu\n",
                  g.stable_id().c_str(), st.busy_percent,
                  g.stable_id().c_str(),
                  st.temp_mc < 0 ? -1.0 : st.temp_mc / 1000.0,
                  g.stable_id().c_str(),
                  st.power_uw < 0 ? -1.0 : st.power_uw / 1e6,
                  g.stable_id().c_str(), (unsigned long long)st.vram_used);
    per_gpu += line;
    // RAS/error-state gauges (-1 = not exposed by this driver build) —
    // the same counters the health model thresholds on, so an operator
    // can alert BEFORE a GPU flips Unhealthy
    GpuHealthCounters hc = read_gpu_health(sysfs_root_, g.card_index);
    std::snprintf(line, sizeof(line),
                  "k3samd_gpu_ras_uncorrectable_errors{gpu=\"%s\"} %ld\n"
                  "k3samd_gpu_ras_correctable_errors{gpu=\"%s\"} %ld\n"
                  "k3samd_gpu_reset_count{gpu=\"%s\"} %ld\n",
                  g.stable_id().c_str(), hc.ras_ue, g.stable_id().c_str(),
                  hc.ras_ce, g.stable_id().c_str(), hc.reset_count);
    per_gpu += line;
  }
  char buf[2048];
  std::snprintf(
      buf, sizeof(buf),
      "# HELP k3samd_gpu_devices Advertised devices by health.\n"
      "# TYPE k3samd_gpu_devices gauge\n"
      "k3samd_gpu_devices{health=\"healthy\"} %zu\n"
      "k3samd_gpu_devices{health=\"unhealthy\"} %zu\n"
      "# TYPE k3samd_allocations_total counter\n"
      "k3samd_allocations_total %llu\n"
      "# TYPE k3samd_allocated_devices_total counter\n"
      "k3samd_allocated_devices_total %llu\n"
      "# TYPE k3samd_allocation_errors_total counter\n"
      "k3samd_allocation_errors_total %llu\n"
      "# TYPE k3samd_list_and_watch_updates_total counter\n"
      "k3samd_list_and_watch_updates_total %llu\n"
      "# TYPE k3samd_registrations_total counter\n"
      "k3samd_registrations_total %llu\n"
      "# TYPE k3samd_health_transitions_total counter\n"
      "k3samd_health_transitions_total %llu\n",
      healthy, unhealthy,
      (unsigned long long)metrics_.allocations_total.load(),
      (unsigned long long)metrics_.allocated_devices_total.load(),
      (unsigned long long)metrics_.allocation_errors_total.load(),
      (unsigned long long)metrics_.list_and_watch_updates_total.load(),
      (unsigned long long)metrics_.registrations_total.load(),
      (unsigned long long)metrics_.health_transitions_total.load());
  return std::string(buf) + per_gpu;
}

bool DevicePlugin::serve(const std::string& plugin_sock,
                         const std::string& kubelet_sock, int health_poll_ms,
                         const std::string& metrics_addr) {
  server_.add_unary(dp::kOptionsPath,
                    [this](const std::string&, std::string& resp) {
                      resp = handle_options();
                      return GrpcStatus::Ok();
                    });
  server_.add_unary(dp::kAllocatePath,
                    [this](const std::string& req, std::string& resp) {
                      return handle_allocate(req, resp);
                    });
  server_.add_unary(dp::kPreferredPath,
                    [this](const std::string& req, std::string& resp) {
                      return handle_preferred(req, resp);
                    });
  server_.add_unary(dp::kPreStartPath,
                    [](const std::string&, std::string& resp) {
                      resp.clear();
                      return GrpcStatus::Ok();
                    });
  server_.add_server_stream(
      dp::kListAndWatchPath,
      [this](const std::string& req, const GrpcServer::WriteFn& write) {
        return handle_list_and_watch(req, write);
      });

  if (!server_.start(plugin_sock)) {
    std::fprintf(stderr, "deviceplugin: cannot bind %s\n",
                 plugin_sock.c_str());
    return false;
  }

  plugin_sock_ = plugin_sock;
  kubelet_sock_ = kubelet_sock;
  stopping_.store(false);

  if (!kubelet_sock.empty()) {
    // kubelet may come up after this DaemonSet pod (node boot ordering):
    // retry with linear backoff before giving up to CrashLoop semantics.
    bool registered = false;
    for (int attempt = 0; attempt <= register_retries_; ++attempt) {
      if (attempt > 0) {
        std::fprintf(stderr,
                     "deviceplugin: Register retry %d/%d in %d ms\n",
                     attempt, register_retries_, register_backoff_ms_);
        std::this_thread::sleep_for(
            std::chrono::milliseconds(register_backoff_ms_));
      }
      if (register_with_kubelet()) {
        registered = true;
        break;
      }
    }
    if (!registered) {
      server_.stop();
      return false;
    }
    // kubelet drops all plugin registrations when it restarts: watch the
    // registration socket's inode and re-register when it changes (the
    // same recovery loop production device plugins implement).
    reregister_thread_ = std::thread([this] {
      // identity = (inode, ctime): tmpfs reuses inode numbers on an
      // unlink+rebind, so the inode alone can miss a kubelet restart
      auto ident = [](const struct stat& st) {
        return std::make_tuple(st.st_ino, st.st_ctim.tv_sec,
                               st.st_ctim.tv_nsec);
      };
      struct stat st{};
      auto last = ::stat(kubelet_sock_.c_str(), &st) == 0
                      ? ident(st)
                      : std::make_tuple((ino_t)0, (time_t)0, 0l);
      while (!stopping_.load()) {
        std::this_thread::sleep_for(std::chrono::milliseconds(500));
        if (stopping_.load()) break;
        struct stat now{};
        if (::stat(kubelet_sock_.c_str(), &now) != 0) continue;
        if (ident(now) != last) {
          std::fprintf(stderr,
                       "deviceplugin: kubelet socket changed, re-registering\n");
          if (register_with_kubelet()) last = ident(now);
        }
      }
    });
  }

  if (!metrics_addr.empty() &&
      !metrics_server_.start(metrics_addr, [this] { return render_metrics(); })) {
    std::fprintf(stderr, "deviceplugin: cannot bind metrics %s\n",
                 metrics_addr.c_str());
  }

  if (health_poll_ms > 0) {
    health_thread_ = std::thread([this, health_poll_ms] {
      while (!stopping_.load()) {
        std::this_thread::sleep_for(std::chrono::milliseconds(health_poll_ms));
        if (stopping_.load()) break;
        poll_config_once();
        poll_health_once();
      }
    });
  }
  return true;
}

bool DevicePlugin::register_with_kubelet() {
  dp::RegisterRequest rr;
  size_t slash = plugin_sock_.find_last_of('/');
  rr.endpoint = slash == std::string::npos ? plugin_sock_
                                           : plugin_sock_.substr(slash + 1);
  rr.resource_name = advertised_resource();
  auto res = grpc_unary_call(kubelet_sock_, dp::kRegisterPath, rr.encode());
  if (res.transport_ok && res.grpc_status == 0)
    metrics_.registrations_total.fetch_add(1);
  if (!res.transport_ok || res.grpc_status != 0) {
    std::fprintf(stderr,
                 "deviceplugin: Register with kubelet failed: %s (grpc=%d %s)\n",
                 res.error.c_str(), res.grpc_status, res.grpc_message.c_str());
    return false;
  }
  return true;
}

void DevicePlugin::stop() {
  stopping_.store(true);
  if (health_thread_.joinable()) health_thread_.join();
  if (reregister_thread_.joinable()) reregister_thread_.join();
  metrics_server_.stop();
  server_.stop();
}

}  // namespace k3samd
