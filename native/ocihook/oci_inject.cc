#include "oci_inject.h"

#include <sys/stat.h>
#include <sys/sysmacros.h>

#include <algorithm>
#include <set>
#include <sstream>

namespace k3samd {

namespace {

// canonical fallbacks when the host node can't be stat'ed (unit tests,
// transform-only mode on CPU boxes): DRI is always char major 226 with
// renderD at minor 128+k and card at minor k; kfd is a misc char device.
constexpr int kDriMajor = 226;
constexpr int kKfdFallbackMajor = 10;
constexpr int kKfdFallbackMinor = 241;

struct DevNode {
  std::string path;  // container path, e.g. /dev/dri/renderD128
  int major;
  int minor;
};

bool stat_cdev(const std::string& host_path, int& major, int& minor) {
  struct stat st;
  if (::stat(host_path.c_str(), &st) != 0) return false;
  if (!S_ISCHR(st.st_mode)) return false;
  major = (int)major(st.st_rdev);
  minor = (int)minor(st.st_rdev);
  return true;
}

std::string env_lookup(const JPtr& env_arr, const std::string& key) {
  if (!env_arr || env_arr->type != JValue::kArray) return {};
  std::string prefix = key + "=";
  for (const auto& e : env_arr->arr) {
    if (e->type == JValue::kString && e->str.rfind(prefix, 0) == 0)
      return e->str.substr(prefix.size());
  }
  return {};
}

std::vector<std::string> split_csv(const std::string& s) {
  std::vector<std::string> out;
  std::stringstream ss(s);
  std::string item;
  while (std::getline(ss, item, ','))
    if (!item.empty()) out.push_back(item);
  return out;
}

bool has_device(const JPtr& devices_arr, const std::string& path) {
  for (const auto& d : devices_arr->arr) {
    JPtr p = d->get("path");
    if (p && p->type == JValue::kString && p->str == path) return true;
  }
  return false;
}

bool has_mount_dest(const JPtr& mounts_arr, const std::string& dest) {
  for (const auto& m : mounts_arr->arr) {
    JPtr p = m->get("destination");
    if (p && p->type == JValue::kString && p->str == dest) return true;
  }
  return false;
}

void add_device(const JPtr& linux_obj, const DevNode& dn,
                InjectReport* report) {
  JPtr devices = linux_obj->ensure_arr("devices");
  if (!has_device(devices, dn.path)) {
    auto d = JValue::make_obj();
    d->set("path", JValue::make_str(dn.path));
    d->set("type", JValue::make_str("c"));
    d->set("major", JValue::make_int(dn.major));
    d->set("minor", JValue::make_int(dn.minor));
    d->set("fileMode", JValue::make_int(0666));
    d->set("uid", JValue::make_int(0));
    d->set("gid", JValue::make_int(0));
    devices->arr.push_back(d);
    if (report) report->devices_added.push_back(dn.path);
  }
  JPtr rules = linux_obj->ensure_obj("resources")->ensure_arr("devices");
  // avoid duplicate allow rules
  for (const auto& r : rules->arr) {
    JPtr maj = r->get("major"), min = r->get("minor"), allow = r->get("allow");
    if (allow && allow->type == JValue::kBool && allow->b && maj && min &&
        maj->as_int(-1) == dn.major && min->as_int(-1) == dn.minor)
      return;
  }
  auto rule = JValue::make_obj();
  rule->set("allow", JValue::make_bool(true));
  rule->set("type", JValue::make_str("c"));
  rule->set("major", JValue::make_int(dn.major));
  rule->set("minor", JValue::make_int(dn.minor));
  rule->set("access", JValue::make_str("rwm"));
  rules->arr.push_back(rule);
}

void add_bind_mount(const JPtr& config, const std::string& dest,
                    const std::string& src, InjectReport* report) {
  JPtr mounts = config->ensure_arr("mounts");
  if (has_mount_dest(mounts, dest)) return;
  auto m = JValue::make_obj();
  m->set("destination", JValue::make_str(dest));
  m->set("type", JValue::make_str("bind"));
  m->set("source", JValue::make_str(src));
  auto opts = JValue::make_arr();
  for (const char* o : {"rbind", "ro", "nosuid", "nodev"})
    opts->arr.push_back(JValue::make_str(o));
  m->set("options", opts);
  mounts->arr.push_back(m);
  if (report) report->mounts_added.push_back(dest);
}

}  // namespace

bool oci_inject_gpus(const JPtr& config, const Topology& topo,
                     const InjectOptions& opts, InjectReport* report) {
  if (!config || config->type != JValue::kObject) return false;
  JPtr process = config->get("process");
  JPtr env = process ? process->get("env") : nullptr;

  std::string visible = env_lookup(env, "K3SAMD_VISIBLE_DEVICES");
  std::string minors_env = env_lookup(env, "K3SAMD_RENDER_MINORS");
  std::string inject_rocm_env = env_lookup(env, "K3SAMD_INJECT_ROCM");

  // fallback: the device plugin also stamps its allocation as an
  // annotation (k3samd.ai/allocated-gpus); some runtimes propagate
  // annotations but scrub env
  if (visible.empty() && minors_env.empty()) {
    JPtr ann = config->get("annotations");
    if (ann && ann->type == JValue::kObject) {
      JPtr a = ann->get("k3samd.ai/allocated-gpus");
      if (a && a->type == JValue::kString) visible = a->str;
    }
  }

  if (visible == "none" || visible == "void") {
    if (report) report->skipped = true;
    return true;
  }

  // Default-deny: a container with NO allocation (no env, no annotation)
  // gets nothing. Injecting every GPU here would let any pod that merely
  // sets runtimeClassName bypass kubelet device accounting — the
  // NVIDIA_VISIBLE_DEVICES=all foot-gun. Injecting all GPUs requires the
  // explicit "all" sentinel, or the operator-level allow_all_default
  // runtime flag (K3SAMD_ALLOW_ALL=1 on the runtime binary, debug only).
  if (visible.empty() && minors_env.empty() && !opts.allow_all_default) {
    if (report) {
      report->skipped = true;
      report->no_allocation = true;
    }
    return true;
  }

  // pick GPUs: "all" sentinel (or operator allow-all default) = every GPU;
  // else by stable id; render-minor list also honored (set by the device
  // plugin alongside the id list)
  std::vector<const GpuDevice*> selected;
  if (visible == "all" || (visible.empty() && minors_env.empty())) {
    for (const auto& g : topo.gpus) selected.push_back(&g);
  } else {
    std::set<std::string> want_ids;
    for (auto& s : split_csv(visible)) want_ids.insert(s);
    std::set<int> want_minors;
    for (auto& s : split_csv(minors_env))
      want_minors.insert(std::atoi(s.c_str()));
    for (const auto& g : topo.gpus) {
      if (want_ids.count(g.stable_id()) ||
          want_minors.count(g.drm_render_minor))
        selected.push_back(&g);
    }
  }

  JPtr linux_obj = config->ensure_obj("linux");

  // /dev/kfd — the ROCm compute entry point (shared across GPUs)
  if (!selected.empty()) {
    DevNode kfd{"/dev/kfd", kKfdFallbackMajor, kKfdFallbackMinor};
    stat_cdev(opts.dev_root + "/kfd", kfd.major, kfd.minor);
    add_device(linux_obj, kfd, report);
  }

  for (const GpuDevice* g : selected) {
    DevNode rn{"/dev/dri/renderD" + std::to_string(g->drm_render_minor),
               kDriMajor, g->drm_render_minor};
    stat_cdev(opts.dev_root + "/dri/renderD" +
                  std::to_string(g->drm_render_minor),
              rn.major, rn.minor);
    add_device(linux_obj, rn, report);
    if (g->card_index >= 0) {
      DevNode cn{"/dev/dri/card" + std::to_string(g->card_index), kDriMajor,
                 g->card_index};
      stat_cdev(opts.dev_root + "/dri/card" + std::to_string(g->card_index),
                cn.major, cn.minor);
      add_device(linux_obj, cn, report);
    }
  }

  bool inject_rocm = opts.inject_rocm_default;
  if (inject_rocm_env == "1" || inject_rocm_env == "true") inject_rocm = true;
  if (inject_rocm_env == "0" || inject_rocm_env == "false")
    inject_rocm = false;
  if (inject_rocm && !selected.empty()) {
    struct stat st;
    if (::stat(opts.rocm_root.c_str(), &st) == 0 && S_ISDIR(st.st_mode))
      add_bind_mount(config, "/opt/rocm", opts.rocm_root, report);
  }

  return true;
}

}  // namespace k3samd
