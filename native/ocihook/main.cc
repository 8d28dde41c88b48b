// k3samd-oci-runtime — OCI runtime wrapper for containerd RuntimeClass `amd`.
//
// Drop-in runc front-end: on `create`/`run` it rewrites the bundle's
// config.json (oci_inject.h: /dev/kfd + selected /dev/dri nodes + cgroup
// rules + optional ROCm mounts), then execs the real runc with unchanged
// arguments. This replaces the reference's nvidia-container-runtime layer
// (/root/reference/README.md:57-69; activated by runtimeClassName,
// README.md:164).
//
// containerd registration (deploy/containerd-runtime.md):
//   [plugins."io.containerd.grpc.v1.cri".containerd.runtimes.amd]
//     runtime_type = "io.containerd.runc.v2"
//     [plugins."io.containerd.grpc.v1.cri".containerd.runtimes.amd.options]
//       BinaryName = "/usr/local/bin/k3samd-oci-runtime"
//
// Test mode:
//   k3samd-oci-runtime --transform-only <config.json>   (writes in place)

#include <cstdio>
#include <cstring>
#include <fstream>
#include <sstream>
#include <unistd.h>

#include "oci_inject.h"

namespace {

std::string read_file(const std::string& p) {
  std::ifstream f(p);
  std::stringstream ss;
  ss << f.rdbuf();
  return ss.str();
}

bool transform_config(const std::string& path) {
  std::string text = read_file(path);
  if (text.empty()) {
    std::fprintf(stderr, "k3samd-oci-runtime: cannot read %s\n", path.c_str());
    return false;
  }
  k3samd::JPtr config;
  try {
    config = k3samd::json_parse(text);
  } catch (const std::exception& e) {
    std::fprintf(stderr, "k3samd-oci-runtime: %s: %s\n", path.c_str(),
                 e.what());
    return false;
  }
  k3samd::InjectOptions opts;
  opts.sysfs_root = k3samd::default_sysfs_root();
  if (const char* d = std::getenv("K3SAMD_DEV_ROOT")) opts.dev_root = d;
  if (const char* r = std::getenv("K3SAMD_ROCM_ROOT")) opts.rocm_root = r;
  if (const char* e = std::getenv("K3SAMD_INJECT_ROCM_DEFAULT"))
    opts.inject_rocm_default = !std::strcmp(e, "1");
  if (const char* e = std::getenv("K3SAMD_ALLOW_ALL"))
    opts.allow_all_default = !std::strcmp(e, "1");

  k3samd::Topology topo = k3samd::enumerate_topology(opts.sysfs_root);
  k3samd::InjectReport report;
  if (!k3samd::oci_inject_gpus(config, topo, opts, &report)) {
    std::fprintf(stderr, "k3samd-oci-runtime: malformed OCI spec %s\n",
                 path.c_str());
    return false;
  }
  std::ofstream out(path, std::ios::trunc);
  out << k3samd::json_serialize(config);
  if (!out) {
    std::fprintf(stderr, "k3samd-oci-runtime: cannot write %s\n",
                 path.c_str());
    return false;
  }
  std::fprintf(stderr,
               "k3samd-oci-runtime: injected %zu device(s), %zu mount(s)%s\n",
               report.devices_added.size(), report.mounts_added.size(),
               report.no_allocation
                   ? " (skipped: no allocation — default-deny; set "
                     "K3SAMD_VISIBLE_DEVICES or K3SAMD_ALLOW_ALL=1)"
                   : (report.skipped ? " (skipped: visible=none)" : ""));
  return true;
}

}  // namespace

int main(int argc, char** argv) {
  // NOTE: --version is deliberately NOT intercepted — containerd probes
  // runtimes with `runc --version` and parses runc's output format, so it
  // must pass through to the real runc below.
  // explicit test mode
  if (argc >= 3 && !std::strcmp(argv[1], "--transform-only")) {
    return transform_config(argv[2]) ? 0 : 1;
  }

  // runc-compatible invocation: find subcommand + bundle. Global flags can
  // take values ("--root /run/..."), so the subcommand is the first token
  // matching runc's command set, not merely the first non-dash token.
  static const char* kRuncCommands[] = {
      "checkpoint", "create", "delete", "events", "exec", "features", "init",
      "kill", "list", "pause", "ps", "restore", "resume", "run", "spec",
      "start", "state", "update"};
  const char* subcommand = nullptr;
  std::string bundle = ".";
  for (int i = 1; i < argc; ++i) {
    const char* a = argv[i];
    if (!subcommand && a[0] != '-') {
      for (const char* c : kRuncCommands) {
        if (!std::strcmp(a, c)) {
          subcommand = a;
          break;
        }
      }
    }
    if ((!std::strcmp(a, "--bundle") || !std::strcmp(a, "-b")) &&
        i + 1 < argc) {
      bundle = argv[++i];
    } else if (!std::strncmp(a, "--bundle=", 9)) {
      bundle = a + 9;
    }
  }

  if (subcommand &&
      (!std::strcmp(subcommand, "create") || !std::strcmp(subcommand, "run"))) {
    std::string cfg = bundle + "/config.json";
    if (!transform_config(cfg)) return 1;  // refuse to start unobserved
  }

  // exec the real runc with identical arguments
  const char* runc = std::getenv("K3SAMD_RUNC_PATH");
  if (!runc) {
    for (const char* cand :
         {"/usr/local/sbin/runc", "/usr/sbin/runc", "/usr/bin/runc",
          "/usr/local/bin/runc"}) {
      if (::access(cand, X_OK) == 0) {
        runc = cand;
        break;
      }
    }
  }
  std::vector<char*> args;
  args.push_back(const_cast<char*>(runc ? runc : "runc"));
  for (int i = 1; i < argc; ++i) args.push_back(argv[i]);
  args.push_back(nullptr);
  if (runc)
    ::execv(runc, args.data());
  else
    ::execvp("runc", args.data());
  std::fprintf(stderr, "k3samd-oci-runtime: failed to exec runc\n");
  return 127;
}
