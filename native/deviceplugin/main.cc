// k3samd-device-plugin — kubelet device plugin daemon for AMD MI355X.
//
// Usage:
//   k3samd-device-plugin [--config /etc/k3samd/config.yaml]
//                        [--plugin-sock /var/lib/kubelet/device-plugins/amd-gpu.sock]
//                        [--kubelet-sock /var/lib/kubelet/device-plugins/kubelet.sock]
//                        [--health-poll-ms 5000] [--oneshot]
//
// The config file is the `version: v1` document the Helm chart embeds in a
// ConfigMap (same schema the reference passes through values.yaml:6-18).
// --oneshot prints the advertised device list as JSON and exits (used by
// docs/tests; no kubelet needed).

#include <csignal>
#include <cstdio>
#include <cstring>
#include <fstream>
#include <semaphore.h>
#include <sstream>

#include "../common/json_writer.h"
#include "../common/version.h"
#include "plugin.h"

namespace {

volatile std::sig_atomic_t g_stop = 0;
void on_signal(int) { g_stop = 1; }

std::string read_file(const std::string& p) {
  std::ifstream f(p);
  std::stringstream ss;
  ss << f.rdbuf();
  return ss.str();
}

}  // namespace

int main(int argc, char** argv) {
  if (k3samd::handle_version_flag(argc, argv, "k3samd-device-plugin")) return 0;
  std::string config_path;
  std::string plugin_sock = "/var/lib/kubelet/device-plugins/amd-gpu.sock";
  std::string kubelet_sock = "/var/lib/kubelet/device-plugins/kubelet.sock";
  int health_poll_ms = 5000;
  bool oneshot = false;
  bool use_cdi = false;
  int register_retries = 5;
  int register_backoff_ms = 1000;
  std::string metrics_addr;

  for (int i = 1; i < argc; ++i) {
    auto arg = [&](const char* name) -> const char* {
      if (std::strcmp(argv[i], name) == 0 && i + 1 < argc) return argv[++i];
      return nullptr;
    };
    if (const char* v = arg("--config")) config_path = v;
    else if (const char* v = arg("--plugin-sock")) plugin_sock = v;
    else if (const char* v = arg("--kubelet-sock")) kubelet_sock = v;
    else if (const char* v = arg("--health-poll-ms")) health_poll_ms = std::atoi(v);
    else if (!std::strcmp(argv[i], "--no-register")) kubelet_sock.clear();
    else if (const char* v = arg("--register-retries")) register_retries = std::atoi(v);
    else if (const char* v = arg("--register-backoff-ms")) register_backoff_ms = std::atoi(v);
    else if (!std::strcmp(argv[i], "--use-cdi")) use_cdi = true;
    else if (const char* v = arg("--metrics-addr")) metrics_addr = v;
    else if (!std::strcmp(argv[i], "--oneshot")) oneshot = true;
    else if (!std::strcmp(argv[i], "--help") || !std::strcmp(argv[i], "-h")) {
      std::printf("k3samd-device-plugin [--config F] [--plugin-sock S] "
                  "[--kubelet-sock S|--no-register] [--health-poll-ms N] "
                  "[--oneshot]\n");
      return 0;
    } else {
      std::fprintf(stderr, "unknown flag %s\n", argv[i]);
      return 2;
    }
  }

  k3samd::PluginConfig cfg;
  if (!config_path.empty()) {
    std::string err;
    if (!k3samd::PluginConfig::from_yaml(read_file(config_path), cfg, &err)) {
      std::fprintf(stderr, "bad config %s: %s\n", config_path.c_str(),
                   err.c_str());
      return 2;
    }
  }

  cfg.use_cdi = use_cdi;
  k3samd::DevicePlugin plugin(cfg, k3samd::default_sysfs_root());
  plugin.set_register_policy(register_retries, register_backoff_ms);
  if (!config_path.empty()) plugin.watch_config(config_path);

  if (oneshot) {
    k3samd::JsonWriter w;
    w.begin_obj();
    w.key("resource").value(plugin.advertised_resource());
    auto devs = plugin.devices();
    w.key("allocatable").value((uint64_t)devs.size());
    w.key("devices").begin_arr();
    for (const auto& d : devs) {
      w.begin_obj();
      w.key("id").value(d.id);
      w.key("gpu_index").value(d.gpu_index);
      w.key("healthy").value(d.healthy);
      w.end_obj();
    }
    w.end_arr();
    w.end_obj();
    std::printf("%s\n", w.str().c_str());
    return 0;
  }

  std::signal(SIGINT, on_signal);
  std::signal(SIGTERM, on_signal);

  if (!plugin.serve(plugin_sock, kubelet_sock, health_poll_ms, metrics_addr))
    return 1;
  std::fprintf(stderr,
               "k3samd-device-plugin: serving %s on %s (%zu devices)%s\n",
               plugin.advertised_resource().c_str(), plugin_sock.c_str(),
               plugin.devices().size(),
               kubelet_sock.empty() ? " [not registered]" : "");

  while (!g_stop) {
    struct timespec ts = {0, 200 * 1000 * 1000};
    nanosleep(&ts, nullptr);
  }
  plugin.stop();
  return 0;
}
