// k3samd device plugin — advertises amd.com/gpu to kubelet.
//
// MI355X-native re-implementation of the role the reference fills with the
// NVIDIA K8s Device Plugin (/root/reference/README.md:105-126,
// values.yaml:6-18): enumerate GPUs (KFD topology), apply time-slicing
// replica fan-out, serve the v1beta1 DevicePlugin gRPC API on a kubelet
// plugin socket, register with kubelet, and watch device health.

#pragma once

#include <atomic>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

#include "../grpc/grpc_transport.h"
#include "../topology/gpu_health.h"
#include "../topology/kfd_topology.h"
#include "dp_messages.h"
#include "metrics.h"

namespace k3samd {

// Parsed device-plugin sharing config (values.yaml:6-18 schema).
struct PluginConfig {
  std::string version = "v1";
  std::string mig_strategy = "none";   // flags.migStrategy (parity knob)
  bool rename_by_default = false;      // sharing.timeSlicing.renameByDefault
  bool fail_requests_greater_than_one = false;
  int replicas = 1;                    // for resource `resource_name`
  std::string resource_name = "amd.com/gpu";
  bool use_cdi = false;                // Allocate returns CDI device names
  std::string cdi_kind = "amd.com/gpu";
  HealthPolicy health;                 // additive `health:` block (RAS/ECC
                                       // thresholds; see gpu_health.h)

  // Parse the `version: v1 / flags / sharing.timeSlicing` YAML document.
  static bool from_yaml(const std::string& text, PluginConfig& out,
                        std::string* err);
};

// One advertised (possibly virtual/time-sliced) device.
struct VirtualDevice {
  std::string id;        // "<stable_id>" or "<stable_id>::<r>"
  int gpu_index;         // index into Topology.gpus
  bool healthy = true;
};

class DevicePlugin {
 public:
  DevicePlugin(PluginConfig cfg, std::string sysfs_root,
               std::string dev_root = "/dev");
  ~DevicePlugin();

  // Re-enumerate the topology and rebuild the advertised device list.
  void refresh_topology();

  // kubelet-registration retry policy (kubelet may start after us).
  void set_register_policy(int retries, int backoff_ms) {
    register_retries_ = retries;
    register_backoff_ms_ = backoff_ms;
  }

  // Watch `path` for config changes (mtime/size) from the health loop;
  // a valid new config rebuilds the device list (pushed over
  // ListAndWatch); an invalid one is logged and ignored.
  void watch_config(const std::string& path);
  // exposed for tests: returns true if the config was reloaded
  bool poll_config_once();

  std::vector<VirtualDevice> devices() const;
  // resource name after renameByDefault is applied
  std::string advertised_resource() const;

  // Serve on `plugin_sock` and (if kubelet_sock non-empty) register with
  // kubelet. Non-blocking; returns false if the socket can't be bound or
  // registration fails. `metrics_addr` ("host:port" or "unix:/path", empty
  // = disabled) exposes Prometheus metrics.
  bool serve(const std::string& plugin_sock, const std::string& kubelet_sock,
             int health_poll_ms = 5000,
             const std::string& metrics_addr = "");
  std::string render_metrics();
  void stop();

  // --- logic, exposed for unit tests ---
  std::string handle_options();
  GrpcStatus handle_allocate(const std::string& req, std::string& resp);
  GrpcStatus handle_preferred(const std::string& req, std::string& resp);
  std::vector<dp::Device> current_device_list();
  // Re-check health from sysfs; returns true if the device list changed.
  bool poll_health_once();

 private:
  GrpcStatus handle_list_and_watch(const std::string& req,
                                   const GrpcServer::WriteFn& write);
  // find physical gpu for a virtual id; -1 if unknown
  int gpu_for_id(const std::string& vid) const;

  PluginConfig cfg_;
  std::string sysfs_root_;
  std::string dev_root_;
  std::string config_path_;
  long config_mtime_ = -1;
  long config_size_ = -1;

  mutable std::recursive_mutex mu_;
  Topology topo_;
  std::vector<VirtualDevice> devices_;
  uint64_t generation_ = 0;  // bumped on device-list changes

  // register with kubelet; returns false on failure
  bool register_with_kubelet();

  GrpcServer server_;
  PluginMetrics metrics_;
  MetricsServer metrics_server_;
  std::thread health_thread_;
  std::thread reregister_thread_;
  int register_retries_ = 5;       // extra attempts after the first
  int register_backoff_ms_ = 1000;
  std::string plugin_sock_;
  std::string kubelet_sock_;
  std::atomic<bool> stopping_{false};
};

}  // namespace k3samd
