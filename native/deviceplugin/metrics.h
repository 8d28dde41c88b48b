// Minimal Prometheus text-format metrics endpoint for the device plugin.
//
// The reference stack's observability is CLI-driven; this
// adds the cluster-native layer on top: a tiny HTTP/1.0 responder serving
// /metrics on a TCP address or unix socket, with the counters an operator
// actually pages on (allocations, device health, stream updates).

#pragma once

#include <atomic>
#include <functional>
#include <string>
#include <thread>

namespace k3samd {

struct PluginMetrics {
  std::atomic<uint64_t> allocations_total{0};
  std::atomic<uint64_t> allocated_devices_total{0};
  std::atomic<uint64_t> allocation_errors_total{0};
  std::atomic<uint64_t> list_and_watch_updates_total{0};
  std::atomic<uint64_t> registrations_total{0};
  std::atomic<uint64_t> health_transitions_total{0};
};

class MetricsServer {
 public:
  // `render` returns the full text exposition (gauges need live state).
  using RenderFn = std::function<std::string()>;

  MetricsServer() = default;
  ~MetricsServer();

  // addr: "unix:/path/metrics.sock" or "127.0.0.1:9400". Returns false if
  // the socket can't be bound.
  bool start(const std::string& addr, RenderFn render);
  void stop();

 private:
  void serve_loop();
  RenderFn render_;
  std::atomic<int> listen_fd_{-1};
  std::thread thread_;
  std::atomic<bool> stopping_{false};
  std::atomic<int> active_{0};  // in-flight connection threads
};

}  // namespace k3samd
