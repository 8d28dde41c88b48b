// Self-contained gRPC-over-HTTP/2 transport (unix sockets, h2c).
//
// The kubelet device-plugin API is gRPC over a unix socket
// (/var/lib/kubelet/device-plugins/kubelet.sock). This image ships no C++
// gRPC, so the plugin implements the slice of HTTP/2 (RFC 7540) + HPACK
// (RFC 7541) + gRPC framing that the protocol needs, natively:
//   server: unary + server-streaming RPCs (ListAndWatch), flow control,
//           PING/SETTINGS/WINDOW_UPDATE/RST_STREAM/GOAWAY handling;
//   client: blocking unary call (Registration.Register).
// Interoperability is tested against grpcio (tests/test_deviceplugin.py,
// the same gRPC core family kubelet's grpc-go belongs to), against
// hand-built raw frames for the shapes grpcio never emits (padding,
// CONTINUATION — tests/test_http2_raw.py), and against hostile byte
// streams (tests/test_chaos.py). RFC 7541 Appendix C vectors validate the
// HPACK decoder (native/selftest).

#pragma once

#include <atomic>
#include <condition_variable>
#include <functional>
#include <map>
#include <memory>
#include <mutex>
#include <set>
#include <string>
#include <thread>
#include <vector>

namespace k3samd {

struct GrpcStatus {
  int code = 0;  // 0 = OK; 12 = UNIMPLEMENTED; 13 = INTERNAL ...
  std::string message;
  static GrpcStatus Ok() { return {}; }
};

class GrpcServer {
 public:
  using UnaryHandler =
      std::function<GrpcStatus(const std::string& request, std::string& response)>;
  // Server-streaming handler: call write() once per message (returns false
  // when the stream/connection is gone — return promptly then).
  using WriteFn = std::function<bool(const std::string& message)>;
  using StreamHandler =
      std::function<GrpcStatus(const std::string& request, const WriteFn& write)>;

  GrpcServer() = default;
  ~GrpcServer();

  void add_unary(const std::string& path, UnaryHandler h);
  void add_server_stream(const std::string& path, StreamHandler h);

  // Bind + listen on `unix_path` (unlinked first) and serve on a background
  // thread. Returns false if the socket can't be bound.
  bool start(const std::string& unix_path);
  void stop();

  bool running() const { return running_.load(); }

 private:
  void serve_loop();
  void connection_loop(const std::shared_ptr<struct H2Conn>& conn);

  std::map<std::string, UnaryHandler> unary_;
  std::map<std::string, StreamHandler> stream_;
  std::atomic<int> listen_fd_{-1};
  std::string bound_path_;
  std::thread accept_thread_;
  std::atomic<bool> running_{false};
  std::atomic<bool> stopping_{false};
  std::mutex conns_mu_;
  std::set<std::shared_ptr<struct H2Conn>> conns_;
  // Every spawned thread (connection readers + stream handlers) stays
  // joinable (detached threads recycle TSan thread slots and produce
  // false reports) and is reaped: each marks a done flag on exit, spawn()
  // joins finished ones, stop() joins everything.
  struct Tracked {
    std::thread t;
    std::shared_ptr<std::atomic<bool>> done;
  };
  void spawn(std::function<void()> fn);
  void reap_locked();
  std::mutex threads_mu_;
  std::vector<Tracked> threads_;
};

struct UnaryCallResult {
  bool transport_ok = false;   // false: connect/protocol failure
  int grpc_status = -1;        // from trailers
  std::string grpc_message;
  std::string response;        // decoded message payload
  std::string error;           // transport error detail
};

// Blocking unary gRPC call over a unix socket (h2c). Used for
// Registration.Register and by tests.
UnaryCallResult grpc_unary_call(const std::string& unix_path,
                                const std::string& method_path,
                                const std::string& request,
                                int timeout_ms = 5000);

}  // namespace k3samd
