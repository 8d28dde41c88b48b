#include "metrics.h"

#include <arpa/inet.h>
#include <chrono>
#include <cstring>
#include <netinet/in.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

namespace k3samd {

MetricsServer::~MetricsServer() { stop(); }

bool MetricsServer::start(const std::string& addr, RenderFn render) {
  render_ = std::move(render);
  int fd = -1;
  if (addr.rfind("unix:", 0) == 0) {
    std::string path = addr.substr(5);
    ::unlink(path.c_str());
    fd = ::socket(AF_UNIX, SOCK_STREAM | SOCK_CLOEXEC, 0);
    if (fd < 0) return false;
    sockaddr_un sa{};
    sa.sun_family = AF_UNIX;
    if (path.size() >= sizeof(sa.sun_path)) {
      ::close(fd);
      return false;
    }
    std::strcpy(sa.sun_path, path.c_str());
    if (::bind(fd, (sockaddr*)&sa, sizeof(sa)) != 0 || ::listen(fd, 8) != 0) {
      ::close(fd);
      return false;
    }
  } else {
    size_t colon = addr.find_last_of(':');
    if (colon == std::string::npos) return false;
    std::string host = addr.substr(0, colon);
    int port = std::atoi(addr.c_str() + colon + 1);
    fd = ::socket(AF_INET, SOCK_STREAM | SOCK_CLOEXEC, 0);
    if (fd < 0) return false;
    int one = 1;
    ::setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in sa{};
    sa.sin_family = AF_INET;
    sa.sin_port = htons((uint16_t)port);
    if (::inet_pton(AF_INET, host.c_str(), &sa.sin_addr) != 1) {
      ::close(fd);
      return false;
    }
    if (::bind(fd, (sockaddr*)&sa, sizeof(sa)) != 0 || ::listen(fd, 8) != 0) {
      ::close(fd);
      return false;
    }
  }
  listen_fd_.store(fd);
  stopping_.store(false);
  thread_ = std::thread([this] { serve_loop(); });
  return true;
}

void MetricsServer::stop() {
  stopping_.store(true);
  int fd = listen_fd_.exchange(-1);
  if (fd >= 0) {
    ::shutdown(fd, SHUT_RDWR);
    ::close(fd);
  }
  if (thread_.joinable()) thread_.join();
}

void MetricsServer::serve_loop() {
  while (!stopping_.load()) {
    int lfd = listen_fd_.load();
    if (lfd < 0) break;
    int cfd = ::accept(lfd, nullptr, nullptr);
    if (cfd < 0) break;
    // A client that connects and sends nothing must not wedge the accept
    // loop (the DaemonSet's livenessProbe hits this endpoint): bound both
    // directions with short socket timeouts and serve each connection on
    // its own thread so one slow peer never blocks the next probe.
    timeval tv{2, 0};
    ::setsockopt(cfd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
    ::setsockopt(cfd, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof(tv));
    // bound concurrent connection threads: a flood (the endpoint may be
    // bound on the pod IP for ServiceMonitor scraping) must not grow
    // threads without limit — excess connections are dropped, and the
    // 2 s timeouts guarantee the pool drains
    if (active_.load() >= 64) {
      ::close(cfd);
      continue;
    }
    active_.fetch_add(1);
    std::thread([this, cfd] {
      char buf[1024];
      ssize_t ignored = ::read(cfd, buf, sizeof(buf));
      (void)ignored;
      std::string body;
      if (!stopping_.load() && render_) body = render_();
      char head[160];
      int n = std::snprintf(head, sizeof(head),
                            "HTTP/1.0 200 OK\r\n"
                            "Content-Type: text/plain; version=0.0.4\r\n"
                            "Content-Length: %zu\r\n\r\n",
                            body.size());
      (void)::send(cfd, head, (size_t)n, MSG_NOSIGNAL);
      (void)::send(cfd, body.data(), body.size(), MSG_NOSIGNAL);
      ::close(cfd);
      active_.fetch_sub(1);
    }).detach();
  }
  // connection threads are bounded by the 2 s socket timeouts; wait for
  // them so render_'s captures stay valid through stop()
  for (int i = 0; i < 500 && active_.load() > 0; ++i)
    std::this_thread::sleep_for(std::chrono::milliseconds(10));
}

}  // namespace k3samd
