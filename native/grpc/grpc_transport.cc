#include "grpc_transport.h"

#include <cerrno>
#include <chrono>
#include <csignal>
#include <cstdio>
#include <cstring>
#include <condition_variable>
#include <poll.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>
#include <vector>

#include "hpack.h"

namespace k3samd {

namespace {

constexpr char kPreface[] = "PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n";
constexpr size_t kPrefaceLen = 24;

enum FrameType : uint8_t {
  kData = 0x0,
  kHeaders = 0x1,
  kPriority = 0x2,
  kRstStream = 0x3,
  kSettings = 0x4,
  kPushPromise = 0x5,
  kPing = 0x6,
  kGoAway = 0x7,
  kWindowUpdate = 0x8,
  kContinuation = 0x9,
};

enum Flags : uint8_t {
  kEndStream = 0x1,
  kAck = 0x1,
  kEndHeaders = 0x4,
  kPadded = 0x8,
  kPriorityFlag = 0x20,
};

constexpr int kDefaultWindow = 65535;
constexpr size_t kMaxFrame = 16384;

struct FrameHeader {
  uint32_t length;
  uint8_t type;
  uint8_t flags;
  uint32_t stream;
};

bool read_full(int fd, void* buf, size_t n, int timeout_ms = -1) {
  uint8_t* p = (uint8_t*)buf;
  while (n > 0) {
    if (timeout_ms >= 0) {
      struct pollfd pfd{fd, POLLIN, 0};
      int pr = ::poll(&pfd, 1, timeout_ms);
      if (pr <= 0) return false;
    }
    ssize_t r = ::read(fd, p, n);
    if (r <= 0) return false;
    p += r;
    n -= (size_t)r;
  }
  return true;
}

bool write_full(int fd, const void* buf, size_t n) {
  const uint8_t* p = (const uint8_t*)buf;
  while (n > 0) {
    // MSG_NOSIGNAL: a peer that closed mid-write must surface as EPIPE,
    // not a process-killing SIGPIPE (churned gRPC channels do this)
    ssize_t r = ::send(fd, p, n, MSG_NOSIGNAL);
    if (r <= 0) return false;
    p += r;
    n -= (size_t)r;
  }
  return true;
}

std::string frame_bytes(uint8_t type, uint8_t flags, uint32_t stream,
                        std::string_view payload) {
  std::string out;
  out.reserve(9 + payload.size());
  uint32_t len = (uint32_t)payload.size();
  out.push_back((char)((len >> 16) & 0xff));
  out.push_back((char)((len >> 8) & 0xff));
  out.push_back((char)(len & 0xff));
  out.push_back((char)type);
  out.push_back((char)flags);
  out.push_back((char)((stream >> 24) & 0x7f));
  out.push_back((char)((stream >> 16) & 0xff));
  out.push_back((char)((stream >> 8) & 0xff));
  out.push_back((char)(stream & 0xff));
  out.append(payload.data(), payload.size());
  return out;
}

std::string grpc_frame(std::string_view message) {
  std::string out;
  out.reserve(5 + message.size());
  out.push_back(0);  // not compressed
  uint32_t len = (uint32_t)message.size();
  out.push_back((char)((len >> 24) & 0xff));
  out.push_back((char)((len >> 16) & 0xff));
  out.push_back((char)((len >> 8) & 0xff));
  out.push_back((char)(len & 0xff));
  out.append(message.data(), message.size());
  return out;
}

// Parse gRPC length-prefixed messages out of a data buffer; returns the
// first complete message (our RPCs all take exactly one request message).
bool first_grpc_message(const std::string& buf, std::string& msg) {
  if (buf.size() < 5) return false;
  uint32_t len = ((uint8_t)buf[1] << 24) | ((uint8_t)buf[2] << 16) |
                 ((uint8_t)buf[3] << 8) | (uint8_t)buf[4];
  if (buf.size() < 5 + (size_t)len) return false;
  msg = buf.substr(5, len);
  return true;
}

}  // namespace

// ---------------------------------------------------------------------------
// Connection state
// ---------------------------------------------------------------------------

struct H2Stream {
  uint32_t id = 0;
  std::vector<Header> headers;
  std::string data;
  bool request_complete = false;
  std::atomic<bool> cancelled{false};
  std::atomic<int> send_window{kDefaultWindow};
  bool dispatched = false;
};

struct H2Conn : std::enable_shared_from_this<H2Conn> {
  int fd = -1;
  std::mutex write_mu;
  std::atomic<bool> closed{false};
  HpackDecoder decoder;
  std::mutex mu;  // guards streams map + windows
  std::condition_variable window_cv;
  std::map<uint32_t, std::shared_ptr<H2Stream>> streams;
  std::atomic<int> send_window{kDefaultWindow};
  int peer_initial_window = kDefaultWindow;

  ~H2Conn() {
    close_fd();
    if (fd >= 0) {
      ::close(fd);
      fd = -1;
    }
  }

  void close_fd() {
    bool was = closed.exchange(true);
    if (!was && fd >= 0) {
      ::shutdown(fd, SHUT_RDWR);
    }
    window_cv.notify_all();
  }

  bool send(std::string_view bytes) {
    std::lock_guard<std::mutex> lk(write_mu);
    if (closed.load()) return false;
    if (!write_full(fd, bytes.data(), bytes.size())) {
      closed.store(true);
      window_cv.notify_all();
      return false;
    }
    return true;
  }

  // Send a DATA frame respecting connection + stream flow-control windows.
  // Splits the payload to whatever window credit is currently available
  // (a peer window smaller than the payload must produce several DATA
  // frames, not a wait for credit that can never arrive).
  bool send_data(const std::shared_ptr<H2Stream>& st, std::string_view payload,
                 bool end_stream) {
    if (payload.empty())
      return send(frame_bytes(kData, end_stream ? kEndStream : 0, st->id,
                              payload));
    size_t off = 0;
    while (off < payload.size()) {
      size_t chunk;
      {
        std::unique_lock<std::mutex> lk(mu);
        bool ok = window_cv.wait_for(lk, std::chrono::seconds(60), [&] {
          return closed.load() || st->cancelled.load() ||
                 (send_window.load() > 0 && st->send_window.load() > 0);
        });
        if (!ok || closed.load() || st->cancelled.load()) return false;
        chunk = std::min(
            {payload.size() - off, kMaxFrame, (size_t)send_window.load(),
             (size_t)st->send_window.load()});
        send_window.fetch_sub((int)chunk);
        st->send_window.fetch_sub((int)chunk);
      }
      bool last = (off + chunk >= payload.size());
      if (!send(frame_bytes(kData, (end_stream && last) ? kEndStream : 0,
                            st->id, payload.substr(off, chunk))))
        return false;
      off += chunk;
    }
    return true;
  }

  void on_window_update(uint32_t stream, uint32_t inc) {
    std::lock_guard<std::mutex> lk(mu);
    if (stream == 0) {
      send_window.fetch_add((int)inc);
    } else {
      auto it = streams.find(stream);
      if (it != streams.end()) it->second->send_window.fetch_add((int)inc);
    }
    window_cv.notify_all();
  }
};

// ---------------------------------------------------------------------------
// Server
// ---------------------------------------------------------------------------

GrpcServer::~GrpcServer() { stop(); }

void GrpcServer::add_unary(const std::string& path, UnaryHandler h) {
  unary_[path] = std::move(h);
}

void GrpcServer::add_server_stream(const std::string& path, StreamHandler h) {
  stream_[path] = std::move(h);
}

bool GrpcServer::start(const std::string& unix_path) {
  // belt-and-braces with MSG_NOSIGNAL: never die on peer-closed sockets
  std::signal(SIGPIPE, SIG_IGN);
  ::unlink(unix_path.c_str());
  int fd = ::socket(AF_UNIX, SOCK_STREAM | SOCK_CLOEXEC, 0);
  if (fd < 0) return false;
  sockaddr_un addr{};
  addr.sun_family = AF_UNIX;
  if (unix_path.size() >= sizeof(addr.sun_path)) return false;
  std::strcpy(addr.sun_path, unix_path.c_str());
  if (::bind(fd, (sockaddr*)&addr, sizeof(addr)) != 0) {
    ::close(fd);
    return false;
  }
  if (::listen(fd, 8) != 0) {
    ::close(fd);
    return false;
  }
  listen_fd_.store(fd);
  bound_path_ = unix_path;
  running_.store(true);
  stopping_.store(false);
  accept_thread_ = std::thread([this] { serve_loop(); });
  return true;
}

void GrpcServer::stop() {
  int fd = listen_fd_.exchange(-1);
  if (!running_.load() && fd < 0) return;
  stopping_.store(true);
  if (fd >= 0) {
    ::shutdown(fd, SHUT_RDWR);
    ::close(fd);
    if (!bound_path_.empty()) ::unlink(bound_path_.c_str());
  }
  {
    std::lock_guard<std::mutex> lk(conns_mu_);
    for (auto& c : conns_) c->close_fd();
  }
  if (accept_thread_.joinable()) accept_thread_.join();
  // join every connection/stream thread (their fds are shut down above,
  // so they exit promptly; new stream threads can only be spawned by
  // connection threads this loop is joining)
  for (;;) {
    Tracked tr;
    {
      std::lock_guard<std::mutex> lk(threads_mu_);
      if (threads_.empty()) break;
      tr = std::move(threads_.back());
      threads_.pop_back();
    }
    if (tr.t.joinable()) tr.t.join();
  }
  running_.store(false);
}

void GrpcServer::reap_locked() {
  for (size_t i = 0; i < threads_.size();) {
    if (threads_[i].done->load()) {
      if (threads_[i].t.joinable()) threads_[i].t.join();  // finishes now
      threads_[i] = std::move(threads_.back());
      threads_.pop_back();
    } else {
      ++i;
    }
  }
}

void GrpcServer::spawn(std::function<void()> fn) {
  auto done = std::make_shared<std::atomic<bool>>(false);
  std::thread t([fn = std::move(fn), done] {
    fn();
    done->store(true);
  });
  std::lock_guard<std::mutex> lk(threads_mu_);
  reap_locked();
  threads_.push_back({std::move(t), std::move(done)});
}

namespace {

void send_trailers(const std::shared_ptr<H2Conn>& conn,
                   const std::shared_ptr<H2Stream>& st, const GrpcStatus& gs,
                   bool with_response_headers) {
  std::string block;
  if (with_response_headers) {
    // trailers-only response: include :status + content-type
    HpackEncoder::encode({{":status", "200"},
                          {"content-type", "application/grpc"}},
                         block);
  }
  HpackEncoder::encode({{"grpc-status", std::to_string(gs.code)},
                        {"grpc-message", gs.message}},
                       block);
  conn->send(frame_bytes(kHeaders, kEndHeaders | kEndStream, st->id, block));
}

void send_response_headers(const std::shared_ptr<H2Conn>& conn,
                           const std::shared_ptr<H2Stream>& st) {
  std::string block;
  HpackEncoder::encode(
      {{":status", "200"}, {"content-type", "application/grpc"}}, block);
  conn->send(frame_bytes(kHeaders, kEndHeaders, st->id, block));
}

std::string find_header(const std::vector<Header>& hs, const std::string& k) {
  for (auto& [n, v] : hs)
    if (n == k) return v;
  return {};
}

}  // namespace

void GrpcServer::serve_loop() {
  while (!stopping_.load()) {
    int lfd = listen_fd_.load();
    if (lfd < 0) break;
    int cfd = ::accept(lfd, nullptr, nullptr);
    if (cfd < 0) {
      if (errno == EINTR || errno == ECONNABORTED) continue;
      if (!stopping_.load())
        std::fprintf(stderr, "grpc-server: accept failed: %s\n",
                     std::strerror(errno));
      break;
    }
    auto conn = std::make_shared<H2Conn>();
    conn->fd = cfd;
    {
      std::lock_guard<std::mutex> lk(conns_mu_);
      // kubelet holds ONE connection for the plugin's lifetime; a cap
      // far above that bounds thread growth if something floods the
      // socket (each connection owns a reader thread)
      if (conns_.size() >= 32) {
        ::close(cfd);
        continue;
      }
      conns_.insert(conn);
    }
    spawn([this, conn] { connection_loop(conn); });
  }
}

// One HTTP/2 connection, start to finish: preface handshake, frame loop,
// RPC dispatch. Runs on its own (tracked) thread; returns when the peer
// hangs up, errors, or the server stops.
void GrpcServer::connection_loop(const std::shared_ptr<H2Conn>& conn) {
  {  // scope: everything before the common close-and-deregister epilogue
      // --- connection handshake ---
      char preface[kPrefaceLen];
      if (!read_full(conn->fd, preface, kPrefaceLen, 10000) ||
          std::memcmp(preface, kPreface, kPrefaceLen) != 0) {
        if (std::getenv("K3SAMD_GRPC_DEBUG"))
          std::fprintf(stderr, "grpc-server: conn fd=%d preface fail errno=%s\n",
                       conn->fd, std::strerror(errno));
        conn->close_fd();
      } else {
        conn->send(frame_bytes(kSettings, 0, 0, {}));  // our (default) settings
        // --- frame loop ---
        std::string header_accum;
        uint32_t header_stream = 0;
        uint8_t header_flags = 0;
        bool in_headers = false;
        for (;;) {
          uint8_t fh[9];
          if (!read_full(conn->fd, fh, 9)) break;
          FrameHeader f;
          f.length = ((uint32_t)fh[0] << 16) | ((uint32_t)fh[1] << 8) | fh[2];
          f.type = fh[3];
          f.flags = fh[4];
          f.stream = ((uint32_t)(fh[5] & 0x7f) << 24) | ((uint32_t)fh[6] << 16) |
                     ((uint32_t)fh[7] << 8) | fh[8];
          if (f.length > 1u << 24) break;  // sanity
          std::string payload(f.length, '\0');
          if (f.length && !read_full(conn->fd, payload.data(), f.length)) break;

          if (in_headers && f.type != kContinuation) break;  // protocol error

          switch (f.type) {
            case kSettings: {
              if (f.flags & kAck) break;
              for (size_t i = 0; i + 6 <= payload.size(); i += 6) {
                uint16_t id = ((uint8_t)payload[i] << 8) | (uint8_t)payload[i + 1];
                uint32_t val = ((uint8_t)payload[i + 2] << 24) |
                               ((uint8_t)payload[i + 3] << 16) |
                               ((uint8_t)payload[i + 4] << 8) |
                               (uint8_t)payload[i + 5];
                if (id == 4) {  // INITIAL_WINDOW_SIZE
                  std::lock_guard<std::mutex> lk(conn->mu);
                  int delta = (int)val - conn->peer_initial_window;
                  conn->peer_initial_window = (int)val;
                  for (auto& [sid, st] : conn->streams)
                    st->send_window.fetch_add(delta);
                  conn->window_cv.notify_all();
                }
              }
              conn->send(frame_bytes(kSettings, kAck, 0, {}));
              break;
            }
            case kPing: {
              if (!(f.flags & kAck))
                conn->send(frame_bytes(kPing, kAck, 0, payload));
              break;
            }
            case kWindowUpdate: {
              if (payload.size() == 4) {
                uint32_t inc = ((uint32_t)(uint8_t)payload[0] << 24) |
                               ((uint32_t)(uint8_t)payload[1] << 16) |
                               ((uint32_t)(uint8_t)payload[2] << 8) |
                               (uint8_t)payload[3];
                conn->on_window_update(f.stream, inc & 0x7fffffff);
              }
              break;
            }
            case kHeaders: {
              std::string_view frag(payload);
              if (f.flags & kPadded) {
                if (frag.empty()) break;
                uint8_t pad = (uint8_t)frag[0];
                frag.remove_prefix(1);
                if (pad <= frag.size()) frag.remove_suffix(pad);
              }
              if (f.flags & kPriorityFlag) {
                if (frag.size() < 5) break;
                frag.remove_prefix(5);
              }
              header_accum.assign(frag);
              header_stream = f.stream;
              header_flags = f.flags;
              in_headers = !(f.flags & kEndHeaders);
              break;
            }
            case kContinuation: {
              header_accum.append(payload);
              if (f.flags & kEndHeaders) in_headers = false;
              break;
            }
            case kData: {
              std::string_view frag(payload);
              if (f.flags & kPadded) {
                if (frag.empty()) break;
                uint8_t pad = (uint8_t)frag[0];
                frag.remove_prefix(1);
                if (pad <= frag.size()) frag.remove_suffix(pad);
              }
              std::shared_ptr<H2Stream> st;
              {
                std::lock_guard<std::mutex> lk(conn->mu);
                auto it = conn->streams.find(f.stream);
                if (it != conn->streams.end()) st = it->second;
              }
              if (st) {
                // bound request buffering (our RPCs are tiny; a peer must
                // not be able to balloon server memory)
                constexpr size_t kMaxRequestBytes = 16u << 20;
                if (st->data.size() + frag.size() > kMaxRequestBytes) {
                  st->cancelled.store(true);
                } else {
                  st->data.append(frag);
                }
                if (f.flags & kEndStream) st->request_complete = true;
              }
              // replenish peer's send window (conn + stream)
              if (!payload.empty()) {
                std::string wu;
                uint32_t inc = (uint32_t)payload.size();
                char w[4] = {(char)((inc >> 24) & 0x7f), (char)((inc >> 16) & 0xff),
                             (char)((inc >> 8) & 0xff), (char)(inc & 0xff)};
                wu += frame_bytes(kWindowUpdate, 0, 0, std::string_view(w, 4));
                wu += frame_bytes(kWindowUpdate, 0, f.stream,
                                  std::string_view(w, 4));
                conn->send(wu);
              }
              break;
            }
            case kRstStream: {
              std::lock_guard<std::mutex> lk(conn->mu);
              auto it = conn->streams.find(f.stream);
              if (it != conn->streams.end()) {
                it->second->cancelled.store(true);
                conn->window_cv.notify_all();
              }
              break;
            }
            case kGoAway:
              goto conn_done;
            default:
              break;  // PRIORITY, PUSH_PROMISE (ignored)
          }

          // finished a header block -> materialize the stream
          if (!in_headers && !header_accum.empty() && header_stream != 0) {
            auto st = std::make_shared<H2Stream>();
            st->id = header_stream;
            if (!conn->decoder.decode(header_accum, st->headers)) break;
            st->send_window.store(conn->peer_initial_window);
            if (header_flags & kEndStream) st->request_complete = true;
            {
              std::lock_guard<std::mutex> lk(conn->mu);
              conn->streams[st->id] = st;
            }
            header_accum.clear();
            header_stream = 0;
          }

          // dispatch any streams whose request is complete
          std::vector<std::shared_ptr<H2Stream>> ready;
          {
            std::lock_guard<std::mutex> lk(conn->mu);
            for (auto& [sid, st] : conn->streams) {
              if (st->request_complete && !st->dispatched) {
                st->dispatched = true;
                ready.push_back(st);
              }
            }
          }
          auto finish_stream = [conn](const std::shared_ptr<H2Stream>& st) {
            // a finished RPC's state is dropped from the map (kubelet keeps
            // one connection for the plugin's lifetime; per-RPC streams
            // must not accumulate)
            std::lock_guard<std::mutex> lk(conn->mu);
            conn->streams.erase(st->id);
          };
          for (auto& st : ready) {
            std::string path = find_header(st->headers, ":path");
            if (st->cancelled.load()) {  // e.g. over-size request
              send_trailers(conn, st, {8, "request too large"}, true);
              finish_stream(st);
              continue;
            }
            std::string req;
            first_grpc_message(st->data, req);
            auto uit = unary_.find(path);
            if (uit != unary_.end()) {
              // Run on a worker thread, NOT this frame-reader thread: a
              // response larger than the peer's flow-control window makes
              // send_data block for WINDOW_UPDATE frames that only the
              // frame-reader can deliver (reachable with big
              // Allocate/GetPreferredAllocation payloads at high
              // time-slicing replica counts).
              UnaryHandler handler = uit->second;
              spawn([conn, st, handler, finish_stream, req] {
                std::string resp;
                GrpcStatus gs = handler(req, resp);
                if (st->cancelled.load() || conn->closed.load()) {
                  finish_stream(st);
                  return;
                }
                if (gs.code == 0) {
                  send_response_headers(conn, st);
                  conn->send_data(st, grpc_frame(resp), false);
                  send_trailers(conn, st, gs, false);
                } else {
                  send_trailers(conn, st, gs, true);
                }
                finish_stream(st);
              });
              continue;
            }
            auto sit = stream_.find(path);
            if (sit != stream_.end()) {
              // cap concurrent server-streams per connection (each owns a
              // thread; kubelet uses exactly one ListAndWatch)
              size_t live;
              {
                std::lock_guard<std::mutex> lk(conn->mu);
                live = conn->streams.size();
              }
              if (live > 64) {
                send_trailers(conn, st, {8, "too many concurrent streams"},
                              true);
                finish_stream(st);
                continue;
              }
              StreamHandler handler = sit->second;
              spawn([conn, st, handler, finish_stream, req] {
                send_response_headers(conn, st);
                auto write = [conn, st](const std::string& msg) -> bool {
                  if (conn->closed.load() || st->cancelled.load()) return false;
                  return conn->send_data(st, grpc_frame(msg), false);
                };
                GrpcStatus gs = handler(req, write);
                if (!conn->closed.load() && !st->cancelled.load())
                  send_trailers(conn, st, gs, false);
                finish_stream(st);
              });
              continue;
            }
            send_trailers(conn, st, {12, "unknown method " + path}, true);
            finish_stream(st);
          }
        }
      }
  }
conn_done:
  conn->close_fd();
  {
    std::lock_guard<std::mutex> lk(conns_mu_);
    conns_.erase(conn);
  }
}

// ---------------------------------------------------------------------------
// Client (blocking unary)
// ---------------------------------------------------------------------------

UnaryCallResult grpc_unary_call(const std::string& unix_path,
                                const std::string& method_path,
                                const std::string& request, int timeout_ms) {
  UnaryCallResult res;
  int fd = ::socket(AF_UNIX, SOCK_STREAM | SOCK_CLOEXEC, 0);
  if (fd < 0) {
    res.error = "socket() failed";
    return res;
  }
  sockaddr_un addr{};
  addr.sun_family = AF_UNIX;
  if (unix_path.size() >= sizeof(addr.sun_path)) {
    res.error = "path too long";
    ::close(fd);
    return res;
  }
  std::strcpy(addr.sun_path, unix_path.c_str());
  if (::connect(fd, (sockaddr*)&addr, sizeof(addr)) != 0) {
    res.error = "connect failed: " + unix_path;
    ::close(fd);
    return res;
  }

  std::string out(kPreface, kPrefaceLen);
  out += frame_bytes(kSettings, 0, 0, {});
  std::string block;
  HpackEncoder::encode({{":method", "POST"},
                        {":scheme", "http"},
                        {":path", method_path},
                        {":authority", "localhost"},
                        {"content-type", "application/grpc"},
                        {"te", "trailers"}},
                       block);
  out += frame_bytes(kHeaders, kEndHeaders, 1, block);
  out += frame_bytes(kData, kEndStream, 1, grpc_frame(request));
  if (!write_full(fd, out.data(), out.size())) {
    res.error = "write failed";
    ::close(fd);
    return res;
  }

  HpackDecoder decoder;
  std::string data;
  std::vector<Header> resp_headers;
  std::string header_accum;
  bool in_headers = false;
  uint8_t saved_flags = 0;
  bool done = false;
  while (!done) {
    uint8_t fh[9];
    if (!read_full(fd, fh, 9, timeout_ms)) {
      res.error = "timeout/eof waiting for response";
      ::close(fd);
      return res;
    }
    uint32_t length = ((uint32_t)fh[0] << 16) | ((uint32_t)fh[1] << 8) | fh[2];
    uint8_t type = fh[3], flags = fh[4];
    uint32_t stream = ((uint32_t)(fh[5] & 0x7f) << 24) |
                      ((uint32_t)fh[6] << 16) | ((uint32_t)fh[7] << 8) | fh[8];
    std::string payload(length, '\0');
    if (length && !read_full(fd, payload.data(), length, timeout_ms)) {
      res.error = "short frame";
      ::close(fd);
      return res;
    }
    switch (type) {
      case kSettings:
        if (!(flags & kAck)) {
          auto ack = frame_bytes(kSettings, kAck, 0, {});
          write_full(fd, ack.data(), ack.size());
        }
        break;
      case kPing:
        if (!(flags & kAck)) {
          auto pong = frame_bytes(kPing, kAck, 0, payload);
          write_full(fd, pong.data(), pong.size());
        }
        break;
      case kHeaders: {
        std::string_view frag(payload);
        if (flags & kPadded) {
          if (frag.empty()) break;
          uint8_t pad = (uint8_t)frag[0];
          frag.remove_prefix(1);
          if (pad <= frag.size()) frag.remove_suffix(pad);
        }
        if (flags & kPriorityFlag) {
          if (frag.size() < 5) break;
          frag.remove_prefix(5);
        }
        header_accum.assign(frag);
        saved_flags = flags;
        in_headers = !(flags & kEndHeaders);
        break;
      }
      case kContinuation:
        header_accum.append(payload);
        if (flags & kEndHeaders) in_headers = false;
        break;
      case kData:
        if (stream == 1) data.append(payload);
        break;
      case kRstStream:
      case kGoAway:
        res.error = "stream reset by peer";
        ::close(fd);
        return res;
      default:
        break;
    }
    if (!in_headers && !header_accum.empty()) {
      std::vector<Header> hs;
      if (!decoder.decode(header_accum, hs)) {
        res.error = "hpack decode failed";
        ::close(fd);
        return res;
      }
      header_accum.clear();
      for (auto& h : hs) resp_headers.push_back(h);
      std::string gs = find_header(hs, "grpc-status");
      if (!gs.empty() || (saved_flags & kEndStream)) {
        res.grpc_status = gs.empty() ? 2 : std::atoi(gs.c_str());
        res.grpc_message = find_header(hs, "grpc-message");
        done = true;
      }
    }
  }
  ::close(fd);
  res.transport_ok = true;
  if (!data.empty()) first_grpc_message(data, res.response);
  return res;
}

}  // namespace k3samd
