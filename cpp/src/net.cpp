#include "hypha/net.h"

#include <arpa/inet.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <poll.h>
#include <string.h>
#include <sys/socket.h>
#include <sys/types.h>
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <condition_variable>
#include <stdexcept>

namespace hypha {

// ---- MsgSocket -------------------------------------------------------------

MsgSocket::~MsgSocket() { close_now(); }

void MsgSocket::close_now() {
  if (ssl_) {
    {
      // graceful close_notify first (if no other thread is mid-io), so the
      // peer sees an orderly TLS shutdown instead of an unexpected EOF
      std::unique_lock<std::mutex> lk(io_mu_, std::try_to_lock);
      if (lk.owns_lock() && ssl_) SSL_shutdown(ssl_);
    }
    if (fd_ >= 0) ::shutdown(fd_, SHUT_RDWR);  // wake any blocked SSL io
    std::lock_guard<std::mutex> lk(io_mu_);
    if (ssl_) SSL_free(ssl_);
    ssl_ = nullptr;
  }
  if (fd_ >= 0) ::shutdown(fd_, SHUT_RDWR);
  if (fd_ >= 0) {
    ::close(fd_);
    fd_ = -1;
  }
}

void MsgSocket::shutdown_now() {
  if (fd_ >= 0) ::shutdown(fd_, SHUT_RDWR);
}

static bool write_all_fd(int fd, const char* p, size_t n) {
  while (n > 0) {
    ssize_t w = ::send(fd, p, n, MSG_NOSIGNAL);
    if (w <= 0) return false;
    p += w;
    n -= w;
  }
  return true;
}

static bool read_all_fd(int fd, char* p, size_t n) {
  while (n > 0) {
    ssize_t r = ::recv(fd, p, n, 0);
    if (r <= 0) return false;
    p += r;
    n -= r;
  }
  return true;
}

static bool write_all_ssl(SSL* ssl, const char* p, size_t n) {
  while (n > 0) {
    int w = SSL_write(ssl, p, (int)n);
    if (w <= 0) return false;
    p += w;
    n -= w;
  }
  return true;
}

static bool read_all_ssl(SSL* ssl, char* p, size_t n) {
  while (n > 0) {
    int r = SSL_read(ssl, p, (int)n);
    if (r <= 0) return false;
    p += r;
    n -= r;
  }
  return true;
}

// Transport byte accounting (reference telemetry/src/bandwidth.rs:33-60:
// every connection counts inbound/outbound bytes). Process-global, lock-free.
static std::atomic<unsigned long long> g_bytes_in{0}, g_bytes_out{0};

BandwidthStats bandwidth_stats() {
  return {g_bytes_in.load(std::memory_order_relaxed),
          g_bytes_out.load(std::memory_order_relaxed)};
}

bool MsgSocket::ssl_wait_readable() {
  while (true) {
    {
      std::lock_guard<std::mutex> lk(io_mu_);
      if (!ssl_) return false;
      if (SSL_pending(ssl_) > 0) return true;  // buffered record bytes
    }
    struct pollfd p = {fd_, POLLIN, 0};
    int rc = ::poll(&p, 1, 200);
    if (fd_ < 0) return false;
    if (rc < 0) return false;
    if (rc > 0) return true;  // readable (or HUP/ERR: the read will report it)
  }
}

bool MsgSocket::send_json(const Json& j) {
  std::string s = j.dump();
  uint32_t len = htonl((uint32_t)s.size());
  bool ok;
  if (ssl_) {
    std::lock_guard<std::mutex> lk(io_mu_);
    if (!ssl_) return false;
    ok = write_all_ssl(ssl_, (const char*)&len, 4) &&
         write_all_ssl(ssl_, s.data(), s.size());
  } else {
    std::lock_guard<std::mutex> lk(write_mu_);
    ok = write_all_fd(fd_, (const char*)&len, 4) &&
         write_all_fd(fd_, s.data(), s.size());
  }
  if (ok) g_bytes_out.fetch_add(4 + s.size(), std::memory_order_relaxed);
  return ok;
}

std::optional<Json> MsgSocket::recv_json() {
  uint32_t len_be;
  std::string s;
  if (ssl_) {
    if (!ssl_wait_readable()) return std::nullopt;
    std::lock_guard<std::mutex> lk(io_mu_);
    if (!ssl_) return std::nullopt;
    if (!read_all_ssl(ssl_, (char*)&len_be, 4)) return std::nullopt;
    uint32_t len = ntohl(len_be);
    if (len > 256u * 1024u * 1024u) return std::nullopt;
    s.assign(len, '\0');
    if (!read_all_ssl(ssl_, s.data(), len)) return std::nullopt;
  } else {
    if (!read_all_fd(fd_, (char*)&len_be, 4)) return std::nullopt;
    uint32_t len = ntohl(len_be);
    if (len > 256u * 1024u * 1024u) return std::nullopt;
    s.assign(len, '\0');
    if (!read_all_fd(fd_, s.data(), len)) return std::nullopt;
  }
  g_bytes_in.fetch_add(4 + s.size(), std::memory_order_relaxed);
  try {
    return Json::parse(s);
  } catch (...) {
    return std::nullopt;
  }
}

bool MsgSocket::send_raw(const char* data, size_t n) {
  bool ok;
  if (ssl_) {
    std::lock_guard<std::mutex> lk(io_mu_);
    if (!ssl_) return false;
    ok = write_all_ssl(ssl_, data, n);
  } else {
    ok = write_all_fd(fd_, data, n);
  }
  if (ok) g_bytes_out.fetch_add(n, std::memory_order_relaxed);
  return ok;
}
bool MsgSocket::recv_raw(char* data, size_t n) {
  bool ok;
  if (ssl_) {
    if (!ssl_wait_readable()) return false;
    std::lock_guard<std::mutex> lk(io_mu_);
    if (!ssl_) return false;
    ok = read_all_ssl(ssl_, data, n);
  } else {
    ok = read_all_fd(fd_, data, n);
  }
  if (ok) g_bytes_in.fetch_add(n, std::memory_order_relaxed);
  return ok;
}

int tcp_connect(const std::string& host, int port, double timeout_s) {
  struct addrinfo hints = {}, *res = nullptr;
  hints.ai_family = AF_INET;
  hints.ai_socktype = SOCK_STREAM;
  std::string ps = std::to_string(port);
  if (getaddrinfo(host.c_str(), ps.c_str(), &hints, &res) != 0) return -1;
  int fd = ::socket(res->ai_family, res->ai_socktype, res->ai_protocol);
  if (fd < 0) {
    freeaddrinfo(res);
    return -1;
  }
  struct timeval tv;
  tv.tv_sec = (long)timeout_s;
  tv.tv_usec = (long)((timeout_s - tv.tv_sec) * 1e6);
  setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof tv);
  setsockopt(fd, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof tv);
  int one = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof one);
  if (::connect(fd, res->ai_addr, res->ai_addrlen) != 0) {
    ::close(fd);
    freeaddrinfo(res);
    return -1;
  }
  freeaddrinfo(res);
  return fd;
}

int tcp_listen(int port, const std::string& host) {
  int fd = ::socket(AF_INET, SOCK_STREAM, 0);
  if (fd < 0) return -1;
  int one = 1;
  setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof one);
  struct sockaddr_in addr = {};
  addr.sin_family = AF_INET;
  // Default loopback-only (reference listen.rs binds explicit addresses;
  // nothing is exposed off-host unless asked): "0.0.0.0" or a concrete
  // interface address opens the daemon to the LAN for multi-host clusters.
  if (inet_pton(AF_INET, host.c_str(), &addr.sin_addr) != 1) {
    ::close(fd);
    return -1;
  }
  addr.sin_port = htons((uint16_t)port);
  if (::bind(fd, (struct sockaddr*)&addr, sizeof addr) != 0 || ::listen(fd, 64) != 0) {
    ::close(fd);
    return -1;
  }
  return fd;
}

int listen_port(int listen_fd) {
  struct sockaddr_in addr = {};
  socklen_t len = sizeof addr;
  getsockname(listen_fd, (struct sockaddr*)&addr, &len);
  return ntohs(addr.sin_port);
}

// ---- Node ------------------------------------------------------------------

Node::Node(std::string name, std::string gateway_host, int gateway_port, TlsConfig tls)
    : name_(std::move(name)), gw_host_(std::move(gateway_host)), gw_port_(gateway_port) {
  gw_primary_host_ = gw_host_;
  gw_primary_port_ = gw_port_;
  if (tls.enabled()) tls_ = std::make_unique<TlsContext>(tls);
}

Node::~Node() { stop(); }

void Node::start(int port) {
  if (running_ || accept_thread_.joinable())
    throw std::runtime_error("node: already started");
  listen_fd_ = tcp_listen(port, listen_host_);
  if (listen_fd_ < 0) throw std::runtime_error("node: cannot listen");
  port_ = listen_port(listen_fd_);
  running_ = true;
  accept_thread_ = std::thread([this] { accept_loop(); });
  if (gw_port_ > 0) {
    if (!gateway_connect()) {
      // clean partial start so the object can be retried/destroyed safely
      // (assigning over a joinable std::thread is std::terminate)
      stop();
      throw std::runtime_error(
          "node: gateway registration failed (unreachable, TLS handshake "
          "rejected, or no ack)");
    }
    gw_thread_ = std::thread([this] { gateway_listen_loop(); });
  }
}

// Establish (or re-establish) the persistent broker connection:
// connect + optional TLS + register + synchronous ack.
int Node::dial_gateway(double timeout_s) {
  // try the active gateway, then every other candidate (primary first)
  auto [ah, ap] = gw_addr();
  int fd = tcp_connect(ah, ap, timeout_s);
  if (fd < 0 && !gw_list_.empty()) {
    std::vector<std::pair<std::string, int>> cands;
    cands.emplace_back(gw_primary_host_, gw_primary_port_);
    cands.insert(cands.end(), gw_list_.begin(), gw_list_.end());
    for (const auto& [h, p] : cands) {
      if (h == ah && p == ap) continue;
      fd = tcp_connect(h, p, 5.0);
      if (fd >= 0) {
        fprintf(stderr, "[net:%s] gateway failover -> %s:%d\n", name_.c_str(),
                h.c_str(), p);
        set_gw_addr(h, p);
        break;
      }
    }
  }
  return fd;
}

bool Node::gateway_connect() {
  int fd = dial_gateway(10.0);
  if (fd < 0) return false;
  // long receive timeout on the event connection
  struct timeval tv = {86400, 0};
  setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof tv);
  SSL* gssl = nullptr;
  if (tls_) {
    gssl = tls_->wrap(fd, false);
    if (!gssl) {
      ::close(fd);
      return false;
    }
  }
  auto sock = std::make_unique<MsgSocket>(fd, gssl);
  Json reg;
  reg["kind"] = "register";
  reg["peer"] = name_;
  reg["addr"] = (advertise_host_.empty() ? std::string("127.0.0.1")
                                         : advertise_host_) +
                ":" + std::to_string(port_);
  if (!sock->send_json(reg)) return false;
  // synchronous ack: the registry entry is visible before we proceed
  auto ack = sock->recv_json();
  if (!ack || ack->get_or("kind", Json("")).as_string() != "registered") return false;
  {
    std::lock_guard<std::mutex> lk(gw_mu_);
    gw_sock_ = std::move(sock);
    // replay topic subscriptions (the broker lost them with the connection)
    std::lock_guard<std::mutex> lk2(mu_);
    for (auto& [topic, _] : subs_) {
      Json sub;
      sub["kind"] = "subscribe";
      sub["topic"] = topic;
      gw_sock_->send_json(sub);
    }
  }
  if (reconnect_cb_) {
    try {
      reconnect_cb_();  // daemons re-announce their KV records
    } catch (const std::exception& e) {
      fprintf(stderr, "[net:%s] reconnect callback error: %s\n", name_.c_str(),
              e.what());
    }
  }
  return true;
}

void Node::stop() {
  if (!running_.exchange(false)) return;
  // Phase 1: WAKE every blocked thread with shutdown() only — the fds stay
  // open, so no thread can race a kernel fd-number reuse. Closing happens
  // in phase 2, after the threads are done with their sockets.
  if (listen_fd_ >= 0) ::shutdown(listen_fd_, SHUT_RDWR);
  {
    // gw_mu_ serializes against the reader thread swapping the socket in
    // gateway_connect() during a reconnection
    std::lock_guard<std::mutex> lk(gw_mu_);
    if (gw_sock_) gw_sock_->shutdown_now();
  }
  {
    std::lock_guard<std::mutex> lk(conn_mu_);
    for (int fd : conn_fds_) ::shutdown(fd, SHUT_RDWR);
  }
  if (accept_thread_.joinable()) accept_thread_.join();
  if (gw_thread_.joinable()) gw_thread_.join();
  {
    // UNBOUNDED drain of the detached conn handlers: every registered fd
    // was shutdown above, so each handler unblocks and exits. Returning
    // before they drain would let a straggler touch a destroyed Node.
    std::unique_lock<std::mutex> lk(conn_mu_);
    conn_cv_.wait(lk, [&] { return live_conns_ == 0; });
  }
  // Phase 2: no reader threads remain — closing is safe now.
  if (listen_fd_ >= 0) {
    ::close(listen_fd_);
    listen_fd_ = -1;
  }
  std::lock_guard<std::mutex> lk(gw_mu_);
  gw_sock_.reset();
}

void Node::on(const std::string& type, Handler h) {
  std::lock_guard<std::mutex> lk(mu_);
  handlers_[type] = std::move(h);
}

void Node::on_stream(const std::string& type, StreamHandler h) {
  std::lock_guard<std::mutex> lk(mu_);
  stream_handlers_[type] = std::move(h);
}

void Node::unregister_conn(int fd) {
  std::lock_guard<std::mutex> lk(conn_mu_);
  conn_fds_.erase(fd);  // removed while the fd is STILL open (no reuse race)
  --live_conns_;
  conn_cv_.notify_all();
}

void Node::accept_loop() {
  while (running_) {
    int fd = ::accept(listen_fd_, nullptr, nullptr);
    if (fd < 0) break;
    int one = 1;
    setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof one);
    {
      // register BEFORE the handler thread exists, so a concurrent stop()
      // always sees the fd; if stop() already swept, wake it ourselves
      std::lock_guard<std::mutex> lk(conn_mu_);
      conn_fds_.insert(fd);
      ++live_conns_;
      if (!running_) ::shutdown(fd, SHUT_RDWR);
    }
    std::thread([this, fd] {
      SSL* ssl = nullptr;
      if (tls_) {
        ssl = tls_->wrap(fd, true);
        if (!ssl) {
          unregister_conn(fd);
          ::close(fd);
          return;  // unauthenticated peer rejected at handshake
        }
      }
      {
        MsgSocket sock(fd, ssl);
        try {
          handle_conn(sock);
        } catch (const std::exception& e) {
          fprintf(stderr, "[net:%s] conn handler error: %s\n", name_.c_str(),
                  e.what());
        }
        unregister_conn(fd);
      }  // the socket closes HERE — after deregistration
    }).detach();
  }
}

void Node::handle_conn(MsgSocket& sock) {
  // With mTLS the only identity we trust is the verified certificate CN
  // (reference model: PeerId is derived from the cert key,
  // rfc/2025-05-30_mtls.md). A self-declared `from` that contradicts it is a
  // spoof attempt and the message is rejected; handlers always see the CN.
  const std::string verified = sock.peer_identity();
  while (running_) {
    auto msg = sock.recv_json();
    if (!msg) break;
    std::string kind = msg->get_or("kind", Json("")).as_string();
    std::string type = msg->get_or("type", Json("")).as_string();
    std::string from = msg->get_or("from", Json("")).as_string();
    if (!verified.empty()) {
      if (!from.empty() && from != verified) {
        Json resp;
        resp["kind"] = "response";
        resp["ok"] = false;
        resp["error"] = "from '" + from + "' does not match certificate CN '" +
                        verified + "'";
        sock.send_json(resp);
        break;
      }
      from = verified;
    }
    if (kind == "request") {
      Handler h;
      {
        std::lock_guard<std::mutex> lk(mu_);
        auto it = handlers_.find(type);
        if (it != handlers_.end()) h = it->second;
      }
      Json resp;
      resp["kind"] = "response";
      if (h) {
        try {
          resp["body"] = h(from, msg->get_or("body", Json(JsonObject{})));
          resp["ok"] = true;
        } catch (const std::exception& e) {
          resp["ok"] = false;
          resp["error"] = std::string(e.what());
        }
      } else {
        resp["ok"] = false;
        resp["error"] = "no handler for " + type;
      }
      if (!sock.send_json(resp)) break;
    } else if (kind == "stream") {
      StreamHandler h;
      {
        std::lock_guard<std::mutex> lk(mu_);
        auto it = stream_handlers_.find(type);
        if (it != stream_handlers_.end()) h = it->second;
      }
      if (h) h(from, msg->get_or("header", Json(JsonObject{})), sock);
      break;  // stream connections are single-use
    } else {
      break;
    }
  }
}

void Node::gateway_listen_loop() {
  while (running_) {
    auto msg = gw_sock_->recv_json();
    if (!msg) {
      // broker connection lost (gateway restart?): keep re-registering until
      // it is back — the reference's libp2p re-establishes gossipsub the
      // same way. Registration also refreshes our addr record.
      while (running_ && !gateway_connect()) {
        struct timespec ts = {0, 500 * 1000 * 1000};
        nanosleep(&ts, nullptr);
      }
      if (!running_) break;
      fprintf(stderr, "[net:%s] gateway reconnected\n", name_.c_str());
      continue;
    }
    try {
    if (msg->get_or("kind", Json("")).as_string() == "relay_offer") {
      long long circuit = msg->get_or("circuit", Json((int64_t)0)).as_int();
      {
        // count the relay thread BEFORE detaching: stop() joins THIS (gw)
        // thread before draining live_conns_, so the count is always
        // visible to it and the relay thread can never outlive the Node
        std::lock_guard<std::mutex> lk(conn_mu_);
        ++live_conns_;
      }
      std::thread([this, circuit] {
        try {
          relay_accept_run(circuit);
        } catch (const std::exception& e) {
          fprintf(stderr, "[net:%s] relay accept error: %s\n", name_.c_str(),
                  e.what());
        }
        std::lock_guard<std::mutex> lk(conn_mu_);
        --live_conns_;
        conn_cv_.notify_all();
      }).detach();
    } else if (msg->get_or("kind", Json("")).as_string() == "pub") {
      std::string topic = msg->at("topic").as_string();
      std::string from = msg->get_or("from", Json("")).as_string();
      std::function<void(const std::string&, const Json&)> cb;
      {
        std::lock_guard<std::mutex> lk(mu_);
        auto it = subs_.find(topic);
        if (it != subs_.end()) cb = it->second;
      }
      if (cb) cb(from, msg->get_or("data", Json(JsonObject{})));
    }
    } catch (const std::exception& e) {
      fprintf(stderr, "[net:%s] gateway event error: %s\n", name_.c_str(), e.what());
    }
  }
}

int Node::relay_dial(const std::string& peer, double timeout_s) {
  auto [gh, gp] = gw_addr();
  if (gp == 0) return -1;
  int fd = tcp_connect(gh, gp, timeout_s);
  if (fd < 0) return -1;
  struct timeval tv;
  tv.tv_sec = (long)(timeout_s + 15.0);
  tv.tv_usec = 0;
  setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof tv);
  MsgSocket hello(fd, nullptr);  // plain framing: first byte 0x00 tells the
                                 // gateway this is a relay leg, not TLS
  Json req;
  req["kind"] = "relay_connect";
  req["to"] = peer;
  req["from"] = name_;
  if (!hello.send_json(req)) return -1;
  auto ack = hello.recv_json();
  if (!ack || !ack->get_or("ok", Json(false)).as_bool()) {
    return -1;  // MsgSocket dtor closes fd
  }
  fprintf(stderr, "[net:%s] relayed circuit to %s via gateway\n", name_.c_str(),
          peer.c_str());
  return hello.release();  // same socket now carries the peer protocol
}

void Node::relay_accept_run(long long circuit) {
  // The surrounding thread is already counted in live_conns_ by its
  // spawner; this function only tracks the FD (insert/erase, no counting)
  // so every blocking phase — hello recv, TLS handshake, message loop —
  // is woken by stop()'s shutdown sweep. Erase always precedes close.
  auto [gh, gp] = gw_addr();
  int fd = tcp_connect(gh, gp, 10.0);
  if (fd < 0) return;
  {
    std::lock_guard<std::mutex> lk(conn_mu_);
    conn_fds_.insert(fd);
    if (!running_) ::shutdown(fd, SHUT_RDWR);
  }
  auto drop_fd = [this](int f) {
    std::lock_guard<std::mutex> lk(conn_mu_);
    conn_fds_.erase(f);
  };
  {
    MsgSocket hello(fd, nullptr);
    Json acc;
    acc["kind"] = "relay_accept";
    acc["circuit"] = (int64_t)circuit;
    acc["from"] = name_;
    if (!hello.send_json(acc)) {
      drop_fd(fd);
      return;  // hello socket closes after deregistration
    }
    auto ack = hello.recv_json();
    if (!ack || !ack->get_or("ok", Json(false)).as_bool()) {
      drop_fd(fd);
      return;
    }
    fd = hello.release();
  }
  // from here the circuit is an ordinary inbound connection: server-side
  // mTLS handshake + the normal typed-message loop
  SSL* ssl = nullptr;
  if (tls_) {
    ssl = tls_->wrap(fd, true);
    if (!ssl) {
      drop_fd(fd);
      ::close(fd);
      return;
    }
  }
  {
    MsgSocket sock(fd, ssl);
    try {
      handle_conn(sock);
    } catch (const std::exception& e) {
      fprintf(stderr, "[net:%s] relay conn handler error: %s\n", name_.c_str(),
              e.what());
    }
    drop_fd(fd);
  }
}

Json Node::gateway_request(const std::string& type, const Json& body) {
  int fd = dial_gateway(10.0);
  if (fd < 0) throw std::runtime_error("gateway unreachable");
  SSL* ssl = nullptr;
  if (tls_) {
    ssl = tls_->wrap(fd, false);
    if (!ssl) {
      ::close(fd);
      throw std::runtime_error("gateway TLS handshake failed");
    }
  }
  MsgSocket sock(fd, ssl);
  Json req;
  req["kind"] = "request";
  req["type"] = type;
  req["from"] = name_;
  req["body"] = body;
  if (!sock.send_json(req)) throw std::runtime_error("gateway send failed");
  auto resp = sock.recv_json();
  if (!resp) throw std::runtime_error("gateway closed");
  return resp->get_or("body", Json(JsonObject{}));
}

bool cidr_contains(const std::string& cidr, const std::string& host) {
  size_t slash = cidr.find('/');
  if (slash == std::string::npos) return false;
  struct in_addr net_a, host_a;
  if (inet_pton(AF_INET, cidr.substr(0, slash).c_str(), &net_a) != 1) return false;
  if (inet_pton(AF_INET, host.c_str(), &host_a) != 1) return false;
  int bits = std::stoi(cidr.substr(slash + 1));
  if (bits <= 0) return true;
  if (bits > 32) return false;
  uint32_t mask = bits == 32 ? 0xFFFFFFFFu : ~(0xFFFFFFFFu >> bits);
  return (ntohl(net_a.s_addr) & mask) == (ntohl(host_a.s_addr) & mask);
}

void Node::check_dialable(const std::string& peer, const std::string& host) const {
  for (const auto& c : exclude_cidrs_)
    if (cidr_contains(c, host))
      throw std::runtime_error("dial refused: peer " + peer + " at " + host +
                               " is in excluded CIDR " + c);
}

Json Node::request(const std::string& peer, const std::string& type, const Json& body,
                   double timeout_s) {
  auto addr = resolve(peer);
  if (!addr) throw std::runtime_error("unknown peer " + peer);
  size_t colon = addr->rfind(':');
  check_dialable(peer, addr->substr(0, colon));
  int fd = tcp_connect(addr->substr(0, colon), std::stoi(addr->substr(colon + 1)), timeout_s);
  if (fd < 0) {
    // stale cache: the peer may have restarted on a new port — re-resolve
    {
      std::lock_guard<std::mutex> lk(mu_);
      addr_cache_.erase(peer);
    }
    addr = resolve(peer);
    if (!addr) throw std::runtime_error("unknown peer " + peer);
    colon = addr->rfind(':');
    check_dialable(peer, addr->substr(0, colon));
    fd = tcp_connect(addr->substr(0, colon), std::stoi(addr->substr(colon + 1)), timeout_s);
  }
  if (fd < 0) fd = relay_dial(peer, timeout_s);  // gateway circuit fallback
  if (fd < 0) throw std::runtime_error("peer unreachable: " + peer);
  SSL* pssl = nullptr;
  if (tls_) {
    pssl = tls_->wrap(fd, false);
    if (!pssl) {
      ::close(fd);
      throw std::runtime_error("TLS handshake failed with " + peer);
    }
  }
  MsgSocket sock(fd, pssl);
  Json req;
  req["kind"] = "request";
  req["type"] = type;
  req["from"] = name_;
  req["body"] = body;
  if (!sock.send_json(req)) throw std::runtime_error("send failed");
  auto resp = sock.recv_json();
  if (!resp) throw std::runtime_error("peer closed: " + peer);
  if (!resp->get_or("ok", Json(false)).as_bool())
    throw std::runtime_error("rpc error from " + peer + ": " +
                             resp->get_or("error", Json("unknown")).as_string());
  return resp->get_or("body", Json(JsonObject{}));
}

std::unique_ptr<MsgSocket> Node::open_stream(const std::string& peer, const std::string& type,
                                             const Json& header) {
  auto addr = resolve(peer);
  if (!addr) throw std::runtime_error("unknown peer " + peer);
  size_t colon = addr->rfind(':');
  check_dialable(peer, addr->substr(0, colon));
  int fd = tcp_connect(addr->substr(0, colon), std::stoi(addr->substr(colon + 1)), 30.0);
  if (fd < 0) {  // stale cache: re-resolve once (peer may have restarted)
    {
      std::lock_guard<std::mutex> lk(mu_);
      addr_cache_.erase(peer);
    }
    addr = resolve(peer);
    if (!addr) throw std::runtime_error("unknown peer " + peer);
    colon = addr->rfind(':');
    check_dialable(peer, addr->substr(0, colon));
    fd = tcp_connect(addr->substr(0, colon), std::stoi(addr->substr(colon + 1)), 30.0);
  }
  if (fd < 0) fd = relay_dial(peer, 30.0);  // gateway circuit fallback
  if (fd < 0) throw std::runtime_error("peer unreachable: " + peer);
  struct timeval tv = {600, 0};
  setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof tv);
  setsockopt(fd, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof tv);
  SSL* sssl = nullptr;
  if (tls_) {
    sssl = tls_->wrap(fd, false);
    if (!sssl) {
      ::close(fd);
      throw std::runtime_error("stream TLS handshake failed with " + peer);
    }
  }
  auto sock = std::make_unique<MsgSocket>(fd, sssl);
  Json req;
  req["kind"] = "stream";
  req["type"] = type;
  req["from"] = name_;
  req["header"] = header;
  if (!sock->send_json(req)) throw std::runtime_error("stream open failed");
  return sock;
}

void Node::publish(const std::string& topic, const Json& data) {
  Json b;
  b["topic"] = topic;
  b["data"] = data;
  gateway_request("publish", b);
}

void Node::subscribe(const std::string& topic,
                     std::function<void(const std::string&, const Json&)> cb) {
  {
    std::lock_guard<std::mutex> lk(mu_);
    subs_[topic] = std::move(cb);
  }
  Json sub;
  sub["kind"] = "subscribe";
  sub["topic"] = topic;
  std::lock_guard<std::mutex> lk(gw_mu_);
  if (gw_sock_) gw_sock_->send_json(sub);  // no broker: recorded locally only
}

void Node::unsubscribe(const std::string& topic) {
  {
    std::lock_guard<std::mutex> lk(mu_);
    subs_.erase(topic);
  }
  Json un;
  un["kind"] = "unsubscribe";
  un["topic"] = topic;
  std::lock_guard<std::mutex> lk(gw_mu_);
  if (gw_sock_) gw_sock_->send_json(un);
}

void Node::kv_put(const std::string& key, const Json& value) {
  Json b;
  b["key"] = key;
  b["value"] = value;
  gateway_request("kv_put", b);
}

std::optional<Json> Node::kv_get(const std::string& key) {
  Json b;
  b["key"] = key;
  Json r = gateway_request("kv_get", b);
  if (r.get_or("found", Json(false)).as_bool()) return r.at("value");
  return std::nullopt;
}

std::optional<std::string> Node::resolve(const std::string& peer) {
  {
    std::lock_guard<std::mutex> lk(mu_);
    auto it = addr_cache_.find(peer);
    if (it != addr_cache_.end()) return it->second;
  }
  // brief retry: peers may register concurrently with the first lookup
  for (int attempt = 0; attempt < 20; ++attempt) {
    auto v = kv_get("addr:" + peer);
    if (v) {
      std::string addr = v->as_string();
      std::lock_guard<std::mutex> lk(mu_);
      addr_cache_[peer] = addr;
      return addr;
    }
    struct timespec ts = {0, 50 * 1000 * 1000};
    nanosleep(&ts, nullptr);
  }
  return std::nullopt;
}

}  // namespace hypha
