// Gateway broker: peer registry (KV/discovery), topic pub/sub fan-out and
// health — the role the reference's gateway plays for the swarm (relay +
// DHT + gossipsub participant, crates/gateway/src/network.rs:41-49), with
// DHT/gossip semantics collapsed into a broker as SURVEY.md §7 step 2 allows.
#pragma once

#include <arpa/inet.h>
#include <netinet/in.h>
#include <sys/socket.h>
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <condition_variable>
#include <map>
#include <memory>
#include <mutex>
#include <set>
#include <string>
#include <thread>
#include <vector>

#include "json.h"
#include "net.h"

namespace hypha {

// Source IP of a connected socket's peer (IPv4/IPv6), "" on failure.
inline std::string observed_ip(int fd) {
  sockaddr_storage ss{};
  socklen_t len = sizeof(ss);
  if (getpeername(fd, (sockaddr*)&ss, &len) != 0) return "";
  char buf[INET6_ADDRSTRLEN] = {0};
  if (ss.ss_family == AF_INET) {
    inet_ntop(AF_INET, &((sockaddr_in*)&ss)->sin_addr, buf, sizeof(buf));
  } else if (ss.ss_family == AF_INET6) {
    inet_ntop(AF_INET6, &((sockaddr_in6*)&ss)->sin6_addr, buf, sizeof(buf));
  }
  return buf;
}

class Gateway {
 public:
  Gateway() = default;
  explicit Gateway(TlsConfig tls) {
    if (tls.enabled()) tls_ = std::make_unique<TlsContext>(tls);
  }
  ~Gateway() { stop(); }

  void start(int port = 0, const std::string& listen_host = "127.0.0.1") {
    listen_fd_ = tcp_listen(port, listen_host);
    if (listen_fd_ < 0) throw std::runtime_error("gateway: cannot listen");
    port_ = listen_port(listen_fd_);
    running_ = true;
    accept_thread_ = std::thread([this] { accept_loop(); });
  }

  void stop() {
    if (!running_.exchange(false)) return;
    // Wake every blocked thread with shutdown() only (fds stay open — no
    // fd-reuse race); each conn thread closes its own socket when it exits.
    if (listen_fd_ >= 0) ::shutdown(listen_fd_, 2);
    {
      std::lock_guard<std::mutex> lk(mu_);
      for (int fd : conn_fds_) ::shutdown(fd, SHUT_RDWR);
      for (int fd : circuit_fds_) ::shutdown(fd, SHUT_RDWR);
    }
    circ_cv_.notify_all();  // abort any circuit handshake still waiting
    if (accept_thread_.joinable()) accept_thread_.join();
    {
      // UNBOUNDED drain of the detached conn handlers: every conn and
      // circuit fd was just shutdown, so each handler unblocks and exits;
      // returning before they drain would let them touch a destroyed
      // Gateway (TSAN caught exactly that with a bounded wait under its
      // ~20x slowdown).
      std::unique_lock<std::mutex> lk(mu_);
      conn_cv_.wait(lk, [&] { return live_conns_ == 0; });
      peers_.clear();
      subs_.clear();
    }
    if (listen_fd_ >= 0) {
      ::close(listen_fd_);
      listen_fd_ = -1;
    }
  }

  int port() const { return port_; }

 private:
  void accept_loop() {
    while (running_) {
      int fd = ::accept(listen_fd_, nullptr, nullptr);
      if (fd < 0) break;
      {
        // register BEFORE the handler thread exists (and before any TLS
        // handshake can block), so a concurrent stop() always sees the fd;
        // if stop() already swept, wake ourselves
        std::lock_guard<std::mutex> lk(mu_);
        conn_fds_.insert(fd);
        ++live_conns_;
        if (!running_) ::shutdown(fd, SHUT_RDWR);
      }
      std::thread([this, fd] {
        SSL* ssl = nullptr;
        bool done = false;
        if (tls_) {
          // Relay legs are PLAIN sockets carrying length-prefixed JSON
          // hellos; the end-to-end peer TLS runs THROUGH the circuit (the
          // relay sees only its records — reference relay-server role,
          // gateway/src/network.rs:44). First byte discriminates: a TLS
          // ClientHello starts 0x16, our frame length prefix starts 0x00.
          char b0 = 0;
          ssize_t pk = ::recv(fd, &b0, 1, MSG_PEEK);
          if (pk == 1 && b0 == 0x00) {
            handle_conn(fd, nullptr, /*relay_only=*/true);
            done = true;
          } else {
            ssl = tls_->wrap(fd, true);
            if (!ssl) {
              // unauthenticated peer rejected at handshake: deregister
              // while the fd is still open, then close
              unregister_conn(fd);
              ::close(fd);
              return;
            }
          }
        }
        if (!done) handle_conn(fd, ssl);
        unregister_conn(fd);
      }).detach();
    }
    // listen_fd_ is closed by stop() after this thread is joined
  }

  void unregister_conn(int fd) {
    std::lock_guard<std::mutex> lk(mu_);
    conn_fds_.erase(fd);  // removed while the fd is STILL open (no reuse race)
    --live_conns_;
    conn_cv_.notify_all();
  }

  // NOTE: the caller (accept_loop lambda) holds the conn-registry entry for
  // `fd` and deregisters after this returns; the socket itself closes when
  // its last shared_ptr drops, which is always after deregistration.
  void handle_conn(int fd, SSL* ssl = nullptr, bool relay_only = false) {
    auto sock = std::make_shared<MsgSocket>(fd, ssl);
    handle_conn_body(sock, relay_only);
  }

  void handle_conn_body(std::shared_ptr<MsgSocket> sock, bool relay_only) {
    std::string peer;  // set once registered (persistent connection)
    // With mTLS the verified cert CN is the only identity we trust: the
    // self-claimed 'peer'/'from' fields must match it (reference model,
    // rfc/2025-05-30_mtls.md — PeerId is derived from the cert key).
    const std::string cn = sock->peer_identity();  // "" on plain sockets
    // A malformed message (Json::at throws) must drop THIS connection, not
    // std::terminate the whole broker: the per-connection thread is detached.
    try {
      while (running_) {
        auto msg = sock->recv_json();
        if (!msg) break;
        std::string kind = msg->get_or("kind", Json("")).as_string();
        if (kind == "relay_connect") {
          relay_connect(sock, *msg);
          return;  // the connection IS the circuit now (or it failed)
        }
        if (kind == "relay_accept") {
          relay_accept(sock, *msg);
          return;
        }
        if (relay_only) break;  // plain leg may only speak relay kinds
        if (kind == "register") {
          std::string claimed = msg->at("peer").as_string();
          if (!cn.empty() && claimed != cn) {
            Json nak;
            nak["kind"] = "error";
            nak["error"] = "peer name does not match certificate CN";
            sock->send_json(nak);
            break;
          }
          peer = claimed;
          // Observed-address substitution (reference external_address.rs /
          // libp2p identify semantics): a node that registered with a
          // loopback or wildcard host but dialed in from another machine is
          // only dialable at the address the broker OBSERVED; keep the
          // node's listen port, swap the host. Loopback connections (the
          // single-host case) keep the registered address, and a node that
          // advertised a concrete non-loopback host is never overridden.
          std::string addr = msg->at("addr").as_string();
          auto colon = addr.rfind(':');
          std::string reg_host =
              colon == std::string::npos ? addr : addr.substr(0, colon);
          if (reg_host == "127.0.0.1" || reg_host == "localhost" ||
              reg_host == "0.0.0.0") {
            std::string obs = observed_ip(sock->fd());
            if (!obs.empty() && obs != "127.0.0.1" && obs != "::1" &&
                obs != "::ffff:127.0.0.1" && colon != std::string::npos)
              addr = obs + addr.substr(colon);
          }
          {
            std::lock_guard<std::mutex> lk(mu_);
            peers_[peer] = sock;
            kv_["addr:" + peer] = Json(addr);
          }
          Json ack;
          ack["kind"] = "registered";
          sock->send_json(ack);
        } else if (kind == "subscribe") {
          std::lock_guard<std::mutex> lk(mu_);
          subs_[msg->at("topic").as_string()].insert(peer);
        } else if (kind == "unsubscribe") {
          std::lock_guard<std::mutex> lk(mu_);
          subs_[msg->at("topic").as_string()].erase(peer);
        } else if (kind == "request") {
          std::string from = msg->get_or("from", Json("")).as_string();
          if (!cn.empty() && !from.empty() && from != cn) {
            Json resp;
            resp["kind"] = "response";
            resp["ok"] = false;
            resp["error"] = "from does not match certificate CN";
            sock->send_json(resp);
            continue;
          }
          Json resp;
          resp["kind"] = "response";
          resp["ok"] = true;
          resp["body"] = handle_request(from,
                                        msg->get_or("type", Json("")).as_string(),
                                        msg->get_or("body", Json(JsonObject{})));
          sock->send_json(resp);
        } else {
          break;
        }
      }
    } catch (const std::exception&) {
      // drop the offending connection; registry cleanup below still runs
    }
    if (!peer.empty()) {
      // identity-guarded cleanup: if the peer already re-registered over a
      // NEW connection, this (old) handler must not erase the fresh entry
      // or its re-subscribed topics
      std::lock_guard<std::mutex> lk(mu_);
      auto it = peers_.find(peer);
      if (it != peers_.end() && it->second == sock) {
        peers_.erase(it);
        for (auto& [_, set] : subs_) set.erase(peer);
      }
    }
  }

  Json handle_request(const std::string& from, const std::string& type, const Json& body) {
    if (type == "publish") {
      std::string topic = body.at("topic").as_string();
      std::vector<std::shared_ptr<MsgSocket>> targets;
      {
        std::lock_guard<std::mutex> lk(mu_);
        auto it = subs_.find(topic);
        if (it != subs_.end())
          for (auto& p : it->second) {
            auto pit = peers_.find(p);
            if (pit != peers_.end() && p != from) targets.push_back(pit->second);
          }
      }
      Json ev;
      ev["kind"] = "pub";
      ev["topic"] = topic;
      ev["from"] = from;
      ev["data"] = body.at("data");
      for (auto& t : targets) t->send_json(ev);
      Json r;
      r["delivered"] = (int64_t)targets.size();
      return r;
    }
    if (type == "kv_put") {
      std::lock_guard<std::mutex> lk(mu_);
      kv_[body.at("key").as_string()] = body.at("value");
      return Json(JsonObject{});
    }
    if (type == "kv_get") {
      std::lock_guard<std::mutex> lk(mu_);
      auto it = kv_.find(body.at("key").as_string());
      Json r;
      r["found"] = it != kv_.end();
      if (it != kv_.end()) r["value"] = it->second;
      return r;
    }
    if (type == "health") {
      Json r;
      r["healthy"] = true;
      return r;
    }
    if (type == "peers") {
      JsonArray arr;
      std::lock_guard<std::mutex> lk(mu_);
      for (auto& [p, _] : peers_) arr.push_back(Json(p));
      Json r;
      r["peers"] = arr;
      return r;
    }
    Json r;
    r["error"] = "unknown request " + type;
    return r;
  }

  int listen_fd_ = -1;
  int port_ = 0;
  std::atomic<bool> running_{false};
  std::thread accept_thread_;
  // ---- relay circuits (opaque byte splice between two legs) ----
  // Circuit state is tiny and guarded by ONE gateway-level mutex/cv pair
  // (circ_mu_/circ_cv_) rather than per-circuit synchronization: circuit
  // setup is rare (one handshake per undialable peer pair), and heap-reused
  // per-circuit mutexes trip gcc-11 libtsan's stale-sync-object false
  // positives while adding no real concurrency.
  struct Circuit {
    std::shared_ptr<MsgSocket> b;  // acceptor leg, set by relay_accept
  };
  std::mutex circ_mu_;
  std::condition_variable circ_cv_;

  void relay_connect(std::shared_ptr<MsgSocket> a, const Json& msg) {
    const std::string to = msg.get_or("to", Json("")).as_string();
    std::shared_ptr<MsgSocket> target;
    {
      std::lock_guard<std::mutex> lk(mu_);
      auto it = peers_.find(to);
      if (it != peers_.end()) target = it->second;
    }
    Json nak;
    nak["ok"] = false;
    if (!target) {
      nak["error"] = "relay: peer not registered: " + to;
      a->send_json(nak);
      return;
    }
    long long id;
    auto circ = std::make_shared<Circuit>();
    {
      std::lock_guard<std::mutex> lk(mu_);
      id = next_circuit_++;
      circuits_[id] = circ;
    }
    Json offer;
    offer["kind"] = "relay_offer";
    offer["circuit"] = (int64_t)id;
    offer["from"] = msg.get_or("from", Json("")).as_string();
    bool pushed = target->send_json(offer);
    std::shared_ptr<MsgSocket> b;
    if (pushed) {
      std::unique_lock<std::mutex> lk(circ_mu_);
      circ_cv_.wait_for(lk, std::chrono::seconds(10),
                        [&] { return (bool)circ->b || !running_; });
      b = circ->b;
    }
    {
      std::lock_guard<std::mutex> lk(mu_);
      circuits_.erase(id);
    }
    if (!b) {
      nak["error"] = "relay: peer did not accept";
      a->send_json(nak);
      return;
    }
    Json ok;
    ok["ok"] = true;
    if (!b->send_json(ok) || !a->send_json(ok)) return;
    // ---- splice: opaque bytes both ways until either side closes ----
    int fa = a->release();
    int fb = b->release();
    {
      // register the raw circuit fds so stop() can wake the pumps (their
      // MsgSockets released ownership and no longer cover them)
      std::lock_guard<std::mutex> lk(mu_);
      circuit_fds_.insert(fa);
      circuit_fds_.insert(fb);
      if (!running_) {
        ::shutdown(fa, 2);
        ::shutdown(fb, 2);
      }
    }
    std::thread rev([fa, fb] { pump(fb, fa); });
    pump(fa, fb);
    ::shutdown(fa, 2);
    ::shutdown(fb, 2);
    rev.join();  // never close an fd another thread may still be using
    {
      // deregister BEFORE closing (the fds are still open here)
      std::lock_guard<std::mutex> lk(mu_);
      circuit_fds_.erase(fa);
      circuit_fds_.erase(fb);
      conn_fds_.erase(fa);
      conn_fds_.erase(fb);
    }
    ::close(fa);
    ::close(fb);
  }

  void relay_accept(std::shared_ptr<MsgSocket> b, const Json& msg) {
    long long id = msg.get_or("circuit", Json((int64_t)0)).as_int();
    std::shared_ptr<Circuit> circ;
    {
      std::lock_guard<std::mutex> lk(mu_);
      auto it = circuits_.find(id);
      if (it != circuits_.end()) circ = it->second;
    }
    if (!circ) {
      Json nak;
      nak["ok"] = false;
      nak["error"] = "relay: unknown circuit";
      b->send_json(nak);
      return;
    }
    {
      std::lock_guard<std::mutex> lk(circ_mu_);
      circ->b = b;
    }
    circ_cv_.notify_all();
    // ownership passes to the connect-side thread via the shared_ptr held
    // in the Circuit: this handler returns immediately and the socket
    // outlives it until the splice finishes.
  }

  static void pump(int src, int dst) {
    char buf[1 << 16];
    while (true) {
      ssize_t n = ::recv(src, buf, sizeof buf, 0);
      if (n <= 0) break;
      ssize_t off = 0;
      while (off < n) {
        // MSG_NOSIGNAL: a torn-down far leg must surface as EPIPE, not a
        // process-killing SIGPIPE in embedders that don't ignore it
        ssize_t w = ::send(dst, buf + off, n - off, MSG_NOSIGNAL);
        if (w <= 0) return;
        off += w;
      }
    }
  }

  std::mutex mu_;
  std::map<long long, std::shared_ptr<Circuit>> circuits_;
  long long next_circuit_ = 1;
  std::map<std::string, std::shared_ptr<MsgSocket>> peers_;
  // Shutdown registries (guarded by mu_): every live inbound conn fd and
  // every raw fd inside an active relay splice. Entries are always removed
  // BEFORE their fd is closed, so stop() can never shutdown a reused fd.
  std::set<int> conn_fds_;
  std::set<int> circuit_fds_;
  int live_conns_ = 0;  // guarded by mu_
  std::condition_variable conn_cv_;
  std::map<std::string, std::set<std::string>> subs_;
  std::map<std::string, Json> kv_;
  std::unique_ptr<TlsContext> tls_;
};

}  // namespace hypha
