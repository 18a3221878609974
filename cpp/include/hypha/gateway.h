// Gateway broker: peer registry (KV/discovery), topic pub/sub fan-out and
// health — the role the reference's gateway plays for the swarm (relay +
// DHT + gossipsub participant, crates/gateway/src/network.rs:41-49), with
// DHT/gossip semantics collapsed into a broker as SURVEY.md §7 step 2 allows.
#pragma once

#include <sys/socket.h>
#include <unistd.h>

#include <atomic>
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

class Gateway {
 public:
  Gateway() = default;
  explicit Gateway(TlsConfig tls) {
    if (tls.enabled()) tls_ = std::make_unique<TlsContext>(tls);
  }
  ~Gateway() { stop(); }

  void start(int port = 0) {
    listen_fd_ = tcp_listen(port);
    if (listen_fd_ < 0) throw std::runtime_error("gateway: cannot listen");
    port_ = listen_port(listen_fd_);
    running_ = true;
    accept_thread_ = std::thread([this] { accept_loop(); });
  }

  void stop() {
    if (!running_.exchange(false)) return;
    if (listen_fd_ >= 0) {
      ::shutdown(listen_fd_, 2);
    }
    if (accept_thread_.joinable()) accept_thread_.join();
    std::lock_guard<std::mutex> lk(mu_);
    for (auto& [_, s] : peers_) s->close_now();
    peers_.clear();
    subs_.clear();
  }

  int port() const { return port_; }

 private:
  void accept_loop() {
    while (running_) {
      int fd = ::accept(listen_fd_, nullptr, nullptr);
      if (fd < 0) break;
      std::thread([this, fd] {
        SSL* ssl = nullptr;
        if (tls_) {
          ssl = tls_->wrap(fd, true);
          if (!ssl) {
            ::close(fd);
            return;  // unauthenticated peer rejected at handshake
          }
        }
        handle_conn(fd, ssl);
      }).detach();
    }
    ::close(listen_fd_);
    listen_fd_ = -1;
  }

  void handle_conn(int fd, SSL* ssl = nullptr) {
    auto sock = std::make_shared<MsgSocket>(fd, ssl);
    std::string peer;  // set once registered (persistent connection)
    // With mTLS the verified cert CN is the only identity we trust: the
    // self-claimed 'peer'/'from' fields must match it (reference model,
    // rfc/2025-05-30_mtls.md — PeerId is derived from the cert key).
    const std::string cn = ssl ? sock->peer_identity() : std::string();
    // A malformed message (Json::at throws) must drop THIS connection, not
    // std::terminate the whole broker: the per-connection thread is detached.
    try {
      while (running_) {
        auto msg = sock->recv_json();
        if (!msg) break;
        std::string kind = msg->get_or("kind", Json("")).as_string();
        if (kind == "register") {
          std::string claimed = msg->at("peer").as_string();
          if (ssl && !cn.empty() && claimed != cn) {
            Json nak;
            nak["kind"] = "error";
            nak["error"] = "peer name does not match certificate CN";
            sock->send_json(nak);
            break;
          }
          peer = claimed;
          {
            std::lock_guard<std::mutex> lk(mu_);
            peers_[peer] = sock;
            kv_["addr:" + peer] = msg->at("addr");
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
          if (ssl && !cn.empty() && !from.empty() && from != cn) {
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
  std::mutex mu_;
  std::map<std::string, std::shared_ptr<MsgSocket>> peers_;
  std::map<std::string, std::set<std::string>> subs_;
  std::map<std::string, Json> kv_;
  std::unique_ptr<TlsContext> tls_;
};

}  // namespace hypha
