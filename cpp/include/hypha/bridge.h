// Job Bridge: the language-agnostic executor API — HTTP over a Unix socket.
// Parity with /root/reference/crates/worker/src/executor/bridge.rs:
//   routes GET /openapi.json, POST /resources/fetch, POST /resources/send,
//   GET /resources/receive (SSE pointer events {path,size,from_peer}),
//   POST /status/send (:154-160); socket mode 0600 (:128-182);
//   safe_join path-traversal guard (:330-346).
// The executor subprocess gets {SOCKET_PATH}/{WORK_DIR}/{JOB_JSON} on its
// command line (process.rs:201-205) and speaks only this API.
#pragma once

#include <fcntl.h>
#include <sys/socket.h>
#include <sys/stat.h>
#include <sys/un.h>
#include <unistd.h>

#include <atomic>
#include <condition_variable>
#include <deque>
#include <functional>
#include <mutex>
#include <string>
#include <chrono>
#include <thread>
#include <vector>

#include "json.h"

namespace hypha {

// Reject paths that escape the work dir (bridge.rs safe_join :330-346).
inline bool safe_subpath(const std::string& rel) {
  if (rel.empty() || rel[0] == '/') return false;
  size_t pos = 0;
  while (pos < rel.size()) {
    size_t next = rel.find('/', pos);
    std::string seg = rel.substr(pos, next == std::string::npos ? std::string::npos : next - pos);
    if (seg == "..") return false;
    if (next == std::string::npos) break;
    pos = next + 1;
  }
  return true;
}

class Bridge {
 public:
  using FetchCb = std::function<Json(const Json& ref)>;
  using SendCb = std::function<Json(const Json& ref, const std::string& path)>;
  using StatusCb = std::function<Json(const Json& progress)>;

  Bridge(std::string socket_path, std::string work_dir)
      : socket_path_(std::move(socket_path)), work_dir_(std::move(work_dir)) {}
  ~Bridge() { stop(); }

  FetchCb fetch_cb;
  SendCb send_cb;
  StatusCb status_cb;

  void start() {
    ::unlink(socket_path_.c_str());
    fd_ = ::socket(AF_UNIX, SOCK_STREAM, 0);
    struct sockaddr_un addr = {};
    addr.sun_family = AF_UNIX;
    snprintf(addr.sun_path, sizeof addr.sun_path, "%s", socket_path_.c_str());
    if (::bind(fd_, (struct sockaddr*)&addr, sizeof addr) != 0 || ::listen(fd_, 16) != 0)
      throw std::runtime_error("bridge: cannot bind " + socket_path_);
    ::chmod(socket_path_.c_str(), 0600);
    running_ = true;
    accept_thread_ = std::thread([this] {
      while (running_) {
        int cfd = ::accept(fd_, nullptr, nullptr);
        if (cfd < 0) break;
        std::thread([this, cfd] { handle(cfd); }).detach();
      }
    });
  }

  void stop() {
    if (!running_.exchange(false)) return;
    ::shutdown(fd_, SHUT_RDWR);
    ::close(fd_);
    {
      std::lock_guard<std::mutex> lk(ev_mu_);
      closed_ = true;
    }
    ev_cv_.notify_all();
    if (accept_thread_.joinable()) accept_thread_.join();
    ::unlink(socket_path_.c_str());
  }

  // Worker side: announce a received resource to the executor (SSE event).
  void push_event(const Json& ev) {
    {
      std::lock_guard<std::mutex> lk(ev_mu_);
      events_.push_back(ev);
    }
    ev_cv_.notify_all();
  }

 private:
  static bool read_until(int fd, std::string& buf, const char* delim) {
    char c;
    size_t dl = strlen(delim);
    while (buf.size() < 1 << 20) {
      ssize_t r = ::recv(fd, &c, 1, 0);
      if (r <= 0) return false;
      buf += c;
      if (buf.size() >= dl && buf.compare(buf.size() - dl, dl, delim) == 0) return true;
    }
    return false;
  }

  static void respond(int fd, int code, const std::string& body,
                      const char* ctype = "application/json") {
    char hdr[256];
    int n = snprintf(hdr, sizeof hdr,
                     "HTTP/1.1 %d %s\r\nContent-Type: %s\r\nContent-Length: %zu\r\n"
                     "Connection: close\r\n\r\n",
                     code, code == 200 ? "OK" : "Error", ctype, body.size());
    ::send(fd, hdr, n, MSG_NOSIGNAL);
    ::send(fd, body.data(), body.size(), MSG_NOSIGNAL);
  }

  void handle(int fd) {
    std::string head;
    if (!read_until(fd, head, "\r\n\r\n")) {
      ::close(fd);
      return;
    }
    // request line
    size_t sp1 = head.find(' ');
    size_t sp2 = sp1 == std::string::npos ? std::string::npos : head.find(' ', sp1 + 1);
    if (sp2 == std::string::npos) {
      respond(fd, 400, "{\"error\":\"malformed request line\"}");
      ::close(fd);
      return;
    }
    std::string method = head.substr(0, sp1);
    std::string path = head.substr(sp1 + 1, sp2 - sp1 - 1);
    // content-length (capped: job configs / status posts are small; file
    // payloads travel over the peer streams, never through the bridge body)
    size_t clen = 0;
    {
      std::string lower;
      for (char ch : head) lower += (char)tolower((unsigned char)ch);
      size_t p = lower.find("content-length:");
      if (p != std::string::npos) clen = strtoul(lower.c_str() + p + 15, nullptr, 10);
    }
    if (clen > (256u << 20)) {
      respond(fd, 400, "{\"error\":\"body too large\"}");
      ::close(fd);
      return;
    }
    std::string body(clen, '\0');
    size_t got = 0;
    while (got < clen) {
      ssize_t r = ::recv(fd, body.data() + got, clen - got, 0);
      if (r <= 0) break;
      got += r;
    }

    try {
      if (path == "/openapi.json") {
        respond(fd, 200,
                "{\"openapi\":\"3.0.0\",\"info\":{\"title\":\"hypha job bridge\","
                "\"version\":\"0.0.1\"},\"paths\":{\"/resources/fetch\":{},"
                "\"/resources/send\":{},\"/resources/receive\":{},\"/status/send\":{}}}");
      } else if (path == "/resources/fetch" && method == "POST") {
        Json req = Json::parse(body);
        Json out = fetch_cb ? fetch_cb(req) : Json(JsonObject{});
        respond(fd, 200, out.dump());
      } else if (path == "/resources/send" && method == "POST") {
        Json req = Json::parse(body);
        std::string rel = req.at("path").as_string();
        if (!safe_subpath(rel)) {
          respond(fd, 400, "{\"error\":\"path escapes work dir\"}");
        } else {
          Json out = send_cb ? send_cb(req.at("reference"), work_dir_ + "/" + rel)
                             : Json(JsonObject{});
          respond(fd, 200, out.dump());
        }
      } else if (path == "/status/send" && method == "POST") {
        Json req = Json::parse(body);
        Json out = status_cb ? status_cb(req) : Json(JsonObject{});
        respond(fd, 200, out.dump());
      } else if (path == "/resources/receive") {
        // SSE: stream pointer events until the connection drops
        const char* hdr =
            "HTTP/1.1 200 OK\r\nContent-Type: text/event-stream\r\n"
            "Cache-Control: no-cache\r\n\r\n";
        ::send(fd, hdr, strlen(hdr), MSG_NOSIGNAL);
        size_t cursor = 0;
        while (running_) {
          Json ev;
          bool have = false;
          {
            std::unique_lock<std::mutex> lk(ev_mu_);
            // 30 s keepalive comments: a DiLoCo round can be quiet for many
            // minutes and executor HTTP clients enforce read timeouts
            ev_cv_.wait_for(lk, std::chrono::seconds(30),
                            [&] { return closed_ || events_.size() > cursor; });
            if (closed_ && events_.size() <= cursor) break;
            if (events_.size() > cursor) {
              ev = events_[cursor++];
              have = true;
            }
          }
          std::string line = have ? "data: " + ev.dump() + "\n\n" : ": keepalive\n\n";
          if (::send(fd, line.data(), line.size(), MSG_NOSIGNAL) <= 0) break;
        }
      } else {
        respond(fd, 404, "{\"error\":\"not found\"}");
      }
    } catch (const std::exception& e) {
      respond(fd, 500, std::string("{\"error\":\"") + e.what() + "\"}");
    }
    ::close(fd);
  }

  std::string socket_path_, work_dir_;
  int fd_ = -1;
  std::atomic<bool> running_{false};
  bool closed_ = false;
  std::thread accept_thread_;
  std::mutex ev_mu_;
  std::condition_variable ev_cv_;
  std::deque<Json> events_;  // retained so late subscribers replay history
};

}  // namespace hypha
