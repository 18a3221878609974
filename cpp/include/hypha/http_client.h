// Minimal blocking HTTP(S) GET client for the worker's URI/HuggingFace
// fetch connector.
//
// Mirrors the reference's HttpHfFetcher + validate_fetch semantics
// (/root/reference/crates/worker/src/connector/mod.rs:226-302 — reqwest GET
// streamed to disk — and /root/reference/crates/worker/src/executor/
// bridge.rs:349-377 — http(s)-scheme validation, safe_join traversal guard).
// The reference left host restriction as a TODO ("check it against an allow
// list to restrict access to _trusted_ sources"); here the allow-list is
// implemented and DENIES by default.
//
// Scope: HTTP/1.1 GET, Content-Length or close-delimited bodies, up to 5
// redirects. https:// URLs are accepted by validation (reference parity) and
// attempted with TLS against the system trust store; in the offline test
// environment only http://127.0.0.1 servers are reachable.
#pragma once

#include <arpa/inet.h>
#include <netdb.h>
#include <openssl/ssl.h>
#include <poll.h>
#include <sys/socket.h>
#include <sys/stat.h>
#include <unistd.h>

#include <cstdio>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

namespace hypha {

struct HttpUrl {
  std::string scheme, host, path;
  int port = 0;
};

inline HttpUrl parse_url(const std::string& url) {
  HttpUrl u;
  auto scheme_end = url.find("://");
  if (scheme_end == std::string::npos)
    throw std::runtime_error("fetch: invalid URI (no scheme): " + url);
  u.scheme = url.substr(0, scheme_end);
  if (u.scheme != "http" && u.scheme != "https")
    throw std::runtime_error("fetch: invalid URI: expected http(s)://..., got `" +
                             url + "`");
  std::string rest = url.substr(scheme_end + 3);
  auto path_start = rest.find('/');
  std::string hostport =
      path_start == std::string::npos ? rest : rest.substr(0, path_start);
  u.path = path_start == std::string::npos ? "/" : rest.substr(path_start);
  auto colon = hostport.rfind(':');
  if (colon != std::string::npos && hostport.find(']') == std::string::npos) {
    u.host = hostport.substr(0, colon);
    u.port = std::atoi(hostport.c_str() + colon + 1);
  } else {
    u.host = hostport;
    u.port = u.scheme == "https" ? 443 : 80;
  }
  if (u.host.empty()) throw std::runtime_error("fetch: invalid URI (no host): " + url);
  return u;
}

// Allow-list entry forms: exact host ("example.com"), wildcard suffix
// ("*.example.com"), or host:port. Empty list = deny everything (the
// reference's TODO closed conservatively).
inline bool fetch_allowed(const std::string& host, int port,
                          const std::vector<std::string>& allow) {
  for (const auto& a : allow) {
    std::string ah = a;
    int ap = -1;
    auto colon = ah.rfind(':');
    if (colon != std::string::npos && ah.find(']') == std::string::npos &&
        ah.find_first_not_of("0123456789", colon + 1) == std::string::npos) {
      ap = std::atoi(ah.c_str() + colon + 1);
      ah = ah.substr(0, colon);
    }
    if (ap != -1 && ap != port) continue;
    if (ah == "*") return true;
    if (ah.size() > 2 && ah[0] == '*' && ah[1] == '.') {
      std::string suffix = ah.substr(1);  // ".example.com"
      if (host.size() > suffix.size() &&
          host.compare(host.size() - suffix.size(), suffix.size(), suffix) == 0)
        return true;
      if (host == ah.substr(2)) return true;  // bare domain matches *.domain
    } else if (host == ah) {
      return true;
    }
  }
  return false;
}

namespace detail {

class HttpConn {
 public:
  HttpConn(const HttpUrl& u, double timeout_s) {
    struct addrinfo hints = {}, *res = nullptr;
    hints.ai_family = AF_UNSPEC;
    hints.ai_socktype = SOCK_STREAM;
    std::string port = std::to_string(u.port);
    if (getaddrinfo(u.host.c_str(), port.c_str(), &hints, &res) != 0 || !res)
      throw std::runtime_error("fetch: cannot resolve host " + u.host);
    fd_ = socket(res->ai_family, SOCK_STREAM, 0);
    if (fd_ < 0) {
      freeaddrinfo(res);
      throw std::runtime_error("fetch: socket failed");
    }
    struct timeval tv;
    tv.tv_sec = (long)timeout_s;
    tv.tv_usec = (long)((timeout_s - (long)timeout_s) * 1e6);
    setsockopt(fd_, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
    setsockopt(fd_, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof(tv));
    int rc = connect(fd_, res->ai_addr, res->ai_addrlen);
    freeaddrinfo(res);
    if (rc != 0) {
      close(fd_);
      throw std::runtime_error("fetch: connect to " + u.host + ":" + port +
                               " failed");
    }
    if (u.scheme == "https") {
      ssl_ctx_ = SSL_CTX_new(TLS_client_method());
      if (!ssl_ctx_) throw std::runtime_error("fetch: SSL_CTX_new failed");
      SSL_CTX_set_default_verify_paths(ssl_ctx_);
      SSL_CTX_set_verify(ssl_ctx_, SSL_VERIFY_PEER, nullptr);
      ssl_ = SSL_new(ssl_ctx_);
      SSL_set_fd(ssl_, fd_);
      SSL_set_tlsext_host_name(ssl_, u.host.c_str());
      if (SSL_connect(ssl_) != 1)
        throw std::runtime_error("fetch: TLS handshake with " + u.host + " failed");
    }
  }

  ~HttpConn() {
    if (ssl_) {
      SSL_shutdown(ssl_);
      SSL_free(ssl_);
    }
    if (ssl_ctx_) SSL_CTX_free(ssl_ctx_);
    if (fd_ >= 0) close(fd_);
  }

  void send_all(const std::string& data) {
    size_t off = 0;
    while (off < data.size()) {
      ssize_t n = ssl_ ? SSL_write(ssl_, data.data() + off, (int)(data.size() - off))
                       : send(fd_, data.data() + off, data.size() - off, MSG_NOSIGNAL);
      if (n <= 0) throw std::runtime_error("fetch: send failed");
      off += (size_t)n;
    }
  }

  // returns bytes read, 0 on orderly EOF
  ssize_t read_some(char* buf, size_t cap) {
    ssize_t n = ssl_ ? SSL_read(ssl_, buf, (int)cap) : recv(fd_, buf, cap, 0);
    if (n < 0) throw std::runtime_error("fetch: recv failed/timeout");
    return n;
  }

 private:
  int fd_ = -1;
  SSL_CTX* ssl_ctx_ = nullptr;
  SSL* ssl_ = nullptr;
};

}  // namespace detail

// GET `url` and stream the body into `out_path` (mode 0600). Applies the
// allow-list to the url host AND every redirect hop. Returns bytes written.
inline long long http_get_to_file(const std::string& url, const std::string& out_path,
                                  const std::vector<std::string>& allow,
                                  double timeout_s = 30.0, int max_redirects = 5,
                                  long long max_bytes = 16LL << 30) {
  std::string cur = url;
  for (int hop = 0; hop <= max_redirects; ++hop) {
    HttpUrl u = parse_url(cur);
    if (!fetch_allowed(u.host, u.port, allow))
      throw std::runtime_error("fetch: host not in allow-list: " + u.host + ":" +
                               std::to_string(u.port));
    detail::HttpConn conn(u, timeout_s);
    std::string req = "GET " + u.path + " HTTP/1.1\r\nHost: " + u.host +
                      "\r\nUser-Agent: hypha-amd/0.2\r\nAccept: */*\r\n"
                      "Connection: close\r\n\r\n";
    conn.send_all(req);

    // read headers
    std::string hdr;
    char buf[16384];
    size_t body_start = std::string::npos;
    while (body_start == std::string::npos) {
      ssize_t n = conn.read_some(buf, sizeof(buf));
      if (n == 0) throw std::runtime_error("fetch: connection closed mid-headers");
      hdr.append(buf, (size_t)n);
      body_start = hdr.find("\r\n\r\n");
      if (hdr.size() > (1 << 20))
        throw std::runtime_error("fetch: oversized response headers");
    }
    std::string head = hdr.substr(0, body_start);
    std::string body0 = hdr.substr(body_start + 4);

    int status = 0;
    if (head.size() > 12) status = std::atoi(head.c_str() + 9);
    auto find_header = [&](const char* name) -> std::string {
      std::string lower;
      lower.reserve(head.size());
      for (char c : head) lower.push_back((char)tolower((unsigned char)c));
      std::string key = std::string("\r\n") + name + ":";
      auto p = lower.find(key);
      if (p == std::string::npos) return "";
      p += key.size();
      auto e = head.find("\r\n", p);
      std::string v = head.substr(p, e - p);
      while (!v.empty() && (v.front() == ' ' || v.front() == '\t')) v.erase(0, 1);
      while (!v.empty() && (v.back() == '\r' || v.back() == ' ')) v.pop_back();
      return v;
    };

    if (status >= 300 && status < 400) {
      std::string loc = find_header("location");
      if (loc.empty()) throw std::runtime_error("fetch: redirect without Location");
      if (loc[0] == '/') loc = u.scheme + "://" + u.host + ":" +
                               std::to_string(u.port) + loc;
      cur = loc;
      continue;
    }
    if (status != 200)
      throw std::runtime_error("fetch: HTTP " + std::to_string(status) + " for " + cur);

    std::string te = find_header("transfer-encoding");
    if (!te.empty() && te != "identity")
      throw std::runtime_error("fetch: unsupported transfer-encoding: " + te);
    long long content_len = -1;
    std::string cl = find_header("content-length");
    if (!cl.empty()) content_len = atoll(cl.c_str());

    FILE* f = fopen(out_path.c_str(), "wb");
    if (!f) throw std::runtime_error("fetch: cannot open " + out_path);
    chmod(out_path.c_str(), 0600);
    long long written = 0;
    auto write_chunk = [&](const char* p, size_t n) {
      if (n && fwrite(p, 1, n, f) != n) {
        fclose(f);
        throw std::runtime_error("fetch: short write to " + out_path);
      }
      written += (long long)n;
      if (written > max_bytes) {
        fclose(f);
        throw std::runtime_error("fetch: response exceeds size cap for " +
                                 out_path);
      }
    };
    if (content_len > max_bytes) {
      fclose(f);
      throw std::runtime_error("fetch: declared size exceeds cap for " + cur);
    }
    write_chunk(body0.data(), body0.size());
    while (content_len < 0 || written < content_len) {
      ssize_t n = conn.read_some(buf, sizeof(buf));
      if (n == 0) break;
      write_chunk(buf, (size_t)n);
    }
    fclose(f);
    if (content_len >= 0 && written != content_len)
      throw std::runtime_error("fetch: short body (" + std::to_string(written) +
                               " of " + std::to_string(content_len) + " bytes)");
    return written;
  }
  throw std::runtime_error("fetch: too many redirects for " + url);
}

}  // namespace hypha
