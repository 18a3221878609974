// Control-plane networking: length-prefixed JSON messages over TCP.
//
// Replaces the reference's libp2p stack (crates/network) with a slim native
// layer exposing the same five interaction patterns:
//   * request/response with typed handler registration (request_response.rs)
//   * topic pub/sub via the gateway broker (gossipsub.rs, topic hypha/worker)
//   * KV/discovery registry hosted by the gateway (kad.rs DHT semantics)
//   * raw byte streams for tensor files (stream_push.rs / stream_pull.rs)
//   * health probes (messages/src/lib.rs health codec)
// Bulk tensor movement on-node is RCCL over xGMI (hypha_amd.parallel.comm);
// this layer carries control messages and the WAN/file tensor path.
#pragma once

#include <atomic>
#include <condition_variable>
#include <functional>
#include <map>
#include <memory>
#include <mutex>
#include <optional>
#include <set>
#include <string>
#include <thread>
#include <vector>

#include "json.h"
#include "tls.h"

namespace hypha {

// ---- low-level framed socket ----------------------------------------------

class MsgSocket {
 public:
  explicit MsgSocket(int fd, SSL* ssl = nullptr) : fd_(fd), ssl_(ssl) {}
  ~MsgSocket();
  MsgSocket(const MsgSocket&) = delete;

  bool send_json(const Json& j);
  std::optional<Json> recv_json();  // nullopt on EOF/error
  bool send_raw(const char* data, size_t n);
  bool recv_raw(char* data, size_t n);
  void close_now();
  // Wake any thread blocked in recv/SSL io WITHOUT closing the fd: the fd
  // number stays reserved, so a concurrent reader can never race a kernel
  // fd-reuse (close_now is only safe once no other thread uses the socket).
  void shutdown_now();
  int fd() const { return fd_; }
  // Detach the fd (relay handshake: a few plain frames are exchanged and
  // then the SAME socket carries the end-to-end TLS/protocol bytes).
  int release() {
    int f = fd_;
    fd_ = -1;
    return f;
  }
  std::string peer_identity() const { return TlsContext::peer_cn(ssl_); }

 private:
  bool ssl_wait_readable();  // poll until data (or error) without the io lock

  int fd_;
  SSL* ssl_ = nullptr;  // owned; non-null when the link is mutually-TLS
  std::mutex write_mu_;  // framing lock for plain sockets
  // OpenSSL forbids concurrent SSL_read/SSL_write on one SSL: all TLS io is
  // serialized by io_mu_, with readers polling for data BEFORE taking it so
  // writers are never starved by a blocked read.
  std::mutex io_mu_;
};

int tcp_connect(const std::string& host, int port, double timeout_s = 5.0);
// Returns listen fd; port 0 = ephemeral. Binds loopback by default —
// pass "0.0.0.0" (or an interface address) to accept off-host peers.
int tcp_listen(int port, const std::string& host = "127.0.0.1");
int listen_port(int listen_fd);

// Process-global transport byte counters (reference telemetry/src/
// bandwidth.rs:33-60: bandwidth::Transport wraps every connection with
// inbound/outbound counters). Counted at the MsgSocket framing layer, so
// requests, pub/sub, KV and tensor streams are all included.
struct BandwidthStats {
  unsigned long long inbound_bytes;
  unsigned long long outbound_bytes;
};
BandwidthStats bandwidth_stats();

// "10.0.0.0/8" contains "10.1.2.3"? (reference utils.rs:18
// find_containing_cidr + dial.rs CIDR exclusion). IPv4 dotted-quad only;
// non-numeric hosts are never matched.
bool cidr_contains(const std::string& cidr, const std::string& host);

// ---- node ------------------------------------------------------------------

// A Node serves typed requests on its own TCP port and talks to peers either
// directly (addresses from the gateway registry) or via the gateway broker
// (pub/sub, KV). Handler threads are per-connection (control-plane rates).
class Node {
 public:
  using Handler = std::function<Json(const std::string& from, const Json& body)>;
  // Stream handler: gets header + the connection for raw payload IO.
  using StreamHandler =
      std::function<void(const std::string& from, const Json& header, MsgSocket& sock)>;

  Node(std::string name, std::string gateway_host, int gateway_port,
       TlsConfig tls = {});
  ~Node();

  // start serving on `port` (0 = ephemeral) and register with the gateway
  void start(int port = 0);
  void stop();
  int port() const { return port_; }
  const std::string& name() const { return name_; }

  void on(const std::string& type, Handler h);
  void on_stream(const std::string& type, StreamHandler h);

  // direct request to a named peer (address resolved via gateway registry)
  Json request(const std::string& peer, const std::string& type, const Json& body,
               double timeout_s = 10.0);
  // open a raw stream to a peer: send header, then return the socket for
  // payload bytes (caller closes)
  std::unique_ptr<MsgSocket> open_stream(const std::string& peer, const std::string& type,
                                         const Json& header);

  // gateway-brokered pub/sub + KV
  void publish(const std::string& topic, const Json& data);
  void subscribe(const std::string& topic, std::function<void(const std::string&, const Json&)> cb);
  void unsubscribe(const std::string& topic);  // gossipsub.rs:232-300 parity
  void kv_put(const std::string& key, const Json& value);
  std::optional<Json> kv_get(const std::string& key);

  std::optional<std::string> resolve(const std::string& peer);

  // Dial exclusion (dial.rs CIDR exclusion): peers whose resolved address
  // falls in an excluded CIDR are refused before connect.
  void set_exclude_cidrs(std::vector<std::string> cidrs) {
    exclude_cidrs_ = std::move(cidrs);
  }

  // External-address advertising (reference external_address.rs:15-137):
  // the host other peers should dial this node at. Unset = register as
  // 127.0.0.1 and let the gateway substitute the address it OBSERVED the
  // connection from (identify-style), which covers multi-host LANs
  // automatically; set it explicitly when the observed address is not the
  // dialable one (NAT with port-forward, multi-homed hosts).
  void set_advertise_host(std::string host) { advertise_host_ = std::move(host); }

  // Bind address for this node's own listener (default loopback-only);
  // "0.0.0.0" exposes it to the LAN for multi-host clusters. Set before
  // start().
  void set_listen_host(std::string host) { listen_host_ = std::move(host); }

 private:
  void check_dialable(const std::string& peer, const std::string& host) const;
  std::vector<std::string> exclude_cidrs_;

  void accept_loop();
  void handle_conn(MsgSocket& sock);
  void unregister_conn(int fd);
  void gateway_listen_loop();

 public:
  // Gateway high availability: additional brokers tried in order when the
  // active one is unreachable (each failover re-registers and replays
  // subscriptions; daemons use on_gateway_reconnect to re-announce their
  // KV records — gateways do not replicate state).
  void add_fallback_gateway(const std::string& host, int port) {
    gw_list_.emplace_back(host, port);
  }
  void on_gateway_reconnect(std::function<void()> cb) {
    reconnect_cb_ = std::move(cb);
  }
  // One-shot RR to the active gateway over the (m)TLS transport — used by
  // the daemons' `probe` subcommand (no registration side effects).
  Json gateway_request(const std::string& type, const Json& body);

 private:
  // Relay fallback (reference gateway relay-server role, network.rs:44):
  // when a peer cannot be dialed directly, the connection is tunneled
  // through the gateway as an opaque byte circuit; with mTLS the
  // end-to-end handshake still runs peer<->peer (the relay sees only
  // TLS records). relay_dial returns a connected fd or -1.
  int relay_dial(const std::string& peer, double timeout_s);
  void relay_accept_run(long long circuit);
  bool gateway_connect();  // (re)connect + register + replay subscriptions
  int dial_gateway(double timeout_s);  // active gateway, then failover

  std::string name_, gw_host_;
  std::string advertise_host_;
  std::string listen_host_ = "127.0.0.1";
  int gw_port_;
  // gw_host_/gw_port_ are rewritten by failover while relay threads read
  // them; all post-start access goes through gw_addr()/set_gw_addr()
  mutable std::mutex gw_addr_mu_;
  std::pair<std::string, int> gw_addr() const {
    std::lock_guard<std::mutex> lk(gw_addr_mu_);
    return {gw_host_, gw_port_};
  }
  void set_gw_addr(const std::string& h, int p) {
    std::lock_guard<std::mutex> lk(gw_addr_mu_);
    gw_host_ = h;
    gw_port_ = p;
  }
  // failover gateway candidates (primary + fallbacks); gw_host_/gw_port_
  // is the ACTIVE one, switched by gateway_connect on unreachability
  std::string gw_primary_host_;
  int gw_primary_port_ = 0;
  std::vector<std::pair<std::string, int>> gw_list_;
  std::function<void()> reconnect_cb_;  // fired after re-register+replay
  std::unique_ptr<TlsContext> tls_;
  int listen_fd_ = -1;
  int port_ = 0;
  std::atomic<bool> running_{false};
  std::thread accept_thread_, gw_thread_;
  std::mutex mu_;
  std::map<std::string, Handler> handlers_;
  std::map<std::string, StreamHandler> stream_handlers_;
  std::map<std::string, std::function<void(const std::string&, const Json&)>> subs_;
  std::map<std::string, std::string> addr_cache_;
  std::unique_ptr<MsgSocket> gw_sock_;       // persistent broker connection
  std::mutex gw_mu_;
  // Inbound-connection registry: stop() shutdowns every live conn fd (waking
  // its detached handler thread) and then waits for the handlers to drain,
  // so sockets are only ever CLOSED by the thread that owns them.
  std::mutex conn_mu_;
  std::condition_variable conn_cv_;
  std::set<int> conn_fds_;   // open fds, removed BEFORE their close
  int live_conns_ = 0;       // guarded by conn_mu_
};

}  // namespace hypha
