// hypha-data: dataset node. Scans a directory of SafeTensors slice files
// (one file = one slice), announces {dataset: num_slices} in the registry,
// and serves pull streams by slice index.
// Parity with /root/reference/crates/data/src/bin/hypha-data.rs:150-209 and
// tensor_data.rs:8-16 (files served verbatim).

#include <dirent.h>
#include <signal.h>
#include <sys/stat.h>

#include <algorithm>
#include <string>
#include <thread>
#include <vector>

#include "hypha/json.h"
#include "hypha/net.h"

using namespace hypha;

int main(int argc, char** argv) {
  std::string name = "data", gw_host = "127.0.0.1", dataset = "dataset", dir = ".";
  std::string advertise_host, listen_host = "127.0.0.1";
  int gw_port = 0, port = 0;
  bool probe = false, init = false;
  std::vector<std::string> exclude_cidrs, fallback_gws;
  TlsConfig tls;
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    auto next = [&] { return std::string(argv[++i]); };
    if (a == "--name") name = next();
    else if (a == "--gateway-host") gw_host = next();
    else if (a == "--gateway-port") gw_port = std::stoi(next());
    else if (a == "--port") port = std::stoi(next());
    else if (a == "--dataset") dataset = next();
    else if (a == "--dataset-path") dir = next();
    else if (a == "--tls-cert") tls.cert_path = next();
    else if (a == "--tls-key") tls.key_path = next();
    else if (a == "--tls-ca") tls.ca_path = next();
    else if (a == "--tls-crl") tls.crl_path = next();
    else if (a == "--exclude-cidr") exclude_cidrs.push_back(next());
    else if (a == "--advertise-host") advertise_host = next();
    else if (a == "--listen-host") listen_host = next();
    else if (a == "--fallback-gateway") fallback_gws.push_back(next());
    else if (a == "probe") probe = true;
    else if (a == "init") init = true;
  }
  signal(SIGPIPE, SIG_IGN);
  if (init) {  // reference CLI Init subcommand: emit a commented config
    printf("# hypha-data configuration (flags)\n"
           "# --name data-0                 node name in the registry\n"
           "# --gateway-host/--gateway-port gateway broker address\n"
           "# --dataset synth               dataset name to announce\n"
           "# --dataset-path /data/slices   dir of SafeTensors slice files\n"
           "# --tls-cert/--tls-key/--tls-ca [--tls-crl]  mTLS identity\n"
           "# --fallback-gateway host:port  additional broker(s) tried on gateway loss\n"
           "# --advertise-host 10.0.0.5     host peers should dial this node at\n"
           "# --listen-host 0.0.0.0         bind address (default loopback-only)\n"
           "#                               (default: gateway-observed address)\n");
    return 0;
  }
  if (probe) {  // readiness = mTLS registration round-trip (data.rs probe)
    try {
      Node pn(name, gw_host, gw_port, tls);  // cert CN must equal `name`
      for (const auto& g : fallback_gws) {
        auto c = g.rfind(':');
        pn.add_fallback_gateway(g.substr(0, c), std::stoi(g.substr(c + 1)));
      }
      // health RR over the (m)TLS transport WITHOUT registering (a probe
      // must never displace a live daemon's registry entry)
      Json r = pn.gateway_request("health", Json(JsonObject{}));
      if (!r.get_or("healthy", Json(false)).as_bool())
        throw std::runtime_error("gateway unhealthy");
    } catch (const std::exception& e) {
      fprintf(stderr, "probe: %s\n", e.what());
      return 1;
    }
    printf("probe: healthy\n");
    return 0;
  }

  std::vector<std::string> files;
  DIR* d = opendir(dir.c_str());
  if (!d) {
    fprintf(stderr, "hypha-data: cannot open %s\n", dir.c_str());
    return 1;
  }
  struct dirent* ent;
  while ((ent = readdir(d)) != nullptr) {
    std::string fn = ent->d_name;
    if (fn.size() > 4 && fn[0] != '.') files.push_back(dir + "/" + fn);
  }
  closedir(d);
  std::sort(files.begin(), files.end());

  Node node(name, gw_host, gw_port, tls);
  node.set_advertise_host(advertise_host);
  node.set_listen_host(listen_host);
  node.set_exclude_cidrs(exclude_cidrs);
  node.on_stream("pull_slice", [&](const std::string& from, const Json& header,
                                   MsgSocket& sock) {
    int64_t index = header.at("index").as_int();
    if (index < 0 || index >= (int64_t)files.size()) return;
    FILE* f = fopen(files[index].c_str(), "rb");
    if (!f) return;
    fseek(f, 0, SEEK_END);
    long size = ftell(f);
    fseek(f, 0, SEEK_SET);
    Json szmsg;
    szmsg["size"] = (int64_t)size;
    sock.send_json(szmsg);
    std::vector<char> buf(1 << 20);
    long left = size;
    while (left > 0) {
      size_t chunk = std::min((long)buf.size(), left);
      if (fread(buf.data(), 1, chunk, f) != chunk) break;
      if (!sock.send_raw(buf.data(), chunk)) break;
      left -= chunk;
    }
    fclose(f);
  });
  for (const auto& g : fallback_gws) {
    auto c = g.rfind(':');
    node.add_fallback_gateway(g.substr(0, c), std::stoi(g.substr(c + 1)));
  }
  // the dataset record does not replicate between gateways: re-announce on
  // every (re)connection, incl. failover to a fallback broker
  Json rec;
  rec["num_slices"] = (int64_t)files.size();
  rec["provider"] = name;
  node.on_gateway_reconnect([&node, rec, dataset] {
    node.kv_put("dataset:" + dataset, rec);
  });
  node.start(port);
  node.kv_put("dataset:" + dataset, rec);
  printf("hypha-data %s serving %zu slices of %s on port %d\n", name.c_str(),
         files.size(), dataset.c_str(), node.port());
  fflush(stdout);
  while (true) std::this_thread::sleep_for(std::chrono::seconds(3600));
}
