// hypha-scheduler: auction-based worker allocation, lease renewal, DiLoCo
// job dispatch, data-slice scheduling and the synchronization-point FSM.
// Native redesign of /root/reference/crates/scheduler
// (bin/hypha-scheduler.rs run() :54-432, allocator.rs, worker.rs renewal
// loop :100-116, task.rs dispatch, scheduling/{batch_scheduler,
// data_scheduler}.rs, tracker/*).

#include <signal.h>
#include <unistd.h>

#include <algorithm>
#include <atomic>
#include <chrono>
#include <condition_variable>
#include <fstream>
#include <map>
#include <mutex>
#include <sstream>
#include <thread>
#include <vector>

#include <netdb.h>

#include "hypha/auction.h"
#include "hypha/batch_scheduler.h"
#include "hypha/json.h"
#include "hypha/net.h"
#include "hypha/trackers.h"

using namespace hypha;

static double now_s() {
  return std::chrono::duration<double>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

struct AllocatedWorker {
  std::string peer;
  std::string lease_id;
  std::string job_id;  // train job dispatched on this worker
  Resources resources;
  std::atomic<bool> alive{true};
};

// Minimal HTTP POST (the metrics bridge to the AIM driver,
// metrics_bridge.rs:126-146). Fire-and-forget; bridge form "host:port".
static void http_post_json(const std::string& bridge, const std::string& path,
                           const std::string& body) {
  size_t colon = bridge.rfind(':');
  if (colon == std::string::npos) return;
  int fd = tcp_connect(bridge.substr(0, colon), std::stoi(bridge.substr(colon + 1)), 2.0);
  if (fd < 0) return;
  char hdr[256];
  int n = snprintf(hdr, sizeof hdr,
                   "POST %s HTTP/1.1\r\nHost: bridge\r\nContent-Type: application/json\r\n"
                   "Content-Length: %zu\r\nConnection: close\r\n\r\n",
                   path.c_str(), body.size());
  ::send(fd, hdr, n, MSG_NOSIGNAL);
  ::send(fd, body.data(), body.size(), MSG_NOSIGNAL);
  char buf[256];
  ::recv(fd, buf, sizeof buf, 0);
  ::close(fd);
}

int main(int argc, char** argv) {
  std::string name = "scheduler", gw_host = "127.0.0.1", config_path, status_bridge;
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
    else if (a == "--config") config_path = next();
    else if (a == "--status-bridge") status_bridge = next();
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
  if (init) {  // reference CLI Init subcommand: emit a commented job config
    printf("# hypha-scheduler configuration\n"
           "# --name scheduler --gateway-host H --gateway-port P\n"
           "# --config job.json             job spec (JSON; see below)\n"
           "# --status-bridge 127.0.0.1:53800  AIM metrics forwarding\n"
           "# --tls-cert/--tls-key/--tls-ca [--tls-crl]  mTLS identity\n"
           "# --advertise-host 10.0.0.5     host peers should dial this node at\n"
           "# --listen-host 0.0.0.0         bind address (default loopback-only)\n#\n"
           "# job config (scheduler_config.rs analogue), JSON:\n"
           "{\n"
           "  \"model\": \"llama3-8b\",\n"
           "  \"dataset\": \"synth\",            # OR \"data_uri\": \"http://host/slice.safetensors\"\n"
           "  # \"preprocessor\": {\"task\": \"tokenizer|feature|image|video|auto\",\n"
           "  #   \"artifact\": {\"uri\": {\"value\": \"http://...\"}},\n"
           "  #   \"input_names\": [\"text\"], \"output_key\": \"input_ids\"},\n"
           "  \"num_workers\": 2,\n"
           "  \"update_rounds\": 100,\n"
           "  \"avg_samples_between_updates\": 1200,\n"
           "  \"batch_size\": 6,\n"
           "  \"seq_len\": 2048,\n"
           "  \"inner_lr\": 4e-4,\n"
           "  \"outer_lr\": 0.7,\n"
           "  \"outer_momentum\": 0.9,\n"
           "  \"worker_price\": 1.0,\n"
           "  \"checkpoint_every\": 0\n"
           "}\n");
    return 0;
  }
  if (probe) {  // readiness = mTLS registration round-trip (worker.rs probe)
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

  // job config (scheduler_config.rs analogue; JSON instead of TOML)
  std::ifstream cf(config_path);
  std::stringstream ss;
  ss << cf.rdbuf();
  Json cfg = Json::parse(ss.str());
  const std::string model = cfg.at("model").as_string();
  const std::string dataset = cfg.get_or("dataset", Json(std::string())).as_string();
  // URI data source (connector/mod.rs HttpHfFetcher): jobs may pull their
  // training data over HTTP(S) through the worker's fetch connector instead
  // of a cluster data node; the worker enforces its --fetch-allow list.
  const std::string data_uri = cfg.get_or("data_uri", Json(std::string())).as_string();
  if (dataset.empty() && data_uri.empty()) {
    fprintf(stderr, "[scheduler] config needs `dataset` or `data_uri`\n");
    return 1;
  }
  const int64_t num_workers = cfg.at("num_workers").as_int();
  const int64_t update_rounds = cfg.at("update_rounds").as_int();
  const int64_t samples_between = cfg.at("avg_samples_between_updates").as_int();
  const int64_t batch_size = cfg.get_or("batch_size", Json(4)).as_int();
  const int64_t max_batch = cfg.get_or("max_batch_size", Json((int64_t)600)).as_int();
  const int64_t seq_len = cfg.get_or("seq_len", Json(128)).as_int();
  const double worker_bid = cfg.get_or("worker_bid", Json(1.0)).as_double();
  const double worker_max = cfg.get_or("worker_max_price", Json(10.0)).as_double();
  // sync mode: "ps" = parameter-server star over push streams (WAN path);
  // "rccl" = scheduler-assigned {rank, world_size, rendezvous} per worker,
  // outer sync as a bucketed RCCL all-reduce over xGMI (on-node path).
  const std::string sync_mode = cfg.get_or("sync", Json(std::string("ps"))).as_string();
  const double rccl_timeout_s = cfg.get_or("rccl_timeout_s", Json(600.0)).as_double();

  Node node(name, gw_host, gw_port, tls);
  node.set_exclude_cidrs(exclude_cidrs);
  node.set_advertise_host(advertise_host);
  node.set_listen_host(listen_host);
  for (const auto& g : fallback_gws) {
    auto c = g.rfind(':');
    node.add_fallback_gateway(g.substr(0, c), std::stoi(g.substr(c + 1)));
  }

  // --- offer collection state (allocator.rs) ---
  std::mutex offer_mu;
  std::condition_variable offer_cv;
  std::vector<WorkerOffer> offers;
  node.on("worker_offer", [&](const std::string& from, const Json& body) {
    WorkerOffer o;
    o.id = body.at("id").as_string();
    o.request_id = body.at("request_id").as_string();
    o.worker = from;
    o.price = body.at("price").as_double();
    o.resources = Resources::from_json(body.at("resources"));
    o.expires_at = now_s() + body.get_or("timeout_s", Json(0.5)).as_double();
    {
      std::lock_guard<std::mutex> lk(offer_mu);
      offers.push_back(o);
    }
    offer_cv.notify_all();
    Json r;
    r["accepted"] = true;
    return r;
  });

  // --- trackers + FSM ---
  SliceTracker slices(dataset, 1);  // real count filled after discovery
  std::string data_provider = "data";
  BatchScheduler fsm(samples_between * num_workers, update_rounds);
  std::mutex fsm_mu;
  std::map<std::string, std::string> job_status;  // job id -> status
  std::mutex status_mu;
  std::condition_variable status_cv;

  node.on("data", [&](const std::string& from, const Json& body) {
    std::lock_guard<std::mutex> lk(fsm_mu);
    auto a = slices.next(from);
    Json r;
    r["data_provider"] = data_provider;
    r["index"] = (int64_t)a.index;
    r["epoch"] = (int64_t)a.epoch;
    return r;
  });

  node.on("progress", [&](const std::string& from, const Json& body) {
    std::string kind = body.get_or("kind", Json("status")).as_string();
    Progress p{};
    if (kind == "status") {
      p.kind = Progress::Status;
      p.batch_size = body.get_or("batch_size", Json((int64_t)1)).as_int();
    } else if (kind == "metrics") {
      p.kind = Progress::Metrics;
    } else if (kind == "update") {
      p.kind = Progress::Update;
    } else if (kind == "updated") {
      p.kind = Progress::Updated;
    } else if (kind == "update-received") {
      p.kind = Progress::UpdateReceived;
    }
    ProgressResponse r;
    {
      std::lock_guard<std::mutex> lk(fsm_mu);
      r = fsm.handle(from, p);
    }
    const char* names[] = {"ok", "continue", "schedule-update", "done", "error"};
    Json out;
    out["kind"] = names[(int)r.kind];
    out["counter"] = r.counter;
    if (kind == "metrics" && body.has("metrics")) {
      // metrics bridge: log + forward to the AIM driver (drivers/aim_driver.py)
      fprintf(stderr, "[metrics] %s %s\n", from.c_str(), body.at("metrics").dump().c_str());
      if (!status_bridge.empty()) {
        for (auto& [mname, mval] : body.at("metrics").as_object()) {
          Json post;
          post["worker_id"] = from;
          post["round"] = body.get_or("round", Json((int64_t)0));
          post["metric_name"] = mname;
          post["value"] = mval;
          http_post_json(status_bridge, "/status", post.dump());
        }
      }
    }
    return out;
  });

  node.on("job_status", [&](const std::string& from, const Json& body) {
    {
      std::lock_guard<std::mutex> lk(status_mu);
      job_status[body.at("id").as_string()] = body.at("status").as_string();
    }
    status_cv.notify_all();
    Json r;
    r["ok"] = true;
    return r;
  });

  node.start(port);
  fprintf(stderr, "[scheduler] up on port %d\n", node.port());

  // --- discover the data provider via the registry (DHT get, kad.rs) ---
  int64_t num_slices = 0;
  for (int i = 0; i < 100 && data_uri.empty(); ++i) {
    auto rec = node.kv_get("dataset:" + dataset);
    if (rec) {
      num_slices = rec->at("num_slices").as_int();
      data_provider = rec->get_or("provider", Json(std::string("data"))).as_string();
      break;
    }
    usleep(200000);
  }
  if (num_slices == 0 && data_uri.empty()) {
    fprintf(stderr, "[scheduler] dataset %s not found\n", dataset.c_str());
    return 1;
  }
  {
    std::lock_guard<std::mutex> lk(fsm_mu);
    slices = SliceTracker(dataset, (int)num_slices);
  }

  // --- allocation + immediate lease renewal -----------------------------
  std::atomic<bool> running{true};
  std::vector<std::thread> renewers;
  std::atomic<long> rq_seq{0};
  // lease renewal at 2/3 of the 10 s timeout (worker.rs:100-116); the first
  // renewal fires immediately to upgrade the 500 ms offer lease
  auto start_renewal = [&](std::shared_ptr<AllocatedWorker> w) {
    renewers.emplace_back([&, w] {
      while (running && w->alive) {
        Json b;
        b["id"] = w->lease_id;
        try {
          Json r = node.request(w->peer, "renew_lease", b, 5.0);
          if (!r.get_or("granted", Json(false)).as_bool()) {
            w->alive = false;
            break;
          }
        } catch (...) {
          w->alive = false;
          break;
        }
        for (int i = 0; i < 66 && running; ++i)
          std::this_thread::sleep_for(std::chrono::milliseconds(100));
      }
      if (!w->alive) fprintf(stderr, "[scheduler] lost worker %s\n", w->peer.c_str());
    });
  };

  auto allocate = [&](size_t count, const std::vector<std::string>& execs,
                      Resources req) -> std::vector<WorkerOffer> {
    GreedyOfferAggregator agg(count, PriceRange{worker_bid, worker_max},
                              now_s() + 5.0);  // 5 s deadline (allocator.rs:25)
    std::string rqid = name + "-rq-" + std::to_string(rq_seq.fetch_add(1));
    {
      std::lock_guard<std::mutex> lk(offer_mu);
      offers.clear();
    }
    Json ad;
    ad["id"] = rqid;
    ad["resources"] = req.to_json();
    JsonArray ex;
    for (auto& e : execs) ex.push_back(Json(e));
    ad["executors"] = ex;
    ad["bid"] = worker_bid;
    ad["timeout_s"] = 5.0;
    node.publish("hypha/worker", ad);
    size_t cursor = 0;
    double last_pub = now_s();
    std::unique_lock<std::mutex> lk(offer_mu);
    while (now_s() < agg.deadline()) {
      offer_cv.wait_for(lk, std::chrono::milliseconds(100));
      bool full = false;
      while (cursor < offers.size()) {
        if (offers[cursor].request_id == rqid) full = agg.add(offers[cursor], now_s());
        ++cursor;
      }
      if (full) break;
      if (now_s() - last_pub > 1.0) {  // republish: capacity may have freed up
        last_pub = now_s();
        lk.unlock();
        node.publish("hypha/worker", ad);
        lk.lock();
      }
    }
    return agg.finalize();
  };

  Resources worker_req = Resources::from_json(
      cfg.get_or("worker_resources", Json::parse("{\"gpu\":1,\"cpu\":1,\"memory\":1,\"storage\":1}")));

  // --- inference mode (job_type: "generate"): allocate inference workers,
  // dispatch generation jobs over the same bridge contract, wait for all
  // completions. No PS, no sync FSM — slices still flow through the data
  // scheduler. (The reference reaches inference through the same executor
  // mechanism; here it is a first-class scheduler mode.)
  if (cfg.get_or("job_type", Json(std::string("diloco"))).as_string() == "generate") {
    auto ioffers = allocate(num_workers, {"inference-transformer"}, worker_req);
    if ((int64_t)ioffers.size() < num_workers) {
      fprintf(stderr, "[scheduler] insufficient inference workers: %zu/%lld\n",
              ioffers.size(), (long long)num_workers);
      return 1;
    }
    std::vector<std::shared_ptr<AllocatedWorker>> iworkers;
    for (auto& o : ioffers) {
      auto w = std::make_shared<AllocatedWorker>();
      w->peer = o.worker;
      w->lease_id = o.id;
      w->resources = o.resources;
      start_renewal(w);
      iworkers.push_back(w);
    }
    for (size_t i = 0; i < iworkers.size(); ++i) {
      Json inf;
      inf["model"] = model;
      Json fetch;
      if (!data_uri.empty()) {
        Json uref;
        uref["value"] = data_uri;
        fetch["uri"] = uref;
      } else {
        Json sref;
        sref["peer"] = name;
        sref["dataset"] = dataset;
        fetch["scheduler"] = sref;
      }
      inf["data"] = fetch;
      inf["batch_size"] = batch_size;
      inf["seq_len"] = seq_len;
      inf["max_new_tokens"] = cfg.get_or("max_new_tokens", Json((int64_t)16));
      inf["num_batches"] = cfg.get_or("num_batches", Json((int64_t)2));
      if (cfg.has("temperature")) inf["temperature"] = cfg.at("temperature");
      if (cfg.has("top_k")) inf["top_k"] = cfg.at("top_k");
      Json ex;
      ex["infer"] = inf;
      Json job;
      job["id"] = std::string("job-infer-") + std::to_string(i);
      job["executor"] = ex;
      Json d;
      d["job"] = job;
      d["lease"] = iworkers[i]->lease_id;
      node.request(iworkers[i]->peer, "dispatch_job", d, 10.0);
    }
    fprintf(stderr, "[scheduler] dispatched %lld inference jobs\n",
            (long long)num_workers);
    {
      std::unique_lock<std::mutex> lk(status_mu);
      while (true) {
        int64_t done = 0, failed = 0;
        for (auto& [id, st] : job_status) {
          if (st == "completed") ++done;
          if (st == "failed" || st == "cancelled") ++failed;
        }
        if (done >= num_workers) break;
        bool lost = false;
        for (auto& w : iworkers) lost = lost || !w->alive;
        if (lost || failed > 0) {
          lk.unlock();
          running = false;
          for (auto& t : renewers) t.join();
          fprintf(stderr, "[scheduler] inference worker lost/failed: aborting\n");
          node.stop();
          return 1;
        }
        status_cv.wait_for(lk, std::chrono::milliseconds(500));
      }
    }
    printf("Job is completed.\n");
    fflush(stdout);
    running = false;
    for (auto& t : renewers) t.join();
    node.stop();
    return 0;
  }

  auto train_offers = allocate(num_workers, {"diloco-transformer"}, worker_req);
  if ((int64_t)train_offers.size() < num_workers) {
    fprintf(stderr, "[scheduler] insufficient workers: %zu/%lld\n", train_offers.size(),
            (long long)num_workers);
    return 1;
  }
  auto shutdown_fail = [&](const char* why) {
    fprintf(stderr, "[scheduler] %s\n", why);
    running = false;
    for (auto& t : renewers) t.join();
    node.stop();
    return 1;
  };
  std::vector<std::shared_ptr<AllocatedWorker>> train_workers;
  for (auto& o : train_offers) {
    auto w = std::make_shared<AllocatedWorker>();
    w->peer = o.worker;
    w->lease_id = o.id;
    w->resources = o.resources;
    start_renewal(w);
    train_workers.push_back(w);
  }
  // ephemeral rendezvous ports for the RCCL group (fresh one per formation)
  auto pick_port = [&]() {
    int fd = tcp_listen(0);
    int p = listen_port(fd);
    ::close(fd);
    return p;
  };

  std::string ps_peer;
  std::shared_ptr<AllocatedWorker> psw;
  std::string ps_job_id = "job-ps";
  if (sync_mode != "rccl") {
  auto ps_offers = allocate(1, {"parameter-server"}, worker_req);
  if (ps_offers.empty()) return shutdown_fail("no parameter server offer");
  ps_peer = ps_offers[0].worker;

  // (train workers already renewing)
  psw = std::make_shared<AllocatedWorker>();
  psw->peer = ps_peer;
  psw->lease_id = ps_offers[0].id;
  start_renewal(psw);
  // give renewers a moment to upgrade the offer leases
  std::this_thread::sleep_for(std::chrono::milliseconds(200));

  // --- dispatch jobs (task.rs / hypha-scheduler.rs:328-370) ---
  JsonArray train_peer_names;
  for (auto& w : train_workers) train_peer_names.push_back(Json(w->peer));

  // the aggregate job first, so the PS is listening before updates flow
  {
    Json agg_cfg;
    agg_cfg["num_workers"] = num_workers;
    Json nesterov;
    nesterov["learning_rate"] = cfg.get_or("outer_lr", Json(0.7)).as_double();
    nesterov["momentum"] = cfg.get_or("outer_momentum", Json(0.9)).as_double();
    Json opt;
    opt["nesterov"] = nesterov;
    agg_cfg["optimizer"] = opt;
    if (cfg.has("weighted_aggregation"))
      agg_cfg["weighted_aggregation"] = cfg.at("weighted_aggregation");
    Json results;
    Json peers;
    peers["peers"] = train_peer_names;
    peers["strategy"] = std::string("all");
    results["peers"] = peers;
    agg_cfg["results"] = results;
    // PS broadcasts carry the PS job id; workers route them to their
    // unique local train job (see hypha_worker.cpp receive_resource)
    Json ex;
    ex["aggregate"] = agg_cfg;
    Json job;
    job["id"] = ps_job_id;
    job["executor"] = ex;
    Json d;
    d["job"] = job;
    d["lease"] = psw->lease_id;
    node.request(ps_peer, "dispatch_job", d, 10.0);
  }
  }  // sync_mode != "rccl"

  // dispatch a train job; rank >= 0 attaches the RCCL rendezvous assignment
  // (rank/world_size/master addr+port) the executor forms its communicator
  // from — the control plane bootstrapping the data plane
  auto dispatch_train = [&](std::shared_ptr<AllocatedWorker> w, const std::string& jid,
                            bool join, int rank = -1, int world = 0, int rdv_port = 0) {
    Json tr;
    tr["model"] = model;
    Json fetch;
    if (!data_uri.empty()) {
      Json uref;
      uref["value"] = data_uri;
      fetch["uri"] = uref;
    } else {
      Json sref;
      sref["peer"] = name;
      sref["dataset"] = dataset;
      fetch["scheduler"] = sref;
    }
    tr["data"] = fetch;
    if (cfg.has("preprocessor")) tr["preprocessor"] = cfg.at("preprocessor");
    if (rank >= 0) {
      Json rc;
      rc["rank"] = (int64_t)rank;
      rc["world_size"] = (int64_t)world;
      rc["master_addr"] = std::string("127.0.0.1");
      rc["master_port"] = (int64_t)rdv_port;
      rc["timeout_s"] = rccl_timeout_s;
      tr["rccl"] = rc;
    } else {
      Json updates;
      Json upeers;
      upeers["peers"] = JsonArray{Json(ps_peer)};
      upeers["strategy"] = std::string("all");
      updates["peers"] = upeers;
      updates["job"] = ps_job_id;  // tag pushes for the PS job
      tr["updates"] = updates;
    }
    Json adam;
    adam["learning_rate"] = cfg.get_or("inner_lr", Json(4e-4)).as_double();
    Json opt;
    opt["adam"] = adam;
    tr["optimizer"] = opt;
    // per-worker batch scales with OFFERED gpu capacity, capped
    // (hypha-scheduler.rs:320-322: floor(gpu_avail/gpu_req).min(max_batch))
    int64_t mult = worker_req.gpu > 0 ? (int64_t)(w->resources.gpu / worker_req.gpu) : 1;
    if (mult < 1) mult = 1;
    int64_t wbatch = std::min<int64_t>(max_batch, batch_size * mult);
    tr["batch_size"] = wbatch;
    tr["seq_len"] = seq_len;
    if (cfg.has("checkpoint_every_rounds"))
      tr["checkpoint_every_rounds"] = cfg.at("checkpoint_every_rounds");
    if (cfg.has("checkpoint_dir")) tr["checkpoint_dir"] = cfg.at("checkpoint_dir");
    if (join) tr["join"] = true;
    Json ex;
    ex["train"] = tr;
    Json job;
    job["id"] = jid;
    job["executor"] = ex;
    Json d;
    d["job"] = job;
    d["lease"] = w->lease_id;
    node.request(w->peer, "dispatch_job", d, 10.0);
    w->job_id = jid;
    std::lock_guard<std::mutex> lk(fsm_mu);
    fsm.add_worker(w->peer, wbatch);
  };

  int rdv_port = sync_mode == "rccl" ? pick_port() : 0;
  for (size_t i = 0; i < train_workers.size(); ++i) {
    if (sync_mode == "rccl")
      dispatch_train(train_workers[i], std::string("job-train-") + std::to_string(i),
                     false, (int)i, (int)train_workers.size(), rdv_port);
    else
      dispatch_train(train_workers[i], std::string("job-train-") + std::to_string(i),
                     false);
  }
  fprintf(stderr, "[scheduler] dispatched %lld train jobs (%s sync)%s\n",
          (long long)num_workers, sync_mode.c_str(),
          sync_mode == "rccl" ? "" : " + 1 aggregate");

  // --- fault tolerance: detect lost workers (lease renewal failure) and
  //     allocate + join a replacement (BASELINE config 3; the reference only
  //     detects the loss — rejoin is this framework's addition) ---
  std::mutex tw_mu;
  std::atomic<long long> live_workers{num_workers};
  std::atomic<int> replace_seq{0};
  std::thread monitor([&] {
    std::set<std::string> handled;
    while (running) {
      std::this_thread::sleep_for(std::chrono::milliseconds(300));
      std::vector<std::shared_ptr<AllocatedWorker>> lost;
      {
        std::lock_guard<std::mutex> lk(tw_mu);
        for (auto& w : train_workers)
          if (!w->alive && !handled.count(w->peer)) {
            handled.insert(w->peer);
            lost.push_back(w);
          }
      }
      for (auto& w : lost) {
        {
          std::lock_guard<std::mutex> lk(fsm_mu);
          if (fsm.finished()) break;  // training over: no point replacing
        }
        fprintf(stderr, "[scheduler] worker %s lost: shrinking round to %lld\n",
                w->peer.c_str(), (long long)(live_workers - 1));
        live_workers -= 1;
        {
          std::lock_guard<std::mutex> lk(fsm_mu);
          fsm.remove_worker(w->peer);
          slices.remove_worker(w->peer);
        }
        if (sync_mode == "rccl") {
          // RCCL elastic path: drop the dead rank, optionally admit a
          // replacement, and re-form the communicator on a fresh rendezvous.
          // Survivors keep ranks 0..n-1 (rank 0 always a survivor: it seeds
          // the post-reform state broadcast); a joiner takes the last rank.
          std::vector<std::shared_ptr<AllocatedWorker>> survivors;
          {
            std::lock_guard<std::mutex> lk(tw_mu);
            train_workers.erase(
                std::remove_if(train_workers.begin(), train_workers.end(),
                               [&](auto& x) { return x->peer == w->peer; }),
                train_workers.end());
            survivors = train_workers;
          }
          std::shared_ptr<AllocatedWorker> joiner;
          auto offers2 = allocate(1, {"diloco-transformer"}, worker_req);
          if (!offers2.empty()) {
            joiner = std::make_shared<AllocatedWorker>();
            joiner->peer = offers2[0].worker;
            joiner->lease_id = offers2[0].id;
            start_renewal(joiner);
            std::this_thread::sleep_for(std::chrono::milliseconds(200));
          } else {
            fprintf(stderr, "[scheduler] no replacement worker available\n");
          }
          int port2 = pick_port();
          int world = (int)survivors.size() + (joiner ? 1 : 0);
          if (joiner) {
            std::string jid = "job-train-r" + std::to_string(replace_seq.fetch_add(1));
            try {
              dispatch_train(joiner, jid, true, (int)survivors.size(), world, port2);
              live_workers += 1;
              {
                std::lock_guard<std::mutex> lk(tw_mu);
                train_workers.push_back(joiner);
              }
              fprintf(stderr, "[scheduler] replacement %s joins as rank %zu\n",
                      joiner->peer.c_str(), survivors.size());
            } catch (const std::exception& e) {
              fprintf(stderr, "[scheduler] replacement dispatch failed: %s\n",
                      e.what());
              joiner.reset();
              world = (int)survivors.size();
            }
          }
          for (size_t i = 0; i < survivors.size(); ++i) {
            Json rf;
            rf["job"] = survivors[i]->job_id;
            rf["rank"] = (int64_t)i;
            rf["world_size"] = (int64_t)world;
            rf["master_addr"] = std::string("127.0.0.1");
            rf["master_port"] = (int64_t)port2;
            rf["timeout_s"] = rccl_timeout_s;
            try {
              node.request(survivors[i]->peer, "rccl_reform", rf, 10.0);
            } catch (const std::exception& e) {
              fprintf(stderr, "[scheduler] reform to %s failed: %s\n",
                      survivors[i]->peer.c_str(), e.what());
            }
          }
          fprintf(stderr, "[scheduler] re-formed RCCL group: world=%d port=%d\n",
                  world, port2);
          continue;
        }
        Json pc;
        pc["job"] = ps_job_id;
        pc["remove_peer"] = w->peer;
        try {
          node.request(ps_peer, "ps_control", pc, 10.0);
        } catch (...) {
        }
        // replacement: re-run the auction for one worker
        auto offers2 = allocate(1, {"diloco-transformer"}, worker_req);
        if (offers2.empty()) {
          fprintf(stderr, "[scheduler] no replacement worker available\n");
          continue;
        }
        auto nw = std::make_shared<AllocatedWorker>();
        nw->peer = offers2[0].worker;
        nw->lease_id = offers2[0].id;
        start_renewal(nw);
        std::this_thread::sleep_for(std::chrono::milliseconds(200));
        std::string jid = "job-train-r" + std::to_string(replace_seq.fetch_add(1));
        try {
          // order matters: the PS must require the joiner's update BEFORE the
          // joiner can start pushing, or a round can close without it
          Json pc1;
          pc1["job"] = ps_job_id;
          pc1["add_member"] = nw->peer;
          node.request(ps_peer, "ps_control", pc1, 10.0);
          dispatch_train(nw, jid, true);  // join: catch up via PS offset
          Json pc2;
          pc2["job"] = ps_job_id;
          pc2["sync_to"] = nw->peer;  // sends the cumulative offset
          node.request(ps_peer, "ps_control", pc2, 10.0);
          live_workers += 1;
          {
            std::lock_guard<std::mutex> lk(tw_mu);
            train_workers.push_back(nw);
          }
          fprintf(stderr, "[scheduler] replacement %s joined as %s\n",
                  nw->peer.c_str(), jid.c_str());
        } catch (const std::exception& e) {
          // benign when training finished while the replacement was joining
          fprintf(stderr, "[scheduler] replacement dispatch failed: %s\n", e.what());
        }
      }
    }
  });

  // --- wait for completion ---
  {
    // done when every live train job completed, or (kill/rejoin edge: a
    // replacement still catching up when the final round closed) the FSM is
    // finished and at least one worker completed
    std::unique_lock<std::mutex> lk(status_mu);
    while (true) {
      long long completed = 0;
      for (auto& [id, st] : job_status)
        if (st == "completed" && id != ps_job_id) ++completed;
      if (completed >= live_workers) break;
      bool fsm_done;
      {
        std::lock_guard<std::mutex> flk(fsm_mu);
        fsm_done = fsm.finished();
      }
      if (fsm_done && completed >= 1) {
        // brief grace period for stragglers, then finish
        status_cv.wait_for(lk, std::chrono::seconds(3));
        break;
      }
      if (psw && !psw->alive && !fsm_done) {
        // the aggregator is a single point per job (as in the reference):
        // its loss mid-run is fatal — fail fast instead of hanging until
        // executor timeouts
        lk.unlock();
        running = false;
        monitor.join();
        return shutdown_fail("parameter server lost mid-run: aborting job");
      }
      status_cv.wait_for(lk, std::chrono::milliseconds(500));
    }
  }
  {
    std::lock_guard<std::mutex> lk(fsm_mu);
    fprintf(stderr, "[scheduler] rounds completed: %lld\n", (long long)fsm.round());
  }
  printf("Job is completed.\n");
  fflush(stdout);
  running = false;
  monitor.join();
  for (auto& t : renewers) t.join();
  node.stop();
  return 0;
}
