// hypha-worker: the worker-node daemon.
// Native redesign of /root/reference/crates/worker (bin/hypha-worker.rs,
// arbiter.rs, lease_manager.rs, job_manager.rs, executor/{process,bridge,
// parameter_server}.rs, connector/mod.rs): sells capacity in the gossip
// auction, maintains lease-based fault tolerance (10 s TTL, 250 ms prune,
// orphan-job cancellation), runs Train jobs as bridge-isolated subprocesses
// and Aggregate jobs as the built-in parameter-server executor (safetensors
// average + Nesterov on CPU — the WAN path; on-node GPU workers use RCCL).

#include <signal.h>
#include <sys/stat.h>
#include <sys/types.h>
#include <sys/wait.h>
#include <unistd.h>

#include <algorithm>
#include <atomic>
#include <chrono>
#include <condition_variable>
#include <cstring>
#include <deque>
#include <map>
#include <mutex>
#include <random>
#include <set>
#include <thread>
#include <vector>

#include "hypha/auction.h"
#include "hypha/bridge.h"
#include "hypha/gateway.h"
#include "hypha/http_client.h"
#include "hypha/json.h"
#include "hypha/leases.h"
#include "hypha/net.h"
#include "hypha/ps_math.h"
#include "hypha/safetensors.h"

using namespace hypha;

static double now_s() {
  return std::chrono::duration<double>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

struct LeaseInfo {
  std::string scheduler;
  Resources resources;
  std::string job_id;  // empty until a job is dispatched on this lease
};

struct Job {
  std::string id;
  std::string lease_id;
  std::string scheduler;
  std::string work_dir;
  int gpu_id = -1;  // device pinned for this job (-1 = none)
  pid_t pid = -1;
  std::shared_ptr<Bridge> bridge;
  std::thread runner;
  std::atomic<bool> cancelled{false};
  // aggregate executor state
  bool is_aggregate = false;
  std::mutex agg_mu;
  std::condition_variable agg_cv;
  std::set<std::string> members;                // peers whose update closes a round
  std::map<std::string, std::pair<std::string, double>> round_files;  // peer -> (file, samples)
  std::vector<std::string> pending_sync_peers;  // joiners awaiting the offset
};

struct WorkerDaemon {
  std::string name;
  Node node;
  StaticResourceManager resman;
  OfferPolicy policy;
  std::string exec_cmd;  // template with {SOCKET_PATH} {WORK_DIR} {JOB_JSON}
  std::string infer_cmd;  // same template, launched for "infer" executors
  std::string work_root;

  std::mutex mu;
  Ledger<LeaseInfo> leases;
  std::map<std::string, std::shared_ptr<Job>> jobs;  // job id -> job
  struct BufferedAd {
    std::string from;
    Json ad;
    double expires_at;
  };
  std::deque<BufferedAd> ad_buffer;
  std::atomic<bool> running{true};
  std::atomic<long> lease_seq{0};
  // Versioned parameter KV (messages lib.rs:698-739 parameter_pull/push —
  // declared in the reference protocol; here a working store):
  // "<job>/<key>" -> version -> blob. Monotonic versions, latest by default.
  std::mutex kv_mu;
  std::map<std::string, std::map<uint64_t, std::string>> param_store;
  // URI/HF fetch connector policy (connector/mod.rs:226-302): hosts a job
  // may fetch from (empty = deny, closing the reference's allow-list TODO)
  // and the HuggingFace endpoint resolve URLs are built against.
  std::vector<std::string> fetch_allow;
  std::string hf_endpoint = "https://huggingface.co";
  // GPU pool: device indices this daemon owns (--gpu-ids). A dispatched
  // process job is pinned to one free device via HIP_VISIBLE_DEVICES so each
  // worker peer IS one GPU (the 8-GPUs-on-one-node deployment: 8 daemons or
  // one daemon with 8 ids). Guarded by `mu`.
  std::vector<int> gpu_pool;

  int acquire_gpu() {
    std::lock_guard<std::mutex> lk(mu);
    if (gpu_pool.empty()) return -1;
    int id = gpu_pool.back();
    gpu_pool.pop_back();
    return id;
  }
  void release_gpu(int id) {
    if (id < 0) return;
    std::lock_guard<std::mutex> lk(mu);
    gpu_pool.push_back(id);
  }

  WorkerDaemon(std::string nm, const std::string& gw_host, int gw_port, Resources total,
               OfferPolicy pol, std::string cmd, std::string root, TlsConfig tls = {})
      : name(std::move(nm)),
        node(name, gw_host, gw_port, std::move(tls)),
        resman(total),
        policy(std::move(pol)),
        exec_cmd(std::move(cmd)),
        work_root(std::move(root)) {}

  void start(int port) {
    node.on("health", [&](const std::string&, const Json&) {
      Json r;
      r["healthy"] = true;
      return r;
    });
    node.on("renew_lease", [&](const std::string& from, const Json& body) {
      std::lock_guard<std::mutex> lk(mu);
      std::string id = body.at("id").as_string();
      auto l = leases.get(id);
      Json r;
      if (!l || l->leasable.scheduler != from) {  // owner-validated (arbiter.rs:155-199)
        r["granted"] = false;
      } else {
        r["granted"] = leases.renew(id, 10.0);  // lease TTL 10 s (arbiter.rs:29)
      }
      return r;
    });
    node.on("dispatch_job", [&](const std::string& from, const Json& body) {
      return dispatch_job(from, body);
    });
    // PS membership control: worker kill/rejoin (BASELINE config 3).
    // The aggregate job closes a round when EVERY current member has pushed
    // (set semantics, not a count — immune to membership/round races).
    // {job, remove_peer?} | {job, sync_to?} (sync_to also adds the member).
    node.on("ps_control", [&](const std::string& from, const Json& body) {
      std::shared_ptr<Job> job;
      {
        std::lock_guard<std::mutex> lk(mu);
        auto it = jobs.find(body.at("job").as_string());
        if (it == jobs.end()) throw std::runtime_error("no such job");
        job = it->second;
      }
      if (!job->is_aggregate) throw std::runtime_error("not an aggregate job");
      {
        std::lock_guard<std::mutex> lk(job->agg_mu);
        if (body.has("add_member")) job->members.insert(body.at("add_member").as_string());
        if (body.has("sync_to")) {
          job->pending_sync_peers.push_back(body.at("sync_to").as_string());
          job->members.insert(body.at("sync_to").as_string());
        }
        if (body.has("remove_peer")) job->members.erase(body.at("remove_peer").as_string());
      }
      job->agg_cv.notify_all();
      Json r;
      r["ok"] = true;
      return r;
    });
    // Scheduler-driven RCCL membership change: forward to the executor as a
    // bridge SSE `reform` event; it re-forms the communicator at its next
    // outer-sync boundary (elastic DiLoCo on the RCCL data plane).
    node.on("rccl_reform", [&](const std::string& from, const Json& body) {
      std::shared_ptr<Job> job;
      {
        std::lock_guard<std::mutex> lk(mu);
        auto it = jobs.find(body.at("job").as_string());
        if (it == jobs.end()) throw std::runtime_error("no such job");
        job = it->second;
      }
      if (job->scheduler != from) throw std::runtime_error("not the job owner");
      if (!job->bridge) throw std::runtime_error("job has no bridge");
      Json ev;
      ev["kind"] = std::string("reform");
      ev["rank"] = body.at("rank");
      ev["world_size"] = body.at("world_size");
      ev["master_addr"] = body.get_or("master_addr", Json(std::string("127.0.0.1")));
      ev["master_port"] = body.at("master_port");
      if (body.has("timeout_s")) ev["timeout_s"] = body.at("timeout_s");
      job->bridge->push_event(ev);
      Json r;
      r["ok"] = true;
      return r;
    });
    node.on_stream("push_resource",
                   [&](const std::string& from, const Json& header, MsgSocket& sock) {
                     receive_resource(from, header, sock);
                   });
    // parameter_push (lib.rs:720-739): header {job, key, version?, size} +
    // payload bytes -> ack {ok, version}. Version defaults to latest+1.
    node.on_stream("param_push",
                   [&](const std::string&, const Json& header, MsgSocket& sock) {
                     int64_t size = header.at("size").as_int();
                     // bound the allocation like recv_json's 256MB frame cap:
                     // a single hostile header must not exhaust worker memory
                     if (size < 0 || size > (int64_t)(256u << 20)) {
                       Json r;
                       r["ok"] = false;
                       r["error"] = "param_push size exceeds cap";
                       sock.send_json(r);
                       return;
                     }
                     std::string data((size_t)size, '\0');
                     if (size > 0 && !sock.recv_raw(data.data(), (size_t)size)) return;
                     Json r;
                     {
                       std::lock_guard<std::mutex> lk(kv_mu);
                       auto& versions = param_store[header.at("job").as_string() + "/" +
                                                    header.at("key").as_string()];
                       uint64_t v = header.has("version")
                                        ? (uint64_t)header.at("version").as_int()
                                        : (versions.empty() ? 1
                                                            : versions.rbegin()->first + 1);
                       versions[v] = std::move(data);
                       r["ok"] = true;
                       r["version"] = (int64_t)v;
                     }
                     sock.send_json(r);
                   });
    // parameter_pull (lib.rs:698-717): header {job, key, version?} ->
    // {found, version, size} + payload (NotFound -> found=false).
    node.on_stream("param_pull",
                   [&](const std::string&, const Json& header, MsgSocket& sock) {
                     std::string blob;
                     Json r;
                     r["found"] = false;
                     {
                       std::lock_guard<std::mutex> lk(kv_mu);
                       auto it = param_store.find(header.at("job").as_string() + "/" +
                                                  header.at("key").as_string());
                       if (it != param_store.end() && !it->second.empty()) {
                         if (header.has("version")) {
                           auto vit =
                               it->second.find((uint64_t)header.at("version").as_int());
                           if (vit != it->second.end()) {
                             r["found"] = true;
                             r["version"] = (int64_t)vit->first;
                             blob = vit->second;
                           }
                         } else {
                           auto& last = *it->second.rbegin();
                           r["found"] = true;
                           r["version"] = (int64_t)last.first;
                           blob = last.second;
                         }
                       }
                     }
                     r["size"] = (int64_t)blob.size();
                     if (!sock.send_json(r)) return;
                     if (!blob.empty()) sock.send_raw(blob.data(), blob.size());
                   });
    node.start(port);
    node.subscribe("hypha/worker", [&](const std::string& from, const Json& ad) {
      double ttl = ad.get_or("timeout_s", Json(5.0)).as_double();
      std::lock_guard<std::mutex> lk(mu);
      ad_buffer.push_back(BufferedAd{from, ad, now_s() + ttl});
    });
    std::thread([this] {
      while (running) {
        try {
          arbiter_loop();
        } catch (const std::exception& e) {
          fprintf(stderr, "[%s] arbiter error: %s\n", name.c_str(), e.what());
        }
      }
    }).detach();
    std::thread([this] {
      while (running) {
        try {
          prune_loop();
        } catch (const std::exception& e) {
          fprintf(stderr, "[%s] pruner error: %s\n", name.c_str(), e.what());
        }
      }
    }).detach();
  }

  // --- dRAP sell side (arbiter.rs:88-437) --------------------------------

  void arbiter_loop() {
    while (running) {
      std::this_thread::sleep_for(std::chrono::milliseconds(200));  // ad batch window
      std::vector<BufferedAd> batch;
      {
        std::lock_guard<std::mutex> lk(mu);
        while (!ad_buffer.empty() && batch.size() < 100) {
          if (ad_buffer.front().expires_at > now_s()) batch.push_back(ad_buffer.front());
          ad_buffer.pop_front();
        }
      }
      if (batch.empty()) continue;
      std::vector<WorkerRequest> ads;
      std::map<std::string, BufferedAd*> by_id;
      for (auto& b : batch) {
        const Json& j = b.ad;
        WorkerRequest rq;
        rq.id = j.at("id").as_string();
        rq.scheduler = b.from;
        rq.resources = Resources::from_json(j.at("resources"));
        Json execs = j.get_or("executors", Json(JsonArray{}));  // bind: get_or returns by value
        for (auto& e : execs.as_array()) rq.executors.push_back(e.as_string());
        rq.bid = j.get_or("bid", Json(0.0)).as_double();
        ads.push_back(rq);
        by_id[rq.id] = &b;
      }
      auto picked = select_requests(ads, policy, resman.available());
      std::set<std::string> answered;
      for (auto& d : picked) {
        // "whole" strategy: offer (and reserve) the full remaining capacity
        Resources offer_res = policy.strategy == "whole" ? resman.available()
                                                         : d.request.resources;
        if (!offer_res.fits_in(resman.available()) ||
            !d.request.resources.fits_in(offer_res))
          continue;
        if (!resman.reserve(offer_res)) continue;
        answered.insert(d.request.id);
        std::string lease_id =
            name + "-lease-" + std::to_string(lease_seq.fetch_add(1));
        {
          std::lock_guard<std::mutex> lk(mu);
          // temporary offer lease: 500 ms TTL (arbiter.rs:383-435)
          leases.insert(lease_id, LeaseInfo{d.request.scheduler, offer_res, ""},
                        0.5);
        }
        Json offer;
        offer["id"] = lease_id;
        offer["request_id"] = d.request.id;
        offer["price"] = d.offer_price;
        offer["resources"] = offer_res.to_json();
        offer["timeout_s"] = 0.5;
        try {
          node.request(d.request.scheduler, "worker_offer", offer, 5.0);
        } catch (const std::exception& e) {
          fprintf(stderr, "[%s] offer to %s failed: %s\n", name.c_str(),
                  d.request.scheduler.c_str(), e.what());
        }
      }
      // requeue unanswered, unexpired ads (resources may free up before the
      // scheduler's offer deadline passes)
      {
        std::lock_guard<std::mutex> lk(mu);
        for (auto& b : batch) {
          std::string id = b.ad.at("id").as_string();
          if (!answered.count(id) && b.expires_at > now_s()) ad_buffer.push_back(b);
        }
      }
    }
  }

  // --- lease pruning (arbiter.rs:98-141, 250 ms tick) --------------------

  void prune_loop() {
    while (running) {
      std::this_thread::sleep_for(std::chrono::milliseconds(250));
      std::vector<Ledger<LeaseInfo>::Lease> expired;
      {
        std::lock_guard<std::mutex> lk(mu);
        expired = leases.drain_expired();
      }
      for (auto& l : expired) {
        resman.release(l.leasable.resources);
        if (!l.leasable.job_id.empty()) cancel_job(l.leasable.job_id);
      }
    }
  }

  void cancel_job(const std::string& job_id) {
    std::shared_ptr<Job> job;
    {
      std::lock_guard<std::mutex> lk(mu);
      auto it = jobs.find(job_id);
      if (it == jobs.end()) return;
      job = it->second;
    }
    job->cancelled = true;
    if (job->pid > 0) {
      kill(job->pid, SIGTERM);
      // SIGKILL escalation after 5 s (process.rs:146-187)
      std::thread([job] {
        std::this_thread::sleep_for(std::chrono::seconds(5));
        pid_t p = job->pid;
        if (p > 0) kill(p, SIGKILL);
      }).detach();
    }
    job->agg_cv.notify_all();
    fprintf(stderr, "[%s] cancelled job %s (lease expired)\n", name.c_str(),
            job_id.c_str());
  }

  // --- job dispatch (job_manager.rs:85-158) ------------------------------

  Json dispatch_job(const std::string& from, const Json& body) {
    std::string lease_id = body.at("lease").as_string();
    Json jobj = body.at("job");
    std::string job_id = jobj.at("id").as_string();
    {
      std::lock_guard<std::mutex> lk(mu);
      auto l = leases.get(lease_id);
      if (!l || l->leasable.scheduler != from)
        throw std::runtime_error("invalid lease " + lease_id);
      // dispatching upgrades the offer lease to the working TTL
      leases.renew(lease_id, 10.0);
      auto info = l->leasable;
      info.job_id = job_id;
      leases.remove(lease_id);
      leases.insert(lease_id, info, 10.0);
    }
    auto job = std::make_shared<Job>();
    job->id = job_id;
    job->lease_id = lease_id;
    job->scheduler = from;
    job->work_dir = work_root + "/hypha-" + job_id;
    mkdir(work_root.c_str(), 0755);
    mkdir(job->work_dir.c_str(), 0755);
    {
      std::lock_guard<std::mutex> lk(mu);
      jobs[job_id] = job;
    }
    Json ex = jobj.at("executor");
    if (ex.has("train")) {
      job->runner = std::thread(
          [this, job, ex] { run_process_job(job, ex.at("train"), exec_cmd); });
    } else if (ex.has("infer")) {
      if (infer_cmd.empty()) throw std::runtime_error("inference not configured");
      job->runner = std::thread(
          [this, job, ex] { run_process_job(job, ex.at("infer"), infer_cmd); });
    } else if (ex.has("aggregate")) {
      job->is_aggregate = true;
      job->runner = std::thread(
          [this, job, ex] { run_aggregate_job(job, ex.at("aggregate")); });
    } else {
      throw std::runtime_error("unknown executor kind");
    }
    job->runner.detach();
    Json r;
    r["accepted"] = true;
    return r;
  }

  // --- process executor (executor/process.rs:78-198) ---------------------

  void run_process_job(std::shared_ptr<Job> job, Json config, const std::string& cmd_tmpl) {
    std::string sock_path = job->work_dir + "/bridge.sock";
    std::string job_json = job->work_dir + "/job.json";
    {
      FILE* f = fopen(job_json.c_str(), "w");
      if (!f) {
        fprintf(stderr, "[%s] cannot write %s\n", name.c_str(), job_json.c_str());
        report_job_status(job, "failed");
        return;
      }
      std::string s = config.dump();
      fwrite(s.data(), 1, s.size(), f);
      fclose(f);
    }
    auto bridge = std::make_shared<Bridge>(sock_path, job->work_dir);
    job->bridge = bridge;
    bridge->fetch_cb = [this, job](const Json& ref) { return connector_fetch(job, ref); };
    bridge->send_cb = [this, job](const Json& ref, const std::string& path) {
      return connector_send(job, ref, path);
    };
    bridge->status_cb = [this, job](const Json& progress) {
      Json body = progress;
      body["job"] = job->id;
      return node.request(job->scheduler, "progress", body, 30.0);
    };
    bridge->start();

    // {SOCKET_PATH}/{WORK_DIR}/{JOB_JSON} substitution (process.rs:201-205)
    std::string cmd = cmd_tmpl;
    auto subst = [&](const std::string& key, const std::string& val) {
      size_t p;
      while ((p = cmd.find(key)) != std::string::npos) cmd.replace(p, key.size(), val);
    };
    subst("{SOCKET_PATH}", sock_path);
    subst("{WORK_DIR}", job->work_dir);
    subst("{JOB_JSON}", job_json);

    // dispatch-span context for the executor's OTLP tracer: job id + the
    // dispatch wall-clock (ns) so the executor can emit a job.dispatch span
    // covering queue/spawn latency across the process boundary
    {
      struct timespec ts;
      clock_gettime(CLOCK_REALTIME, &ts);
      long long ns = (long long)ts.tv_sec * 1000000000LL + ts.tv_nsec;
      cmd = "HYPHA_JOB_ID=" + job->id + " HYPHA_DISPATCH_TS_NS=" +
            std::to_string(ns) + " " + cmd;
    }

    // pin one GPU from this daemon's pool to the job for its whole lease
    // lifetime: the executor sees exactly one device (cuda:0) regardless of
    // which physical GPU it landed on
    job->gpu_id = acquire_gpu();
    if (job->gpu_id >= 0) {
      cmd = "HIP_VISIBLE_DEVICES=" + std::to_string(job->gpu_id) +
            " CUDA_VISIBLE_DEVICES=" + std::to_string(job->gpu_id) + " " + cmd;
      fprintf(stderr, "[%s] job %s pinned to GPU %d\n", name.c_str(),
              job->id.c_str(), job->gpu_id);
    }

    pid_t pid = fork();
    if (pid == 0) {
      execl("/bin/sh", "sh", "-c", cmd.c_str(), (char*)nullptr);
      _exit(127);
    }
    job->pid = pid;
    int status = 0;
    waitpid(pid, &status, 0);
    job->pid = -1;
    release_gpu(job->gpu_id);
    job->gpu_id = -1;
    bridge->stop();
    bool ok = WIFEXITED(status) && WEXITSTATUS(status) == 0;
    report_job_status(job, job->cancelled ? "cancelled" : (ok ? "completed" : "failed"));
  }

  void report_job_status(std::shared_ptr<Job> job, const std::string& status) {
    Json st;
    st["id"] = job->id;
    st["status"] = status;
    try {
      node.request(job->scheduler, "job_status", st, 5.0);
    } catch (...) {
    }
    std::lock_guard<std::mutex> lk(mu);
    jobs.erase(job->id);
  }

  // --- connector (connector/mod.rs) --------------------------------------

  Json connector_fetch(std::shared_ptr<Job> job, const Json& ref) {
    // {"scheduler": {"peer": ..., "dataset": ...}} -> ask scheduler for a
    // slice index, then pull from the data provider (mod.rs:436-507)
    if (ref.has("scheduler")) {
      const Json& sc = ref.at("scheduler");
      Json q;
      q["dataset"] = sc.at("dataset");
      Json a = node.request(sc.at("peer").as_string(), "data", q, 30.0);
      std::string provider = a.at("data_provider").as_string();
      int64_t index = a.at("index").as_int();
      Json hdr;
      hdr["dataset"] = sc.at("dataset");
      hdr["index"] = index;
      auto stream = node.open_stream(provider, "pull_slice", hdr);
      auto szmsg = stream->recv_json();
      if (!szmsg) throw std::runtime_error("pull: no size header");
      size_t size = (size_t)szmsg->at("size").as_int();
      std::string out = job->work_dir + "/slice-" + std::to_string(index) + ".safetensors";
      FILE* f = fopen(out.c_str(), "wb");
      std::vector<char> buf(1 << 20);
      size_t left = size;
      while (left > 0) {
        size_t chunk = std::min(left, buf.size());
        if (!stream->recv_raw(buf.data(), chunk)) break;
        fwrite(buf.data(), 1, chunk, f);
        left -= chunk;
      }
      fclose(f);
      if (left != 0) throw std::runtime_error("pull: short read");
      Json r;
      r["files"] = JsonArray{Json(out)};
      return r;
    }
    // {"uri": {"value": "http(s)://..."}} -> GET into {work_dir}/artifacts/
    // (HttpHfFetcher, connector/mod.rs:237-255; artifacts dir + 0600 perms
    // mirror bridge.rs fetch_resource; allow-list enforced per hop)
    if (ref.has("uri")) {
      const std::string uri = ref.at("uri").at("value").as_string();
      parse_url(uri);  // validate_fetch: http(s) scheme or throw
      std::string dir = job->work_dir + "/artifacts";
      mkdir(dir.c_str(), 0700);
      std::string name = uri.substr(uri.rfind('/') + 1);
      if (name.empty() || name.find("..") != std::string::npos) name = "item-0";
      std::string out = dir + "/" + name;
      long long size = http_get_to_file(uri, out, fetch_allow);
      Json item;
      item["path"] = Json(std::string("artifacts/") + name);
      item["size"] = Json((int64_t)size);
      Json r;
      r["files"] = JsonArray{Json(out)};
      r["items"] = JsonArray{item};
      return r;
    }
    // {"huggingface": {"repository": ..., "revision": ..., "filenames": [...]}}
    // -> resolve each file against the HF endpoint (connector/mod.rs:256-297;
    // offline deployments point --hf-endpoint at a mirror)
    if (ref.has("huggingface")) {
      const Json& hf = ref.at("huggingface");
      std::string repo = hf.at("repository").as_string();
      std::string rev = hf.has("revision") ? hf.at("revision").as_string() : "main";
      if (repo.find("..") != std::string::npos)
        throw std::runtime_error("fetch: invalid repository");
      std::string dir = job->work_dir + "/artifacts";
      mkdir(dir.c_str(), 0700);
      Json files = JsonArray{};
      Json items = JsonArray{};
      if (hf.has("filenames")) {
        for (const auto& fn : hf.at("filenames").as_array()) {
          std::string f = fn.as_string();
          if (f.empty() || f[0] == '/' || f.find("..") != std::string::npos)
            throw std::runtime_error("fetch: path traversal is not allowed");
          std::string url = hf_endpoint + "/" + repo + "/resolve/" + rev + "/" + f;
          std::string base = f.substr(f.rfind('/') + 1);
          std::string out = dir + "/" + base;
          long long size = http_get_to_file(url, out, fetch_allow);
          files.as_array().push_back(Json(out));
          Json item;
          item["path"] = Json(std::string("artifacts/") + base);
          item["size"] = Json((int64_t)size);
          items.as_array().push_back(item);
        }
      }
      Json r;
      r["files"] = files;
      r["items"] = items;
      return r;
    }
    throw std::runtime_error("unsupported fetch reference");
  }

  Json connector_send(std::shared_ptr<Job> job, const Json& ref, const std::string& path) {
    // {"peers": {"peers": [...], "strategy": "all"}} push (mod.rs:305-433)
    FILE* f = fopen(path.c_str(), "rb");
    if (!f) throw std::runtime_error("send: no such file " + path);
    fseek(f, 0, SEEK_END);
    long size = ftell(f);
    fclose(f);
    std::string base = path.substr(path.rfind('/') + 1);
    // SelectionStrategy (connector/mod.rs:330-380): "all" -> every peer;
    // "one" -> first; "random" -> uniform pick (the reference collapses
    // Random to first — we honor the declared semantics).
    const Json& pj = ref.at("peers");
    std::vector<Json> targets;
    {
      const JsonArray& peers = pj.at("peers").as_array();
      if (peers.empty()) throw std::runtime_error("send: no peers provided");
      std::string strategy = pj.get_or("strategy", Json("all")).as_string();
      if (strategy == "all") {
        targets.assign(peers.begin(), peers.end());
      } else if (strategy == "one") {
        targets.push_back(peers.front());
      } else if (strategy == "random") {
        targets.push_back(peers[(size_t)rand() % peers.size()]);
      } else {
        throw std::runtime_error("send: unknown strategy " + strategy);
      }
    }
    // stream in 1 MB chunks: pseudo-gradient files are model-sized (16 GB
    // for 8B) — never buffer the whole payload in host memory
    std::vector<char> buf(1 << 20);
    for (auto& peer : targets) {
      Json hdr;
      hdr["name"] = base;
      hdr["size"] = (int64_t)size;
      hdr["job"] = ref.get_or("job", Json(job->id));
      if (ref.has("samples")) hdr["samples"] = ref.at("samples");
      auto stream = node.open_stream(peer.as_string(), "push_resource", hdr);
      FILE* fp = fopen(path.c_str(), "rb");
      if (!fp) throw std::runtime_error("send: cannot reopen " + path);
      long left = size;
      while (left > 0) {
        size_t chunk = std::min((long)buf.size(), left);
        if (fread(buf.data(), 1, chunk, fp) != chunk ||
            !stream->send_raw(buf.data(), chunk)) {
          fclose(fp);
          throw std::runtime_error("send: stream write failed to " + peer.as_string());
        }
        left -= chunk;
      }
      fclose(fp);
    }
    Json r;
    r["sent"] = true;
    return r;
  }

  void receive_resource(const std::string& from, const Json& header, MsgSocket& sock) {
    std::string jid = header.get_or("job", Json("")).as_string();
    std::shared_ptr<Job> job;
    {
      std::lock_guard<std::mutex> lk(mu);
      auto it = jobs.find(jid);
      if (it != jobs.end()) {
        job = it->second;
      } else {
        // fall back to the unique local train job (PS broadcasts carry the
        // PS job id; each worker runs at most one train job per scheduler)
        for (auto& [id, j] : jobs)
          if (!j->is_aggregate) {
            job = j;
            break;
          }
      }
    }
    size_t size = (size_t)header.at("size").as_int();
    std::string dir = job ? job->work_dir : work_root + "/orphan";
    mkdir(dir.c_str(), 0755);
    // peer-supplied names must stay inside the work dir (the safe_join rule
    // the bridge enforces, bridge.rs:330-346): keep only the basename and
    // drop dot-leading components
    auto sanitize = [](std::string s) {
      size_t slash = s.rfind('/');
      if (slash != std::string::npos) s = s.substr(slash + 1);
      if (s.empty() || s[0] == '.') s = "unnamed";
      return s;
    };
    std::string out = dir + "/recv-" + sanitize(from) + "-" +
                      sanitize(header.at("name").as_string());
    FILE* f = fopen(out.c_str(), "wb");
    if (!f) {
      fprintf(stderr, "[%s] recv: cannot open %s\n", name.c_str(), out.c_str());
      return;
    }
    std::vector<char> buf(1 << 20);
    size_t left = size;
    while (left > 0) {
      size_t chunk = std::min(left, buf.size());
      if (!sock.recv_raw(buf.data(), chunk)) break;
      fwrite(buf.data(), 1, chunk, f);
      left -= chunk;
    }
    fclose(f);
    if (left != 0 || !job) {
      fprintf(stderr, "[%s] recv incomplete from %s: left=%zu job=%d\n", name.c_str(),
              from.c_str(), left, job != nullptr);
      return;
    }
    if (job->is_aggregate) {
      {
        std::lock_guard<std::mutex> lk(job->agg_mu);
        // one update per peer per round; "samples" = the round's sample
        // count for optional weighted aggregation (heterogeneous batches)
        double w = header.get_or("samples", Json(1.0)).as_double();
        job->round_files[from] = {out, w > 0 ? w : 1.0};
      }
      job->agg_cv.notify_all();
      fprintf(stderr, "[%s] PS got file from %s\n", name.c_str(), from.c_str());
    } else if (job->bridge) {
      Json ev;
      ev["path"] = out;
      ev["size"] = (int64_t)size;
      ev["from_peer"] = from;
      job->bridge->push_event(ev);
    }
  }

  // --- built-in aggregate executor (parameter_server.rs:44-304) ----------

  void run_aggregate_job(std::shared_ptr<Job> job, Json config) {
    {
      std::lock_guard<std::mutex> lk(job->agg_mu);
      for (auto& pj : config.at("results").at("peers").at("peers").as_array())
        job->members.insert(pj.as_string());
      fprintf(stderr, "[%s] PS start: %zu members\n", name.c_str(), job->members.size());
    }
    double lr = config.at("optimizer").at("nesterov").at("learning_rate").as_double();
    double mu_ = config.at("optimizer").at("nesterov").at("momentum").as_double();
    const bool weighted = config.get_or("weighted_aggregation", Json(false)).as_bool();
    SafeTensors momentum;  // persists across rounds (parameter_server.rs:393-398)
    SafeTensors cum;       // cumulative sum of updates: joiner catch-up state
    bool have_momentum = false;

    auto serve_joiners = [&]() {
      std::vector<std::string> peers;
      {
        std::lock_guard<std::mutex> lk(job->agg_mu);
        peers.swap(job->pending_sync_peers);
      }
      if (peers.empty()) return;
      std::string cpath = job->work_dir + "/global_offset.safetensors";
      if (!have_momentum) {
        SafeTensors empty;  // no rounds yet: zero offset (deterministic init)
        empty.save(cpath);
      } else {
        cum.save(cpath);
      }
      for (auto& p : peers) {
        Json ref;
        Json jp;
        jp["peers"] = JsonArray{Json(p)};
        jp["strategy"] = std::string("all");
        ref["peers"] = jp;
        ref["offset"] = true;
        try {
          connector_send(job, ref, cpath);
        } catch (const std::exception& e) {
          fprintf(stderr, "[%s] PS joiner sync to %s failed: %s\n", name.c_str(),
                  p.c_str(), e.what());
        }
      }
    };

    while (!job->cancelled) {
      // wait until every current member has pushed this round's update
      std::vector<std::string> files;
      std::vector<double> file_weights;
      std::vector<std::string> targets;
      {
        std::unique_lock<std::mutex> lk(job->agg_mu);
        auto complete = [&] {
          if (job->members.empty()) return false;
          for (auto& m : job->members)
            if (!job->round_files.count(m)) return false;
          return true;
        };
        job->agg_cv.wait(lk, [&] {
          return job->cancelled || !job->pending_sync_peers.empty() || complete();
        });
        if (job->cancelled) break;
        if (!complete()) {
          lk.unlock();
          serve_joiners();
          continue;
        }
        for (auto& m : job->members) {
          files.push_back(job->round_files.at(m).first);
          file_weights.push_back(job->round_files.at(m).second);
        }
        targets.assign(job->members.begin(), job->members.end());
        job->round_files.clear();  // stale non-member leftovers dropped too
      }
      fprintf(stderr, "[%s] PS round: %zu files\n", name.c_str(), files.size());
      // running average + outer Nesterov (shared ps_math.h; golden-tested
      // vs torch SGD(nesterov=True) in tests/test_control_plane.py);
      // weighted_aggregation weights each delta by its sample count
      SafeTensors avg = weighted ? ps_weighted_average(files, file_weights)
                                 : ps_average(files);
      if (!have_momentum) {
        momentum = ps_zeros_like(avg);
        cum = momentum;  // zeros, same shapes
        have_momentum = true;
      }
      SafeTensors update = ps_nesterov(avg, momentum, lr, mu_);
      // cumulative offset for joiners: theta_global = theta_init + cum
      ps_add_(cum, update);
      std::string upath = job->work_dir + "/update.safetensors";
      update.save(upath);
      fprintf(stderr, "[%s] PS broadcasting to %zu targets\n", name.c_str(), targets.size());
      // broadcast to every CURRENT member (parameter_server.rs:232-269)
      Json send_ref;
      Json jp;
      JsonArray names;
      for (auto& t : targets) names.push_back(Json(t));
      jp["peers"] = names;
      jp["strategy"] = std::string("all");
      send_ref["peers"] = jp;
      connector_send(job, send_ref, upath);
      // notify scheduler (Progress::Updated, parameter_server.rs:274-282)
      Json prog;
      prog["kind"] = "updated";
      prog["job"] = job->id;
      try {
        Json resp = node.request(job->scheduler, "progress", prog, 30.0);
        if (resp.get_or("kind", Json("")).as_string() == "done") break;
      } catch (const std::exception& e) {
        fprintf(stderr, "[%s] PS progress failed: %s\n", name.c_str(), e.what());
        break;
      }
      serve_joiners();
    }
    report_job_status(job, job->cancelled ? "cancelled" : "completed");
  }
};

int main(int argc, char** argv) {
  std::string name = "worker", gw_host = "127.0.0.1", cmd, icmd, work_root = "/tmp/hypha-work";
  std::string advertise_host, listen_host = "127.0.0.1";
  int gw_port = 0, port = 0;
  bool probe = false, init = false;
  std::vector<int> gpu_ids;
  std::vector<std::string> exclude_cidrs, fetch_allow, fallback_gws;
  std::string hf_endpoint = "https://huggingface.co";
  TlsConfig tls;
  Resources total{1, 4, 16, 100};
  OfferPolicy policy{1.0, 0.0, {"diloco-transformer", "parameter-server"}};
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    auto next = [&] { return std::string(argv[++i]); };
    if (a == "--name") name = next();
    else if (a == "--gateway-host") gw_host = next();
    else if (a == "--gateway-port") gw_port = std::stoi(next());
    else if (a == "--port") port = std::stoi(next());
    else if (a == "--gpu") total.gpu = std::stod(next());
    else if (a == "--gpu-ids") {  // physical device indices this daemon owns
      std::string v = next();
      size_t pos = 0;
      while (pos <= v.size()) {
        size_t c = v.find(',', pos);
        if (c == std::string::npos) c = v.size();
        if (c > pos) gpu_ids.push_back(std::stoi(v.substr(pos, c - pos)));
        pos = c + 1;
      }
    }
    else if (a == "--cpu") total.cpu = std::stod(next());
    else if (a == "--memory") total.memory = std::stod(next());
    else if (a == "--storage") total.storage = std::stod(next());
    else if (a == "--price") policy.price = std::stod(next());
    else if (a == "--floor") policy.floor = std::stod(next());
    else if (a == "--offer-strategy") policy.strategy = next();
    else if (a == "--executors") {  // comma-separated supported executor list
      policy.supported_executors.clear();
      std::string v = next();
      size_t pos = 0;
      while (pos <= v.size()) {
        size_t c = v.find(',', pos);
        if (c == std::string::npos) c = v.size();
        if (c > pos) policy.supported_executors.push_back(v.substr(pos, c - pos));
        pos = c + 1;
      }
    }
    else if (a == "--exec-cmd") cmd = next();
    else if (a == "--infer-cmd") icmd = next();
    else if (a == "--work-root") work_root = next();
    else if (a == "--tls-cert") tls.cert_path = next();
    else if (a == "--tls-key") tls.key_path = next();
    else if (a == "--tls-ca") tls.ca_path = next();
    else if (a == "--tls-crl") tls.crl_path = next();
    else if (a == "--exclude-cidr") exclude_cidrs.push_back(next());
    else if (a == "--advertise-host") advertise_host = next();
    else if (a == "--listen-host") listen_host = next();
    else if (a == "--fallback-gateway") fallback_gws.push_back(next());
    else if (a == "--fetch-allow") fetch_allow.push_back(next());
    else if (a == "--hf-endpoint") hf_endpoint = next();
    else if (a == "probe") probe = true;
    else if (a == "init") init = true;
  }
  signal(SIGPIPE, SIG_IGN);
  if (init) {  // reference CLI Init subcommand: emit a commented config
    printf("# hypha-worker configuration (flags)\n"
           "# --name worker-0               node name in the registry\n"
           "# --gateway-host/--gateway-port gateway broker address\n"
           "# --gpu 1 --cpu 4 --memory 16 --storage 100   sellable resources\n"
           "# --price 1.0 --floor 0.0       auction offer policy\n"
           "# --offer-strategy flexible|whole  offer requested vs full capacity\n"
           "# --executors diloco-transformer,parameter-server  roles to sell\n"
           "# --exec-cmd 'python -m hypha_amd.runtime.executor --socket "
           "{SOCKET_PATH} --work-dir {WORK_DIR} --job {JOB_JSON}'\n"
           "# --infer-cmd '... hypha_amd.runtime.infer_executor ...'  enables "
           "the inference-transformer executor\n"
           "# --work-root /tmp/hypha-work   per-job working directories\n"
           "# --tls-cert/--tls-key/--tls-ca [--tls-crl]  mTLS identity\n"
           "# --fetch-allow host[:port]     allow URI/HF fetches from this host\n"
           "#                               (repeatable; *.domain wildcards; default: deny all)\n"
           "# --hf-endpoint https://huggingface.co  HuggingFace resolve endpoint\n"
           "# --fallback-gateway host:port  additional broker(s) tried on gateway loss\n"
           "# --advertise-host 10.0.0.5     host peers should dial this node at\n"
           "# --listen-host 0.0.0.0         bind address (default loopback-only)\n"
           "#                               (default: gateway-observed address)\n");
    return 0;
  }
  if (probe) {
    // readiness = a full mTLS registration round-trip (register + ack),
    // matching the reference's health RR over the authenticated transport
    // (hypha-worker.rs:312-354) rather than a bare TCP connect
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
  if (!icmd.empty()) {
    auto& se = policy.supported_executors;
    bool has = false;
    for (auto& e : se) has = has || e == "inference-transformer";
    if (!has) se.push_back("inference-transformer");
  }
  WorkerDaemon daemon(name, gw_host, gw_port, total, policy, cmd, work_root, tls);
  daemon.gpu_pool = gpu_ids;
  daemon.infer_cmd = icmd;
  daemon.fetch_allow = fetch_allow;
  for (const auto& g : fallback_gws) {
    auto c = g.rfind(':');
    daemon.node.add_fallback_gateway(g.substr(0, c), std::stoi(g.substr(c + 1)));
  }
  daemon.hf_endpoint = hf_endpoint;
  daemon.node.set_exclude_cidrs(exclude_cidrs);
  daemon.node.set_advertise_host(advertise_host);
  daemon.node.set_listen_host(listen_host);
  daemon.start(port);
  printf("hypha-worker %s ready on port %d\n", name.c_str(), daemon.node.port());
  fflush(stdout);
  while (true) std::this_thread::sleep_for(std::chrono::seconds(3600));
}
