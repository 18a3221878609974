// pybind11 bindings for the C++ control-plane core (hypha_amd._core).
// Exposes the scheduler/worker logic classes so the Python test-suite can
// exercise them deterministically (injected clocks — the analogue of the
// reference's paused-tokio-time unit tests).

#include <pybind11/functional.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "hypha/auction.h"
#include "hypha/batch_scheduler.h"
#include "hypha/gateway.h"
#include "hypha/http_client.h"
#include "hypha/json.h"
#include "hypha/leases.h"
#include "hypha/net.h"
#include "hypha/ps_math.h"
#include "hypha/resources.h"
#include "hypha/simulation.h"
#include "hypha/trackers.h"

namespace py = pybind11;
using namespace hypha;

// Json <-> Python object conversion for the net bindings
static hypha::Json py_to_json(const py::object& o) {
  if (o.is_none()) return hypha::Json(nullptr);
  if (py::isinstance<py::bool_>(o)) return hypha::Json(o.cast<bool>());
  if (py::isinstance<py::int_>(o)) return hypha::Json((int64_t)o.cast<int64_t>());
  if (py::isinstance<py::float_>(o)) return hypha::Json(o.cast<double>());
  if (py::isinstance<py::str>(o)) return hypha::Json(o.cast<std::string>());
  if (py::isinstance<py::list>(o) || py::isinstance<py::tuple>(o)) {
    hypha::JsonArray arr;
    for (auto item : o.cast<py::sequence>()) arr.push_back(py_to_json(py::reinterpret_borrow<py::object>(item)));
    return hypha::Json(std::move(arr));
  }
  if (py::isinstance<py::dict>(o)) {
    hypha::JsonObject obj;
    for (auto item : o.cast<py::dict>())
      obj[item.first.cast<std::string>()] =
          py_to_json(py::reinterpret_borrow<py::object>(item.second));
    return hypha::Json(std::move(obj));
  }
  throw std::runtime_error("unsupported python type for json");
}

static py::object json_to_py(const hypha::Json& j) {
  if (j.is_null()) return py::none();
  if (j.is_bool()) return py::bool_(j.as_bool());
  if (j.is_number()) {
    double d = j.as_double();
    if (d == (int64_t)d) return py::int_((int64_t)d);
    return py::float_(d);
  }
  if (j.is_string()) return py::str(j.as_string());
  if (j.is_array()) {
    py::list l;
    for (auto& e : j.as_array()) l.append(json_to_py(e));
    return l;
  }
  py::dict d;
  for (auto& [k, v] : j.as_object()) d[py::str(k)] = json_to_py(v);
  return d;
}

// shared callback holder whose LAST release may happen on a non-Python
// thread (e.g. unsubscribe erasing the map, or a server thread dropping its
// copy): the deleter re-acquires the GIL before touching the refcount.
static std::shared_ptr<py::function> make_cb_holder(py::function cb) {
  return std::shared_ptr<py::function>(new py::function(std::move(cb)),
                                       [](py::function* f) {
                                         py::gil_scoped_acquire gil;
                                         delete f;
                                       });
}

PYBIND11_MODULE(_core, m) {
  m.doc() = "hypha_amd C++ control-plane core";

  // ---- json round-trip (used by tests to validate the wire format) ----
  m.def("json_roundtrip", [](const std::string& s) { return Json::parse(s).dump(); });
  m.def("cidr_contains", &cidr_contains, py::arg("cidr"), py::arg("host"));
  m.def("bandwidth_stats", [] {
    auto s = bandwidth_stats();
    py::dict d;
    d["inbound_bytes"] = s.inbound_bytes;
    d["outbound_bytes"] = s.outbound_bytes;
    return d;
  });

  // ---- resources ----
  py::class_<Resources>(m, "Resources")
      .def(py::init([](double gpu, double cpu, double memory, double storage) {
             return Resources{gpu, cpu, memory, storage};
           }),
           py::arg("gpu") = 0, py::arg("cpu") = 0, py::arg("memory") = 0,
           py::arg("storage") = 0)
      .def_readwrite("gpu", &Resources::gpu)
      .def_readwrite("cpu", &Resources::cpu)
      .def_readwrite("memory", &Resources::memory)
      .def_readwrite("storage", &Resources::storage)
      .def("fits_in", &Resources::fits_in)
      .def("partial_cmp",
           [](const Resources& a, const Resources& b) -> py::object {
             auto c = a.partial_cmp(b);
             if (!c) return py::none();
             return py::int_(*c);
           })
      .def("__add__", &Resources::operator+)
      .def("__sub__", &Resources::operator-)
      .def("__eq__", &Resources::operator==);

  py::class_<WeightedResourceEvaluator>(m, "WeightedResourceEvaluator")
      .def(py::init<>())
      .def("weighted_units", &WeightedResourceEvaluator::weighted_units)
      .def("score", &WeightedResourceEvaluator::score);

  // ---- leases ----
  using StrLedger = Ledger<std::string>;
  py::class_<StrLedger>(m, "Ledger")
      .def(py::init([](py::object clock) {
             if (clock.is_none()) return StrLedger(system_clock_fn());
             auto fn = clock.cast<std::function<double()>>();
             return StrLedger(fn);
           }),
           py::arg("clock") = py::none())
      .def("insert", &StrLedger::insert)
      .def("get",
           [](StrLedger& l, const std::string& id) -> py::object {
             auto r = l.get(id);
             if (!r) return py::none();
             return py::make_tuple(r->id, r->leasable, r->timeout);
           })
      .def("remove", &StrLedger::remove)
      .def("renew", &StrLedger::renew)
      .def("list_expired",
           [](StrLedger& l) {
             std::vector<std::string> out;
             for (auto& e : l.list_expired()) out.push_back(e.id);
             return out;
           })
      .def("drain_expired",
           [](StrLedger& l) {
             std::vector<std::string> out;
             for (auto& e : l.drain_expired()) out.push_back(e.id);
             return out;
           })
      .def("size", &StrLedger::size)
      .def("ids", &StrLedger::ids);

  // ---- trackers ----
  py::class_<SliceTracker>(m, "SliceTracker")
      .def(py::init<const std::string&, int>())
      .def("next",
           [](SliceTracker& t, const std::string& peer) {
             auto a = t.next(peer);
             return py::make_tuple(a.index, a.epoch);
           })
      .def("set_statistic", &SliceTracker::set_statistic)
      .def("remove_worker", &SliceTracker::remove_worker)
      .def("available_count", &SliceTracker::available_count)
      .def("owner", &SliceTracker::owner)
      .def_property_readonly("epoch", &SliceTracker::epoch);

  py::class_<ProgressTracker>(m, "ProgressTracker")
      .def(py::init<int64_t, int64_t>())
      .def("on_status", &ProgressTracker::on_status)
      .def("next_round", &ProgressTracker::next_round)
      .def("training_finished", &ProgressTracker::training_finished)
      .def_property_readonly("counter", &ProgressTracker::counter)
      .def_property_readonly("round", &ProgressTracker::round);

  py::class_<RunningMean>(m, "RunningMean")
      .def(py::init<>())
      .def("record", &RunningMean::record)
      .def_readonly("mean", &RunningMean::mean)
      .def_readonly("count", &RunningMean::count);

  // ---- simulation ----
  // direct simulation entry for property tests: workers = {peer: (batch,
  // mean_ms_or_None)}; returns a Projection
  m.def("simulate_project",
        [](const std::map<std::string, std::pair<int64_t, py::object>>& workers,
           int64_t counter, double time_cap_ms, int64_t update_cap) {
          std::map<std::string, WorkerEntry> entries;
          for (auto& [peer, be] : workers) {
            WorkerEntry e;
            e.peer = peer;
            e.batch_size = be.first;
            if (!be.second.is_none()) e.stat.record(be.second.cast<double>());
            entries[peer] = e;
          }
          BasicSimulation sim;
          sim.time_cap_ms = time_cap_ms;
          sim.update_cap = update_cap;
          return sim.project(entries, counter);
        },
        py::arg("workers"), py::arg("counter"), py::arg("time_cap_ms") = 10000.0,
        py::arg("update_cap") = 3);

  py::class_<Projection>(m, "Projection")
      .def_readonly("time_ms", &Projection::time_ms)
      .def_readonly("remaining", &Projection::remaining)
      .def_readonly("batches_per_worker", &Projection::batches_per_worker)
      .def_readonly("capped", &Projection::capped);

  // ---- batch scheduler FSM ----
  py::enum_<WorkerState>(m, "WorkerState")
      .value("Training", WorkerState::Training)
      .value("UpdateScheduled", WorkerState::UpdateScheduled)
      .value("Updating", WorkerState::Updating)
      .value("Done", WorkerState::Done);

  py::class_<BatchScheduler>(m, "BatchScheduler")
      .def(py::init([](int64_t samples_per_round, int64_t rounds, py::object clock) {
             if (clock.is_none())
               return BatchScheduler(samples_per_round, rounds);
             auto fn = clock.cast<std::function<double()>>();
             return BatchScheduler(samples_per_round, rounds, fn);
           }),
           py::arg("samples_per_round"), py::arg("rounds"),
           py::arg("clock") = py::none())
      .def("add_worker", &BatchScheduler::add_worker)
      .def("remove_worker", &BatchScheduler::remove_worker)
      .def("worker_state", &BatchScheduler::worker_state)
      .def_property_readonly("round", &BatchScheduler::round)
      .def_property_readonly("counter", &BatchScheduler::counter)
      .def("finished", &BatchScheduler::finished)
      .def("handle",
           [](BatchScheduler& s, const std::string& peer, const std::string& kind,
              int64_t batch_size) {
             Progress p{};
             if (kind == "status") {
               p.kind = Progress::Status;
               p.batch_size = batch_size;
             } else if (kind == "metrics") {
               p.kind = Progress::Metrics;
             } else if (kind == "update") {
               p.kind = Progress::Update;
             } else if (kind == "updated") {
               p.kind = Progress::Updated;
             } else if (kind == "update-received") {
               p.kind = Progress::UpdateReceived;
             } else {
               throw std::runtime_error("bad kind " + kind);
             }
             auto r = s.handle(peer, p);
             const char* names[] = {"ok", "continue", "schedule-update", "done",
                                    "error"};
             return py::make_tuple(names[(int)r.kind], r.counter);
           },
           py::arg("peer"), py::arg("kind"), py::arg("batch_size") = 0);

  // ---- auction ----
  py::class_<PriceRange>(m, "PriceRange")
      .def(py::init([](double bid, double mx) { return PriceRange{bid, mx}; }),
           py::arg("bid"), py::arg("max"))
      .def_readwrite("bid", &PriceRange::bid)
      .def_readwrite("max", &PriceRange::max);

  py::class_<WorkerRequest>(m, "WorkerRequest")
      .def(py::init([](std::string id, std::string scheduler, Resources r,
                       std::vector<std::string> execs, double bid, double timeout) {
             return WorkerRequest{std::move(id), std::move(scheduler), r,
                                  std::move(execs), bid, timeout};
           }),
           py::arg("id"), py::arg("scheduler"), py::arg("resources"),
           py::arg("executors"), py::arg("bid"), py::arg("timeout_s") = 5.0)
      .def_readwrite("id", &WorkerRequest::id)
      .def_readwrite("bid", &WorkerRequest::bid)
      .def_readwrite("resources", &WorkerRequest::resources);

  py::class_<WorkerOffer>(m, "WorkerOffer")
      .def(py::init([](std::string id, std::string request_id, std::string worker,
                       double price, Resources r, double expires_at) {
             return WorkerOffer{std::move(id), std::move(request_id),
                                std::move(worker), price, r, expires_at};
           }),
           py::arg("id"), py::arg("request_id"), py::arg("worker"),
           py::arg("price"), py::arg("resources"), py::arg("expires_at"))
      .def_readwrite("id", &WorkerOffer::id)
      .def_readwrite("worker", &WorkerOffer::worker)
      .def_readwrite("price", &WorkerOffer::price);

  py::class_<GreedyOfferAggregator>(m, "GreedyOfferAggregator")
      .def(py::init<size_t, PriceRange, double>())
      .def("add", &GreedyOfferAggregator::add)
      .def("finalize", &GreedyOfferAggregator::finalize)
      .def_property_readonly("deadline", &GreedyOfferAggregator::deadline);

  py::class_<OfferPolicy>(m, "OfferPolicy")
      .def(py::init([](double price, double floor, std::vector<std::string> execs) {
             return OfferPolicy{price, floor, std::move(execs)};
           }),
           py::arg("price"), py::arg("floor"), py::arg("supported_executors"));

  py::class_<ArbiterDecision>(m, "ArbiterDecision")
      .def_readonly("request", &ArbiterDecision::request)
      .def_readonly("offer_price", &ArbiterDecision::offer_price);

  m.def("select_requests", &select_requests);
  // URI fetch connector internals (connector/mod.rs:226-302 + the
  // allow-list the reference left as a TODO) - unit-testable offline
  m.def("parse_url", [](const std::string& u) {
    HttpUrl p = parse_url(u);
    py::dict d;
    d["scheme"] = p.scheme;
    d["host"] = p.host;
    d["port"] = p.port;
    d["path"] = p.path;
    return d;
  });
  m.def("fetch_allowed", [](const std::string& host, int port,
                            const std::vector<std::string>& allow) {
    return fetch_allowed(host, port, allow);
  });
  m.def("http_get_to_file",
        [](const std::string& url, const std::string& out,
           const std::vector<std::string>& allow, double timeout) {
          py::gil_scoped_release rel;
          return http_get_to_file(url, out, allow, timeout);
        },
        py::arg("url"), py::arg("out"), py::arg("allow"),
        py::arg("timeout") = 10.0);

  // parameter-server file pipeline (the production code path, for golden tests)
  m.def("ps_weighted_average_files",
        [](const std::vector<std::string>& delta_files, const std::vector<double>& w,
           const std::string& out) {
          SafeTensors avg = ps_weighted_average(delta_files, w);
          avg.save(out);
        });
  m.def("ps_aggregate_files",
        [](const std::vector<std::string>& delta_files, const std::string& momentum_io,
           const std::string& update_out, double lr, double mu) {
          SafeTensors avg = ps_average(delta_files);
          SafeTensors momentum;
          FILE* f = fopen(momentum_io.c_str(), "rb");
          if (f) {
            fclose(f);
            momentum = SafeTensors::load(momentum_io);
          } else {
            momentum = ps_zeros_like(avg);
          }
          SafeTensors update = ps_nesterov(avg, momentum, lr, mu);
          momentum.save(momentum_io);
          update.save(update_out);
        });

  // ---- in-process control-plane networking (multi-peer tests) ----
  py::class_<Gateway>(m, "Gateway")
      .def(py::init<>())
      .def(py::init([](const std::string& cert, const std::string& key,
                       const std::string& ca, const std::string& crl) {
             return new Gateway(TlsConfig{cert, key, ca, crl});
           }),
           py::arg("tls_cert"), py::arg("tls_key"), py::arg("tls_ca"),
           py::arg("tls_crl") = "")
      .def("start", &Gateway::start, py::arg("port") = 0,
           py::arg("listen_host") = "127.0.0.1",
           py::call_guard<py::gil_scoped_release>())
      .def("stop", &Gateway::stop, py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("port", &Gateway::port);

  py::class_<Node>(m, "Node")
      .def(py::init([](std::string name, std::string gh, int gp, std::string cert,
                       std::string key, std::string ca, std::string crl) {
             return new Node(std::move(name), std::move(gh), gp,
                             TlsConfig{cert, key, ca, crl});
           }),
           py::arg("name"), py::arg("gateway_host") = "127.0.0.1",
           py::arg("gateway_port") = 0, py::arg("tls_cert") = "",
           py::arg("tls_key") = "", py::arg("tls_ca") = "",
           py::arg("tls_crl") = "")
      .def("start", &Node::start, py::arg("port") = 0,
           py::call_guard<py::gil_scoped_release>())
      .def("stop", &Node::stop, py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("port", &Node::port)
      .def("on",
           [](Node& n, const std::string& type, py::function cb) {
             // shared_ptr so std::function copies in server threads never
             // touch Python refcounts without the GIL
             auto cbp = make_cb_holder(std::move(cb));
             n.on(type, [cbp](const std::string& from, const Json& body) -> Json {
               py::gil_scoped_acquire gil;
               try {
                 py::object r = (*cbp)(from, json_to_py(body));
                 return py_to_json(r);
               } catch (py::error_already_set& e) {
                 std::string msg = e.what();
                 e.restore();
                 PyErr_Clear();
                 throw std::runtime_error(msg);
               }
             });
           })
      .def("request",
           [](Node& n, const std::string& peer, const std::string& type, py::object body,
              double timeout_s) {
             Json b = py_to_json(body);
             Json r;
             {
               py::gil_scoped_release rel;
               r = n.request(peer, type, b, timeout_s);
             }
             return json_to_py(r);
           },
           py::arg("peer"), py::arg("type"), py::arg("body"), py::arg("timeout_s") = 10.0)
      .def("publish",
           [](Node& n, const std::string& topic, py::object data) {
             Json d = py_to_json(data);
             py::gil_scoped_release rel;
             n.publish(topic, d);
           })
      .def("subscribe",
           [](Node& n, const std::string& topic, py::function cb) {
             auto cbp = make_cb_holder(std::move(cb));
             n.subscribe(topic, [cbp](const std::string& from, const Json& data) {
               py::gil_scoped_acquire gil;
               try {
                 (*cbp)(from, json_to_py(data));
               } catch (py::error_already_set& e) {
                 e.restore();
                 PyErr_Clear();
               }
             });
           })
      .def("unsubscribe", &Node::unsubscribe, py::arg("topic"),
           py::call_guard<py::gil_scoped_release>())
      .def("kv_put",
           [](Node& n, const std::string& k, py::object v) {
             Json j = py_to_json(v);
             py::gil_scoped_release rel;
             n.kv_put(k, j);
           })
      .def("kv_get",
           [](Node& n, const std::string& k) -> py::object {
             std::optional<Json> r;
             {
               py::gil_scoped_release rel;
               r = n.kv_get(k);
             }
             if (!r) return py::none();
             return json_to_py(*r);
           })
      .def("resolve",
           [](Node& n, const std::string& p) -> py::object {
             auto r = n.resolve(p);
             if (!r) return py::none();
             return py::str(*r);
           })
      .def("push_blob",
           [](Node& n, const std::string& peer, const std::string& type,
              py::object header, py::bytes payload) {
             Json h = py_to_json(header);
             std::string data = payload;
             h["size"] = (int64_t)data.size();
             py::gil_scoped_release rel;
             auto stream = n.open_stream(peer, type, h);
             if (!stream->send_raw(data.data(), data.size()))
               throw std::runtime_error("stream write failed");
           })
      .def("set_exclude_cidrs", &Node::set_exclude_cidrs, py::arg("cidrs"))
      .def("set_advertise_host", &Node::set_advertise_host, py::arg("host"))
      .def("set_listen_host", &Node::set_listen_host, py::arg("host"))
      .def("stream_call",
           // bidirectional stream RPC: send header (+payload), read a JSON
           // reply and an optional `size`-byte body. Client side of the
           // param_push/param_pull versioned-KV protocol (lib.rs:698-739).
           [](Node& n, const std::string& peer, const std::string& type,
              py::object header, py::object payload) {
             Json h = py_to_json(header);
             std::string data;
             if (!payload.is_none()) {
               data = std::string(py::cast<py::bytes>(payload));
               h["size"] = (int64_t)data.size();
             }
             Json reply;
             std::string body;
             {
               py::gil_scoped_release rel;
               auto s = n.open_stream(peer, type, h);
               if (!data.empty() && !s->send_raw(data.data(), data.size()))
                 throw std::runtime_error("stream_call: write failed");
               auto r = s->recv_json();
               if (!r) throw std::runtime_error("stream_call: no reply");
               reply = *r;
               int64_t sz = reply.get_or("size", Json((int64_t)0)).as_int();
               if (sz > 0) {
                 body.resize((size_t)sz);
                 if (!s->recv_raw(body.data(), (size_t)sz))
                   throw std::runtime_error("stream_call: short body");
               }
             }
             py::object b = body.empty() ? py::object(py::none())
                                         : py::object(py::bytes(body));
             return py::make_tuple(json_to_py(reply), b);
           },
           py::arg("peer"), py::arg("type"), py::arg("header"),
           py::arg("payload") = py::none())
      .def("add_fallback_gateway", &Node::add_fallback_gateway)
      .def("on_gateway_reconnect",
           [](Node& n, py::function cb) {
             auto cbp = make_cb_holder(std::move(cb));
             n.on_gateway_reconnect([cbp] {
               py::gil_scoped_acquire gil;
               try {
                 (*cbp)();
               } catch (py::error_already_set& e) {
                 e.restore();
                 PyErr_Clear();
               }
             });
           })
      .def("on_blob",
           [](Node& n, const std::string& type, py::function cb) {
             auto cbp = make_cb_holder(std::move(cb));
             n.on_stream(type, [cbp](const std::string& from, const Json& header,
                                     MsgSocket& sock) {
               size_t size = (size_t)header.get_or("size", Json((int64_t)0)).as_int();
               std::string data(size, '\0');
               bool ok = size == 0 || sock.recv_raw(data.data(), size);
               py::gil_scoped_acquire gil;
               py::object blob = ok ? py::object(py::bytes(data)) : py::object(py::none());
               try {
                 (*cbp)(from, json_to_py(header), blob);
               } catch (py::error_already_set& e) {
                 e.restore();
                 PyErr_Clear();
               }
             });
           });

  py::class_<StaticResourceManager>(m, "StaticResourceManager")
      .def(py::init<Resources>())
      .def("reserve", &StaticResourceManager::reserve)
      .def("release", &StaticResourceManager::release)
      .def("available", &StaticResourceManager::available)
      .def("total", &StaticResourceManager::total);
}
