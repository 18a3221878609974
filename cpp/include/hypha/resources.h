// Resources 4-vector + evaluators.
// Parity with /root/reference/crates/resources/src/lib.rs: partial order
// requires all dims comparable (:123-143); WeightedResourceEvaluator default
// weights gpu=25 cpu=1 mem=0.1 storage=0.01, score = price / weighted-units
// (:158-189).
#pragma once

#include <optional>

#include "json.h"

namespace hypha {

struct Resources {
  double gpu = 0, cpu = 0, memory = 0, storage = 0;

  Resources operator+(const Resources& o) const {
    return {gpu + o.gpu, cpu + o.cpu, memory + o.memory, storage + o.storage};
  }
  Resources operator-(const Resources& o) const {
    return {gpu - o.gpu, cpu - o.cpu, memory - o.memory, storage - o.storage};
  }
  bool operator==(const Resources& o) const {
    return gpu == o.gpu && cpu == o.cpu && memory == o.memory && storage == o.storage;
  }

  // partial order: defined only when every dimension agrees in direction
  std::optional<int> partial_cmp(const Resources& o) const {
    bool le = gpu <= o.gpu && cpu <= o.cpu && memory <= o.memory && storage <= o.storage;
    bool ge = gpu >= o.gpu && cpu >= o.cpu && memory >= o.memory && storage >= o.storage;
    if (le && ge) return 0;
    if (le) return -1;
    if (ge) return 1;
    return std::nullopt;
  }
  bool fits_in(const Resources& capacity) const {
    auto c = partial_cmp(capacity);
    return c.has_value() && *c <= 0;
  }

  Json to_json() const {
    JsonObject o;
    o["gpu"] = gpu;
    o["cpu"] = cpu;
    o["memory"] = memory;
    o["storage"] = storage;
    return Json(o);
  }
  static Resources from_json(const Json& j) {
    Resources r;
    r.gpu = j.get_or("gpu", Json(0.0)).as_double();
    r.cpu = j.get_or("cpu", Json(0.0)).as_double();
    r.memory = j.get_or("memory", Json(0.0)).as_double();
    r.storage = j.get_or("storage", Json(0.0)).as_double();
    return r;
  }
};

struct WeightedResourceEvaluator {
  double w_gpu = 25.0, w_cpu = 1.0, w_memory = 0.1, w_storage = 0.01;

  double weighted_units(const Resources& r) const {
    return r.gpu * w_gpu + r.cpu * w_cpu + r.memory * w_memory + r.storage * w_storage;
  }
  // lower is better: price per weighted resource unit
  double score(double price, const Resources& r) const {
    double u = weighted_units(r);
    if (u <= 0) return 1e300;
    return price / u;
  }
};

}  // namespace hypha
