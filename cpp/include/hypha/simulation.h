// Discrete-event projection of worker batch completions.
// Parity with /root/reference/crates/scheduler/src/simulation.rs
// (BasicSimulation::project :16-68): min-heap over each worker's next
// finish time; pop earliest, counter -= its batch_size, count one batch for
// it, push its next finish; stop when the counter is exhausted, the time cap
// is hit, or a worker would exceed the per-worker update cap. Returns
// (projected time ms, remaining counter, batches per worker, capped?).
#pragma once

#include <cstdint>
#include <map>
#include <queue>
#include <string>
#include <vector>

#include "trackers.h"

namespace hypha {

struct Projection {
  double time_ms = 0;
  int64_t remaining = 0;
  std::map<std::string, int64_t> batches_per_worker;
  bool capped = false;
};

struct BasicSimulation {
  double time_cap_ms = 10000.0;  // batch_scheduler.rs:88
  int64_t update_cap = 3;        // batch_scheduler.rs:89

  Projection project(const std::map<std::string, WorkerEntry>& workers,
                     int64_t counter) const {
    Projection out;
    out.remaining = counter;
    if (workers.empty() || counter <= 0) return out;

    using Ev = std::pair<double, std::string>;  // (finish time, peer)
    std::priority_queue<Ev, std::vector<Ev>, std::greater<Ev>> heap;
    for (auto& [peer, e] : workers) {
      double mean = e.stat.known() ? e.stat.mean : 1e18;
      heap.push({mean, peer});
      out.batches_per_worker[peer] = 0;
    }
    while (out.remaining > 0) {
      auto [t, peer] = heap.top();
      heap.pop();
      if (t > time_cap_ms) {
        out.capped = true;
        break;
      }
      auto& e = workers.at(peer);
      if (out.batches_per_worker[peer] + 1 > update_cap) {
        out.capped = true;
        break;
      }
      out.batches_per_worker[peer] += 1;
      out.remaining -= e.batch_size;
      out.time_ms = t;
      double mean = e.stat.known() ? e.stat.mean : 1e18;
      heap.push({t + mean, peer});
    }
    return out;
  }
};

}  // namespace hypha
