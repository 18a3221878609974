// Decentralized resource-allocation (dRAP) auction logic, both sides.
// Parity targets:
//   buy side  — /root/reference/crates/scheduler/src/allocator.rs
//     (GreedyOfferAggregator :276-419: rank offers by price per weighted
//      resource unit, reject above price.max, deadline shrinks to the
//      earliest offer expiry, early-return at capacity; Candidates dedup
//      :194-264)
//   sell side — /root/reference/crates/worker/src/arbiter.rs
//     (filter ads by supported executors :338-349, bid >= floor :352-361,
//      resources fit :364-372; score, sort desc, greedily offer :383-435)
#pragma once

#include <algorithm>
#include <cstdint>
#include <map>
#include <set>
#include <string>
#include <vector>

#include "resources.h"

namespace hypha {

struct PriceRange {
  double bid = 0;  // advertised price
  double max = 0;  // ceiling the buyer will accept
};

struct WorkerRequest {
  std::string id;          // request id
  std::string scheduler;   // requesting peer
  Resources resources;     // per-worker requirement
  std::vector<std::string> executors;  // required executor names
  double bid = 0;
  double timeout_s = 5.0;  // offer deadline (allocator.rs:25)
};

struct WorkerOffer {
  std::string id;          // lease id on the worker
  std::string request_id;
  std::string worker;      // offering peer
  double price = 0;
  Resources resources;
  double expires_at = 0;   // absolute seconds (500 ms TTL at the worker)
};

// --- buy side --------------------------------------------------------------

class GreedyOfferAggregator {
 public:
  GreedyOfferAggregator(size_t count, PriceRange price, double deadline_abs)
      : count_(count), price_(price), deadline_(deadline_abs) {}

  // Returns true once enough acceptable offers have arrived (early return).
  bool add(const WorkerOffer& offer, double now) {
    if (now > deadline_) return accepted_count() >= count_;
    if (offer.price > price_.max) return false;
    if (seen_workers_.count(offer.worker)) return false;  // dedup per worker
    seen_workers_.insert(offer.worker);
    offers_.push_back(offer);
    // deadline shrinks to the earliest offer expiry (allocator.rs deadline)
    if (offer.expires_at > now && offer.expires_at < deadline_)
      deadline_ = offer.expires_at;
    return accepted_count() >= count_;
  }

  double deadline() const { return deadline_; }
  size_t accepted_count() const { return std::min(offers_.size(), count_); }

  // Best `count` offers by price per weighted resource unit (ascending).
  std::vector<WorkerOffer> finalize() {
    std::stable_sort(offers_.begin(), offers_.end(),
                     [&](const WorkerOffer& a, const WorkerOffer& b) {
                       return eval_.score(a.price, a.resources) <
                              eval_.score(b.price, b.resources);
                     });
    if (offers_.size() > count_) offers_.resize(count_);
    return offers_;
  }

 private:
  size_t count_;
  PriceRange price_;
  double deadline_;
  WeightedResourceEvaluator eval_;
  std::vector<WorkerOffer> offers_;
  std::set<std::string> seen_workers_;
};

// --- sell side -------------------------------------------------------------

struct OfferPolicy {
  double price = 1.0;   // asking price per weighted unit scale
  double floor = 0.0;   // minimum acceptable bid
  std::vector<std::string> supported_executors;
  // OfferStrategy (worker/config.rs:54-61): "flexible" offers exactly the
  // requested resources; "whole" offers the worker's FULL capacity
  // (all-or-nothing co-tenancy; the scheduler scales the batch to it).
  std::string strategy = "flexible";
};

struct ArbiterDecision {
  WorkerRequest request;
  double offer_price;
};

// Score a batch of ads (arbiter batching window: 100 msgs / 200 ms,
// arbiter.rs:25-26) and pick which to answer, best-paying first, greedily
// reserving from `available`.
inline std::vector<ArbiterDecision> select_requests(
    std::vector<WorkerRequest> ads, const OfferPolicy& policy,
    Resources available) {
  WeightedResourceEvaluator eval;
  std::vector<WorkerRequest> eligible;
  for (auto& ad : ads) {
    bool exec_ok = true;
    for (auto& e : ad.executors) {
      if (std::find(policy.supported_executors.begin(),
                    policy.supported_executors.end(),
                    e) == policy.supported_executors.end()) {
        exec_ok = false;
        break;
      }
    }
    if (!exec_ok) continue;
    if (ad.bid < policy.floor) continue;
    if (!ad.resources.fits_in(available)) continue;
    eligible.push_back(ad);
  }
  // highest bid per weighted unit first (seller's preference)
  std::stable_sort(eligible.begin(), eligible.end(),
                   [&](const WorkerRequest& a, const WorkerRequest& b) {
                     return eval.score(a.bid, a.resources) >
                            eval.score(b.bid, b.resources);
                   });
  std::vector<ArbiterDecision> out;
  Resources remaining = available;
  for (auto& ad : eligible) {
    if (!ad.resources.fits_in(remaining)) continue;
    remaining = remaining - ad.resources;
    out.push_back({ad, std::max(ad.bid, policy.floor)});
  }
  return out;
}

// --- worker-side resource accounting ---------------------------------------
// Parity: /root/reference/crates/worker/src/resources.rs
// (StaticResourceManager: configured totals minus reservations,
//  double-checked reserve :18-80)

class StaticResourceManager {
 public:
  explicit StaticResourceManager(Resources total) : total_(total) {}

  Resources total() const { return total_; }
  Resources available() const { return total_ - reserved_; }

  bool reserve(const Resources& r) {
    if (!r.fits_in(available())) return false;
    reserved_ = reserved_ + r;
    return true;
  }
  void release(const Resources& r) {
    reserved_ = reserved_ - r;
    if (reserved_.gpu < 0) reserved_.gpu = 0;
    if (reserved_.cpu < 0) reserved_.cpu = 0;
    if (reserved_.memory < 0) reserved_.memory = 0;
    if (reserved_.storage < 0) reserved_.storage = 0;
  }

 private:
  Resources total_;
  Resources reserved_;
};

}  // namespace hypha
