// Generic lease ledger.
// Parity with /root/reference/crates/leases/src/lib.rs: Lease{id, leasable,
// timeout}; Ledger insert/get/remove/renew/list_expired (:40-131); renew sets
// timeout = now + duration; wall-clock by design (:23-27). The clock is
// injectable so scheduler-FSM tests run deterministically (the reference uses
// paused tokio time for the same purpose).
#pragma once

#include <chrono>
#include <cstdint>
#include <functional>
#include <map>
#include <optional>
#include <string>
#include <vector>

namespace hypha {

using Clock = std::function<double()>;  // seconds, monotonically increasing

inline Clock system_clock_fn() {
  return [] {
    return std::chrono::duration<double>(
               std::chrono::system_clock::now().time_since_epoch())
        .count();
  };
}

template <typename T>
class Ledger {
 public:
  struct Lease {
    std::string id;
    T leasable;
    double timeout;  // absolute seconds
  };

  explicit Ledger(Clock clock = system_clock_fn()) : clock_(std::move(clock)) {}

  void insert(const std::string& id, T leasable, double duration_s) {
    leases_[id] = Lease{id, std::move(leasable), clock_() + duration_s};
  }

  std::optional<Lease> get(const std::string& id) const {
    auto it = leases_.find(id);
    if (it == leases_.end()) return std::nullopt;
    return it->second;
  }

  bool remove(const std::string& id) { return leases_.erase(id) > 0; }

  // renew = now + duration (not timeout + duration) — leases/src/lib.rs:103-114
  bool renew(const std::string& id, double duration_s) {
    auto it = leases_.find(id);
    if (it == leases_.end()) return false;
    if (it->second.timeout < clock_()) return false;  // already expired
    it->second.timeout = clock_() + duration_s;
    return true;
  }

  std::vector<Lease> list_expired() const {
    std::vector<Lease> out;
    double now = clock_();
    for (auto& [id, l] : leases_)
      if (l.timeout < now) out.push_back(l);
    return out;
  }

  std::vector<Lease> drain_expired() {
    std::vector<Lease> out = list_expired();
    for (auto& l : out) leases_.erase(l.id);
    return out;
  }

  size_t size() const { return leases_.size(); }
  std::vector<std::string> ids() const {
    std::vector<std::string> out;
    for (auto& [id, _] : leases_) out.push_back(id);
    return out;
  }

 private:
  Clock clock_;
  std::map<std::string, Lease> leases_;
};

}  // namespace hypha
