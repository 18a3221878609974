// Scheduler-side trackers: data slices, round progress, worker states.
// Parity targets (semantics, not code):
//   SliceTracker     — /root/reference/crates/scheduler/src/tracker/slice.rs
//                      (available->processed with owner, cache-steal from the
//                       slowest peer on exhaustion :66-90, epoch rollover
//                       :92-99, remove_worker reclaim :105-114)
//   ProgressTracker  — tracker/progress.rs (counter -= batch_size :42-47,
//                      next_round :49-54, training_finished :64-66)
//   WorkerTracker    — tracker/worker.rs (peer/batch-size/last-update/
//                      statistic arrays, states Training/UpdateScheduled/
//                      Updating/Done :7-12)
//   RunningMean      — statistics.rs (incremental mean, "unknown = +inf")
#pragma once

#include <algorithm>
#include <cstdint>
#include <deque>
#include <map>
#include <optional>
#include <string>
#include <vector>

namespace hypha {

// Incremental mean of per-batch wall ms. Unmeasured workers rank slowest
// (the reference initialises with u64::MAX).
struct RunningMean {
  double mean = 1e18;
  uint64_t count = 0;
  void record(double sample_ms) {
    if (count == 0) {
      mean = sample_ms;
    } else {
      mean += (sample_ms - mean) / double(count + 1);
    }
    ++count;
  }
  bool known() const { return count > 0; }
};

// ---------------------------------------------------------------------------

class SliceTracker {
 public:
  struct Assignment {
    int index;
    int epoch;
  };

  SliceTracker(const std::string& dataset, int num_slices)
      : dataset_(dataset), num_slices_(num_slices) {
    for (int i = 0; i < num_slices; ++i) available_.push_back(i);
    owner_.assign(num_slices, "");
  }

  const std::string& dataset() const { return dataset_; }
  int epoch() const { return epoch_; }
  int num_slices() const { return num_slices_; }

  void set_statistic(const std::string& peer, double mean_batch_ms) {
    stats_[peer] = mean_batch_ms;
  }

  // Next slice for `peer`: an available slice if any; otherwise steal the
  // most recent slice of the slowest other peer; otherwise roll the epoch.
  Assignment next(const std::string& peer) {
    if (available_.empty()) {
      if (auto stolen = steal_from_slowest(peer)) {
        return {*stolen, epoch_};
      }
      rollover();
    }
    int idx = available_.front();
    available_.pop_front();
    assign(idx, peer);
    return {idx, epoch_};
  }

  // Reclaim a removed worker's current-epoch slices (slice.rs:105-114).
  void remove_worker(const std::string& peer) {
    auto it = assigned_.find(peer);
    if (it != assigned_.end()) {
      for (int idx : it->second) {
        owner_[idx] = "";
        available_.push_back(idx);
      }
      assigned_.erase(it);
    }
    stats_.erase(peer);
  }

  size_t available_count() const { return available_.size(); }
  const std::string& owner(int idx) const { return owner_[idx]; }

 private:
  void assign(int idx, const std::string& peer) {
    if (!owner_[idx].empty()) {
      auto& v = assigned_[owner_[idx]];
      v.erase(std::remove(v.begin(), v.end(), idx), v.end());
    }
    owner_[idx] = peer;
    assigned_[peer].push_back(idx);
  }

  std::optional<int> steal_from_slowest(const std::string& requester) {
    std::string slowest;
    double worst = -1;
    for (auto& [peer, slices] : assigned_) {
      if (peer == requester || slices.empty()) continue;
      auto it = stats_.find(peer);
      double mean = it == stats_.end() ? 1e18 : it->second;
      if (mean > worst) {
        worst = mean;
        slowest = peer;
      }
    }
    if (slowest.empty()) return std::nullopt;
    // steal only from a strictly slower peer (unknown == unknown rolls over)
    auto rit = stats_.find(requester);
    double rmean = rit == stats_.end() ? 1e18 : rit->second;
    if (!(worst > rmean)) return std::nullopt;
    int idx = assigned_[slowest].back();
    assign(idx, requester);
    return idx;
  }

  void rollover() {
    ++epoch_;
    available_.clear();
    for (int i = 0; i < num_slices_; ++i) available_.push_back(i);
    assigned_.clear();
    owner_.assign(num_slices_, "");
  }

  std::string dataset_;
  int num_slices_;
  int epoch_ = 0;
  std::deque<int> available_;
  std::vector<std::string> owner_;
  std::map<std::string, std::vector<int>> assigned_;  // per-epoch assignments
  std::map<std::string, double> stats_;
};

// ---------------------------------------------------------------------------

class ProgressTracker {
 public:
  ProgressTracker(int64_t samples_per_round, int64_t update_epochs)
      : samples_per_round_(samples_per_round),
        update_epochs_(update_epochs),
        counter_(samples_per_round) {}

  void on_status(int64_t batch_size) { counter_ -= batch_size; }
  int64_t counter() const { return counter_; }
  int64_t round() const { return update_counter_; }

  void next_round() {
    counter_ = samples_per_round_;
    ++update_counter_;
  }
  bool training_finished() const { return update_counter_ >= update_epochs_; }

 private:
  int64_t samples_per_round_;
  int64_t update_epochs_;
  int64_t counter_;
  int64_t update_counter_ = 0;
};

// ---------------------------------------------------------------------------

enum class WorkerState { Training, UpdateScheduled, Updating, Done };

struct WorkerEntry {
  std::string peer;
  int64_t batch_size = 1;
  double last_status_time = -1;  // seconds
  RunningMean stat;
  WorkerState state = WorkerState::Training;
  int64_t scheduled_counter = -1;  // batches until update, once scheduled
};

class WorkerTracker {
 public:
  void add(const std::string& peer, int64_t batch_size) {
    entries_[peer] = WorkerEntry{peer, batch_size};
  }
  bool has(const std::string& peer) const { return entries_.count(peer) > 0; }
  WorkerEntry& at(const std::string& peer) { return entries_.at(peer); }
  void remove(const std::string& peer) { entries_.erase(peer); }
  std::map<std::string, WorkerEntry>& entries() { return entries_; }
  size_t size() const { return entries_.size(); }

  bool all_in(WorkerState s) const {
    for (auto& [_, e] : entries_)
      if (e.state != s) return false;
    return !entries_.empty();
  }

 private:
  std::map<std::string, WorkerEntry> entries_;
};

}  // namespace hypha
