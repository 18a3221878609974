// DiLoCo synchronization-point scheduler (the control-plane hot loop).
// Parity with /root/reference/crates/scheduler/src/scheduling/
// batch_scheduler.rs: per-worker state machine
//   Training --ScheduleUpdate{counter}--> UpdateScheduled --Update--> Updating
//   --(aggregate Updated -> next_round)--UpdateReceived--> Training|Done
// (state diagram :42-53, schedule() :54-163, run :168-220; caps :88-89).
//
// Each worker Status{batch_size} message decrements the round counter and
// feeds the worker's running-mean batch time; a discrete-event simulation
// projects completion, and when the projected remaining count reaches zero
// the scheduler hands each worker a personal batch countdown.
#pragma once

#include <cstdint>
#include <map>
#include <optional>
#include <string>

#include "leases.h"
#include "simulation.h"
#include "trackers.h"

namespace hypha {

struct Progress {
  enum Kind { Status, Metrics, Update, Updated, UpdateReceived } kind;
  int64_t batch_size = 0;                 // Status
  int64_t round = 0;                      // Metrics
  std::map<std::string, double> metrics;  // Metrics
};

struct ProgressResponse {
  enum Kind { Ok, Continue, ScheduleUpdate, Done, Error } kind;
  int64_t counter = 0;  // ScheduleUpdate payload
  std::string error;
};

class BatchScheduler {
 public:
  BatchScheduler(int64_t samples_per_round, int64_t rounds,
                 Clock clock = system_clock_fn())
      : progress_(samples_per_round, rounds), clock_(std::move(clock)) {}

  void add_worker(const std::string& peer, int64_t batch_size) {
    workers_.add(peer, batch_size);
  }
  void remove_worker(const std::string& peer) { workers_.remove(peer); }

  int64_t round() const { return progress_.round(); }
  int64_t counter() const { return progress_.counter(); }
  bool finished() const { return progress_.training_finished(); }
  WorkerState worker_state(const std::string& peer) {
    return workers_.at(peer).state;
  }

  // Handle one progress message from `peer` (or the aggregator for Updated).
  ProgressResponse handle(const std::string& peer, const Progress& msg) {
    switch (msg.kind) {
      case Progress::Status:
        return on_status(peer, msg.batch_size);
      case Progress::Metrics:
        return {ProgressResponse::Ok};
      case Progress::Update: {
        if (workers_.has(peer)) workers_.at(peer).state = WorkerState::Updating;
        return {ProgressResponse::Ok};
      }
      case Progress::Updated: {
        // aggregate applied: advance the round; workers go back to Training
        // when their UpdateReceived arrives. Done tells the aggregator to
        // exit once the configured rounds are exhausted.
        progress_.next_round();
        round_scheduled_ = false;
        if (progress_.training_finished()) return {ProgressResponse::Done};
        return {ProgressResponse::Ok};
      }
      case Progress::UpdateReceived: {
        if (progress_.training_finished()) {
          if (workers_.has(peer)) workers_.at(peer).state = WorkerState::Done;
          return {ProgressResponse::Done};
        }
        if (workers_.has(peer)) {
          workers_.at(peer).state = WorkerState::Training;
          workers_.at(peer).scheduled_counter = -1;
        }
        return {ProgressResponse::Continue};
      }
    }
    return {ProgressResponse::Error, 0, "unknown progress kind"};
  }

 private:
  ProgressResponse on_status(const std::string& peer, int64_t batch_size) {
    if (!workers_.has(peer)) workers_.add(peer, batch_size);
    auto& e = workers_.at(peer);
    double now = clock_() * 1000.0;  // ms
    if (e.last_status_time >= 0) e.stat.record(now - e.last_status_time);
    e.last_status_time = now;
    e.batch_size = batch_size;
    progress_.on_status(batch_size);

    // already told this worker when to stop: just acknowledge
    if (e.state == WorkerState::UpdateScheduled) return {ProgressResponse::Ok};

    if (round_scheduled_) {
      // a sync point is set: deliver this worker its personal countdown
      e.state = WorkerState::UpdateScheduled;
      int64_t c = scheduled_counters_.count(peer) ? scheduled_counters_[peer] : 0;
      return {ProgressResponse::ScheduleUpdate, c};
    }

    if (progress_.counter() <= 0) {
      // round exhausted outright: everyone updates at their next status
      schedule_now();
      e.state = WorkerState::UpdateScheduled;
      return {ProgressResponse::ScheduleUpdate, scheduled_counters_[peer]};
    }

    // project completion: when the whole remaining counter fits inside the
    // simulation caps, fix each worker's remaining batch count now
    Projection p = sim_.project(workers_.entries(), progress_.counter());
    if (!p.capped && p.remaining <= 0) {
      round_scheduled_ = true;
      scheduled_counters_ = p.batches_per_worker;
      e.state = WorkerState::UpdateScheduled;
      return {ProgressResponse::ScheduleUpdate, scheduled_counters_[peer]};
    }
    return {ProgressResponse::Continue};
  }

  void schedule_now() {
    round_scheduled_ = true;
    scheduled_counters_.clear();
    for (auto& [peer, _] : workers_.entries()) scheduled_counters_[peer] = 0;
  }

  ProgressTracker progress_;
  WorkerTracker workers_;
  BasicSimulation sim_;
  Clock clock_;
  bool round_scheduled_ = false;
  std::map<std::string, int64_t> scheduled_counters_;
};

}  // namespace hypha
