// Parameter-server tensor math: running average of pseudo-gradient
// safetensors + outer Nesterov with persistent momentum.
// Shared by the worker daemon's aggregate executor and the pybind test
// bindings so the golden-value test (vs torch.optim.SGD(nesterov=True) —
// the reference's parameter_server.rs:448-525 pattern) exercises exactly
// the production code path.
#pragma once

#include <string>
#include <vector>

#include "safetensors.h"

namespace hypha {

// avg <- mean of the tensors across `files` (running mean, matching
// parameter_server.rs:194-209's (a+b)/2 pairwise stream generalized to N).
inline SafeTensors ps_average(const std::vector<std::string>& files) {
  SafeTensors avg = SafeTensors::load(files[0]);
  for (size_t i = 1; i < files.size(); ++i) {
    SafeTensors next = SafeTensors::load(files[i]);
    for (auto& [nm, tm] : avg.tensors) {
      auto& ntm = next.tensors.at(nm);
      for (int64_t e = 0; e < tm.numel(); ++e)
        avg.set_elem(tm, e, (avg.get_elem(tm, e) * i + next.get_elem(ntm, e)) / (i + 1));
    }
  }
  return avg;
}

// Weighted mean: avg <- sum(w_i * delta_i) / sum(w_i). The reference notes
// sample-count weighting as a missing TODO (parameter_server.rs:192-193);
// with capacity-proportional batch sizes a 2x-batch worker's pseudo-gradient
// represents 2x the samples, so weighting by contributed samples is the
// principled mean. Opt-in (job config "weighted_aggregation"): the unweighted
// path stays default for reference parity.
inline SafeTensors ps_weighted_average(const std::vector<std::string>& files,
                                       const std::vector<double>& weights) {
  SafeTensors avg = SafeTensors::load(files[0]);
  double wsum = weights[0];
  for (auto& [nm, tm] : avg.tensors)
    for (int64_t e = 0; e < tm.numel(); ++e)
      avg.set_elem(tm, e, (float)(avg.get_elem(tm, e) * weights[0]));
  for (size_t i = 1; i < files.size(); ++i) {
    SafeTensors next = SafeTensors::load(files[i]);
    for (auto& [nm, tm] : avg.tensors) {
      auto& ntm = next.tensors.at(nm);
      for (int64_t e = 0; e < tm.numel(); ++e)
        avg.set_elem(tm, e,
                     (float)(avg.get_elem(tm, e) + weights[i] * next.get_elem(ntm, e)));
    }
    wsum += weights[i];
  }
  for (auto& [nm, tm] : avg.tensors)
    for (int64_t e = 0; e < tm.numel(); ++e)
      avg.set_elem(tm, e, (float)(avg.get_elem(tm, e) / wsum));
  return avg;
}

// m <- mu*m + g ; update = lr*(mu*m + g)   (parameter_server.rs:386-446)
// momentum is updated in place; returns the update tensors.
inline SafeTensors ps_nesterov(const SafeTensors& g, SafeTensors& momentum, double lr,
                               double mu) {
  SafeTensors update = g;
  for (auto& [nm, tm] : g.tensors) {
    auto& mtm = momentum.tensors.at(nm);
    auto& utm = update.tensors.at(nm);
    for (int64_t e = 0; e < tm.numel(); ++e) {
      float gv = g.get_elem(tm, e);
      float mv = (float)(mu * momentum.get_elem(mtm, e) + gv);
      momentum.set_elem(mtm, e, mv);
      update.set_elem(utm, e, (float)(lr * (mu * mv + gv)));
    }
  }
  return update;
}

inline SafeTensors ps_zeros_like(const SafeTensors& t) {
  SafeTensors z = t;
  for (auto& [nm, tm] : z.tensors)
    for (int64_t e = 0; e < tm.numel(); ++e) z.set_elem(tm, e, 0.f);
  return z;
}

}  // namespace hypha
