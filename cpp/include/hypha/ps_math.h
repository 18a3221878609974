// Parameter-server tensor math: running average of pseudo-gradient
// safetensors + outer Nesterov with persistent momentum.
// Shared by the worker daemon's aggregate executor and the pybind test
// bindings so the golden-value test (vs torch.optim.SGD(nesterov=True) —
// the reference's parameter_server.rs:448-525 pattern) exercises exactly
// the production code path.
//
// Hot loops dispatch ONCE per tensor on dtype and then run over raw arrays
// (the naive per-element get_elem/set_elem path string-compares dtypes per
// element — ~20x slower, which matters at WAN scale where a pseudo-gradient
// is model-sized). The generic fallback handles mixed-dtype pairs.
#pragma once

#include <string>
#include <vector>

#include "safetensors.h"

namespace hypha {

// y[e] <- f(y[e], x[e]) with per-tensor dtype dispatch.
template <class F>
inline void zip2_(SafeTensors& Y, TensorMeta& ty, const SafeTensors& X,
                  const TensorMeta& tx, F&& f) {
  const int64_t n = ty.numel();
  if (ty.dtype == "F32" && tx.dtype == "F32") {
    float* y = reinterpret_cast<float*>(Y.data.data() + ty.begin);
    const float* x = reinterpret_cast<const float*>(X.data.data() + tx.begin);
    for (int64_t e = 0; e < n; ++e) y[e] = f(y[e], x[e]);
  } else if (ty.dtype == "BF16" && tx.dtype == "BF16") {
    uint16_t* y = reinterpret_cast<uint16_t*>(Y.data.data() + ty.begin);
    const uint16_t* x = reinterpret_cast<const uint16_t*>(X.data.data() + tx.begin);
    for (int64_t e = 0; e < n; ++e)
      y[e] = SafeTensors::f32_to_bf16(
          f(SafeTensors::bf16_to_f32(y[e]), SafeTensors::bf16_to_f32(x[e])));
  } else {
    for (int64_t e = 0; e < n; ++e)
      Y.set_elem(ty, e, f(Y.get_elem(ty, e), X.get_elem(tx, e)));
  }
}

// y[e] <- f(y[e]) in place.
template <class F>
inline void map_(SafeTensors& Y, TensorMeta& ty, F&& f) {
  const int64_t n = ty.numel();
  if (ty.dtype == "F32") {
    float* y = reinterpret_cast<float*>(Y.data.data() + ty.begin);
    for (int64_t e = 0; e < n; ++e) y[e] = f(y[e]);
  } else if (ty.dtype == "BF16") {
    uint16_t* y = reinterpret_cast<uint16_t*>(Y.data.data() + ty.begin);
    for (int64_t e = 0; e < n; ++e)
      y[e] = SafeTensors::f32_to_bf16(f(SafeTensors::bf16_to_f32(y[e])));
  } else {
    for (int64_t e = 0; e < n; ++e) Y.set_elem(ty, e, f(Y.get_elem(ty, e)));
  }
}

// avg <- mean of the tensors across `files` (running mean, matching
// parameter_server.rs:194-209's (a+b)/2 pairwise stream generalized to N).
inline SafeTensors ps_average(const std::vector<std::string>& files) {
  SafeTensors avg = SafeTensors::load(files[0]);
  for (size_t i = 1; i < files.size(); ++i) {
    SafeTensors next = SafeTensors::load(files[i]);
    const float fi = (float)i;
    for (auto& [nm, tm] : avg.tensors)
      zip2_(avg, tm, next, next.tensors.at(nm),
            [fi](float a, float b) { return (a * fi + b) / (fi + 1.f); });
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
  const float w0 = (float)weights[0];
  for (auto& [nm, tm] : avg.tensors) map_(avg, tm, [w0](float a) { return a * w0; });
  for (size_t i = 1; i < files.size(); ++i) {
    SafeTensors next = SafeTensors::load(files[i]);
    const float wi = (float)weights[i];
    for (auto& [nm, tm] : avg.tensors)
      zip2_(avg, tm, next, next.tensors.at(nm),
            [wi](float a, float b) { return a + wi * b; });
    wsum += weights[i];
  }
  const float inv = (float)(1.0 / wsum);
  for (auto& [nm, tm] : avg.tensors) map_(avg, tm, [inv](float a) { return a * inv; });
  return avg;
}

// m <- mu*m + g ; update = lr*(mu*m + g)   (parameter_server.rs:386-446)
// momentum is updated in place; returns the update tensors.
inline SafeTensors ps_nesterov(const SafeTensors& g, SafeTensors& momentum, double lr,
                               double mu) {
  SafeTensors update = g;  // copies shapes + data; overwritten below
  const float lrf = (float)lr, muf = (float)mu;
  for (auto& [nm, tm] : update.tensors) {
    auto& mtm = momentum.tensors.at(nm);
    // two fused passes over (momentum, g-copy): m <- mu*m + g, then
    // update <- lr*(mu*m + g) reading the refreshed momentum
    zip2_(momentum, mtm, update, tm,
          [muf](float m, float gv) { return muf * m + gv; });
    zip2_(update, tm, momentum, mtm,
          [lrf, muf](float gv, float m) { return lrf * (muf * m + gv); });
  }
  return update;
}

// y += x elementwise (the PS's cumulative joiner offset).
inline void ps_add_(SafeTensors& y, const SafeTensors& x) {
  for (auto& [nm, tm] : y.tensors)
    zip2_(y, tm, x, x.tensors.at(nm), [](float a, float b) { return a + b; });
}

inline SafeTensors ps_zeros_like(const SafeTensors& t) {
  SafeTensors z = t;
  for (auto& [nm, tm] : z.tensors) map_(z, tm, [](float) { return 0.f; });
  return z;
}

}  // namespace hypha
