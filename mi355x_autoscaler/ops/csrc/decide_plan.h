// Pre-bound fused-decide launch plan (shared by orchestrator.cpp and
// bindings.cpp): binds the 33-tensor argument set once so each decide
// re-enqueues with zero pybind argument conversion.
#pragma once
#include <torch/extension.h>

#include <vector>

struct DecidePlan {
  std::vector<torch::Tensor> t;   // fused_decide_launch argument order
  int64_t hash_slots = 0, seg_max = 0, want_pod_pool = 0;
  // optional side-stream MFMA demand segsum (pool_demand): bound once,
  // launched by run_decide_plan_auto on the orchestrator's own side
  // stream — the Python stream-context/event dance cost ~28 us/decide
  bool has_side = false;
  torch::Tensor wi_win, wi_start, wi_end;  // cached fastpath work items
  torch::Tensor demand_a, demand_b;        // double-buffered [K,4] i64
  int64_t K = 0;
  int flip = 0;
};
