// Pre-bound fused-decide launch plan (shared by orchestrator.cpp and
// bindings.cpp): binds the 33-tensor argument set once so each decide
// re-enqueues with zero pybind argument conversion.
#pragma once
#include <torch/extension.h>

#include <vector>

struct DecidePlan {
  std::vector<torch::Tensor> t;   // fused_decide_launch argument order
  int64_t hash_slots = 0, seg_max = 0, want_pod_pool = 0;
};
