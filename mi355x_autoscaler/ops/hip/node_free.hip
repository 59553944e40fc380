// Node free-capacity kernel: free = allocatable - Σ (bound pod requests).
//
// Replaces the reference's per-node Python subtraction loop
// [R: autoscaler/cluster.py running-pods-by-node accumulation —
// reconstructed]. Integer atomics are exact and commutative, so the
// result is deterministic under any pod order / shard split; sharded
// ranks sum partial deltas and all-reduce (parallel/engine_sharded.py).
#include "pipeline_device.h"

extern "C" __global__ void node_free_kernel(
    const int32_t* __restrict__ pod_node, const int32_t* __restrict__ pod_req,
    int64_t P, int32_t* __restrict__ node_free) {
  dev_node_usage_subtract(pod_node, pod_req, P, node_free);
}

#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

void node_free_subtract(torch::Tensor pod_node, torch::Tensor pod_req,
                        torch::Tensor node_free) {
  const int64_t P = pod_node.size(0);
  if (P == 0) return;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const int block = 256;
  const int grid = (int)std::min<int64_t>((P + block - 1) / block, 2048);
  hipLaunchKernelGGL(node_free_kernel, dim3(grid), dim3(block), 0,
                     stream.stream(), pod_node.data_ptr<int32_t>(),
                     pod_req.data_ptr<int32_t>(), P,
                     node_free.data_ptr<int32_t>());
}
