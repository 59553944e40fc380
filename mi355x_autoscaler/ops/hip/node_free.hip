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

// Per-node int64 usage totals (maintain utilization): the sorted-MFMA
// segsum path cost a 10k-element radix sort + ~40 torch dispatches per
// maintain; bound-pod usage is a plain commutative accumulation, so
// one atomic pass is exact, deterministic, and shard-composable (the
// MFMA segmented sums remain on the decide demand path).
extern "C" __global__ void node_usage_kernel(
    const int32_t* __restrict__ pod_node, const int32_t* __restrict__ pod_req,
    int64_t P, unsigned long long* __restrict__ usage) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < P;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int j = pod_node[i];
    if (j < 0) continue;
#pragma unroll
    for (int r = 0; r < NRES; ++r) {
      const int32_t v = pod_req[i * NRES + r];
      if (v > 0) atomicAdd(&usage[(int64_t)j * NRES + r], (unsigned long long)v);
    }
  }
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

void node_usage_sum(torch::Tensor pod_node, torch::Tensor pod_req,
                    torch::Tensor usage) {
  const int64_t P = pod_node.size(0);
  if (P == 0) return;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const int block = 256;
  const int grid = (int)std::min<int64_t>((P + block - 1) / block, 2048);
  hipLaunchKernelGGL(node_usage_kernel, dim3(grid), dim3(block), 0,
                     stream.stream(), pod_node.data_ptr<int32_t>(),
                     pod_req.data_ptr<int32_t>(), P,
                     reinterpret_cast<unsigned long long*>(usage.data_ptr<int64_t>()));
}
