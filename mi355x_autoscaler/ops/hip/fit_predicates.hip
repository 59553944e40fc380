// Kernel A — pod-group -> node eligibility predicates, LDS-staged.
//
// Replaces the reference hot loop's per-(pod,node) selector match +
// schedulability check [R: autoscaler/cluster.py · fulfill_pending
// inner loop; autoscaler/kube.py selector match — reconstructed].
//
// Layout: each 256-thread block stages the label hash-sets of 256
// nodes into LDS (256 x MAXLAB x 8 B = 64 KiB) once, then loops over
// ALL segments; each of the 4 waves owns one 64-node word and emits a
// 64-bit eligibility ballot per segment:
//
//   elig[g][word] bit j = !unschedulable[node] &&
//                         selector(g) ⊆ labels[node]
//
// The segment selector array (<= 8 x i64) is wave-uniform, served by
// the scalar cache; the per-lane label walk hits LDS only. This is the
// "per-wavefront pod/node slots staged in LDS" structure the north
// star names, with nodes resident across the whole segment loop so
// HBM label traffic is paid once per block, not once per segment.
//
// Grid: ceil(N/256) blocks. Output: [G, W] int64 (W = ceil(N/64)).
#include "pipeline_device.h"

#define FIT_BLOCK 256
#define NODES_PER_BLOCK FIT_BLOCK

extern "C" __global__ __launch_bounds__(FIT_BLOCK) void fit_eligibility_kernel(
    const int64_t* __restrict__ node_labels,      // [N, MAXLAB] sorted
    const int32_t* __restrict__ node_label_count, // [N]
    const int32_t* __restrict__ node_unsched,     // [N]
    const int64_t* __restrict__ seg_sel,          // [G, MAXSEL] sorted
    const int32_t* __restrict__ seg_sel_count,    // [G]
    int32_t N, const int32_t* __restrict__ g_ptr,
    uint64_t* __restrict__ elig)                  // [G, W]
{
  const int G = *g_ptr;  // device-resident so the host never syncs for it
  __shared__ int64_t lds_labels[NODES_PER_BLOCK * MAXLAB];
  __shared__ int32_t lds_nlab[NODES_PER_BLOCK];
  dev_fit_tile(blockIdx.x * NODES_PER_BLOCK, node_labels, node_label_count,
               node_unsched, seg_sel, seg_sel_count, N, G, elig, lds_labels,
               lds_nlab);
}

// Segment -> agent-pool routing: for each segment, the FIRST pool
// (ascending index == sorted pool name, matching the oracle) whose
// prototype labels contain the selector AND whose unit capacity fits
// one pod. One wave per segment; lanes fan out over pools.
// [R: autoscaler/cluster.py units-needed pool choice — reconstructed]
extern "C" __global__ void seg_pool_kernel(
    const int64_t* __restrict__ seg_sel, const int32_t* __restrict__ seg_sel_count,
    const int32_t* __restrict__ seg_req,          // [G, NRES]
    const int64_t* __restrict__ pool_labels,      // [K, MAXLAB]
    const int32_t* __restrict__ pool_label_count, // [K]
    const int32_t* __restrict__ pool_unit,        // [K, NRES]
    const int32_t* __restrict__ g_ptr, int32_t K,
    int32_t* __restrict__ seg_pool)               // [G] first pool or -1
{
  const int g = blockIdx.x;
  if (g >= *g_ptr) return;
  dev_route_one(g, K, seg_sel, seg_sel_count, seg_req, pool_labels,
                pool_label_count, pool_unit, seg_pool);
}

#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

void fit_eligibility(torch::Tensor node_labels, torch::Tensor node_label_count,
                     torch::Tensor node_unsched, torch::Tensor seg_sel,
                     torch::Tensor seg_sel_count, torch::Tensor g_dev,
                     torch::Tensor elig) {
  const int N = (int)node_labels.size(0);
  if (N == 0) return;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const int grid = (N + NODES_PER_BLOCK - 1) / NODES_PER_BLOCK;
  hipLaunchKernelGGL(fit_eligibility_kernel, dim3(grid), dim3(FIT_BLOCK), 0,
                     stream.stream(), node_labels.data_ptr<int64_t>(),
                     node_label_count.data_ptr<int32_t>(),
                     node_unsched.data_ptr<int32_t>(),
                     seg_sel.data_ptr<int64_t>(),
                     seg_sel_count.data_ptr<int32_t>(), N,
                     g_dev.data_ptr<int32_t>(),
                     reinterpret_cast<uint64_t*>(elig.data_ptr<int64_t>()));
}

void seg_pool_route(torch::Tensor seg_sel, torch::Tensor seg_sel_count,
                    torch::Tensor seg_req, torch::Tensor pool_labels,
                    torch::Tensor pool_label_count, torch::Tensor pool_unit,
                    torch::Tensor g_dev, torch::Tensor seg_pool) {
  const int Gmax = (int)seg_sel.size(0);
  const int K = (int)pool_labels.size(0);
  if (Gmax == 0) return;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  hipLaunchKernelGGL(seg_pool_kernel, dim3(Gmax), dim3(WAVE), 0, stream.stream(),
                     seg_sel.data_ptr<int64_t>(),
                     seg_sel_count.data_ptr<int32_t>(),
                     seg_req.data_ptr<int32_t>(),
                     pool_labels.data_ptr<int64_t>(),
                     pool_label_count.data_ptr<int32_t>(),
                     pool_unit.data_ptr<int32_t>(), g_dev.data_ptr<int32_t>(), K,
                     seg_pool.data_ptr<int32_t>());
}
