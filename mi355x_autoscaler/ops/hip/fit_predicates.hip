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
#include "common.h"

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

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;
  const int node0 = blockIdx.x * NODES_PER_BLOCK;
  const int my_node = node0 + tid;
  const int W = (N + WAVE - 1) / WAVE;

  // stage this block's node labels into LDS, b128 rows: thread t copies
  // node t's row (coalesced across the row via 4-wide int64 chunks).
  if (my_node < N) {
    const int64_t* src = node_labels + (int64_t)my_node * MAXLAB;
    int64_t* dst = lds_labels + (int64_t)tid * MAXLAB;
#pragma unroll
    for (int q = 0; q < MAXLAB; ++q) dst[q] = src[q];
    int nl = node_label_count[my_node];
    // fold schedulability into the count: unschedulable => impossible
    lds_nlab[tid] = node_unsched[my_node] ? -1 : nl;
  } else {
    lds_nlab[tid] = -1;
  }
  __syncthreads();

  // each wave owns word (node0/64 + wid); loop over all segments
  const int my_word = node0 / WAVE + wid;
  const int lds_off = wid * WAVE + lane;  // this lane's node slot in LDS

  for (int g = 0; g < G; ++g) {
    const int nsel = seg_sel_count[g];
    const int64_t* sel = seg_sel + (int64_t)g * MAXSEL;  // wave-uniform
    bool ok = false;
    const int nlab = lds_nlab[lds_off];
    if (nlab >= 0) {
      ok = sel_subset(sel, nsel, lds_labels + (int64_t)lds_off * MAXLAB, nlab);
    }
    const uint64_t ballot = __ballot(ok);
    if (lane == 0 && my_word < W) {
      elig[(int64_t)g * W + my_word] = ballot;
    }
  }
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
  const int lane = threadIdx.x & (WAVE - 1);
  const int nsel = seg_sel_count[g];
  const int64_t* sel = seg_sel + (int64_t)g * MAXSEL;
  int32_t rq[NRES];
#pragma unroll
  for (int r = 0; r < NRES; ++r) rq[r] = seg_req[g * NRES + r];

  int best = INT32_MAX;
  for (int k0 = 0; k0 < K; k0 += WAVE) {
    const int k = k0 + lane;
    bool ok = false;
    if (k < K) {
      ok = sel_subset(sel, nsel, pool_labels + (int64_t)k * MAXLAB,
                      pool_label_count[k]);
      if (ok) {
#pragma unroll
        for (int r = 0; r < NRES; ++r) ok &= rq[r] <= pool_unit[k * NRES + r];
      }
    }
    const uint64_t ballot = __ballot(ok);
    if (ballot) {
      best = k0 + __ffsll((unsigned long long)ballot) - 1;
      break;
    }
  }
  if (lane == 0) seg_pool[g] = best == INT32_MAX ? -1 : best;
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
