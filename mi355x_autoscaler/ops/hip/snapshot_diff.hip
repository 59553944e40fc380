// Kernel E — delta-apply watch events into the HBM-resident snapshot.
//
// [SURVEY §2.2 row E]: instead of re-uploading the full pod SoA every
// cycle, the host watch cache (snapshot/delta.py) tracks the rows
// touched since the last decision and this kernel scatters just those
// rows into the resident device arrays. Update payload is a compact
// [M]-row SoA (same column layout); uid-hash-ordered row indices make
// the writes coalesce within warps for bursty (append-heavy) deltas.
#include "common.h"

extern "C" __global__ void snapshot_scatter_kernel(
    const int64_t* __restrict__ rows, int64_t M,
    const int32_t* __restrict__ u_req,       // [M, NRES]
    const int32_t* __restrict__ u_flags,     // [M]
    const int64_t* __restrict__ u_sel_hash,  // [M]
    const int64_t* __restrict__ u_sel_pairs, // [M, MAXSEL]
    const int32_t* __restrict__ u_sel_count, // [M]
    const int32_t* __restrict__ u_node,      // [M]
    const int32_t* __restrict__ u_pdb,       // [M]
    int32_t* __restrict__ pod_req, int32_t* __restrict__ pod_flags,
    int64_t* __restrict__ pod_sel_hash, int64_t* __restrict__ pod_sel_pairs,
    int32_t* __restrict__ pod_sel_count, int32_t* __restrict__ pod_node,
    int32_t* __restrict__ pod_pdb) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < M;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t r = rows[i];
#pragma unroll
    for (int q = 0; q < NRES; ++q) pod_req[r * NRES + q] = u_req[i * NRES + q];
    pod_flags[r] = u_flags[i];
    pod_sel_hash[r] = u_sel_hash[i];
#pragma unroll
    for (int q = 0; q < MAXSEL; ++q)
      pod_sel_pairs[r * MAXSEL + q] = u_sel_pairs[i * MAXSEL + q];
    pod_sel_count[r] = u_sel_count[i];
    pod_node[r] = u_node[i];
    pod_pdb[r] = u_pdb[i];
  }
}

#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

void snapshot_scatter(torch::Tensor rows, torch::Tensor u_req,
                      torch::Tensor u_flags, torch::Tensor u_sel_hash,
                      torch::Tensor u_sel_pairs, torch::Tensor u_sel_count,
                      torch::Tensor u_node, torch::Tensor u_pdb,
                      torch::Tensor pod_req, torch::Tensor pod_flags,
                      torch::Tensor pod_sel_hash, torch::Tensor pod_sel_pairs,
                      torch::Tensor pod_sel_count, torch::Tensor pod_node,
                      torch::Tensor pod_pdb) {
  const int64_t M = rows.size(0);
  if (M == 0) return;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const int block = 256;
  const int grid = (int)std::min<int64_t>((M + block - 1) / block, 2048);
  hipLaunchKernelGGL(snapshot_scatter_kernel, dim3(grid), dim3(block), 0,
                     stream.stream(), rows.data_ptr<int64_t>(), M,
                     u_req.data_ptr<int32_t>(), u_flags.data_ptr<int32_t>(),
                     u_sel_hash.data_ptr<int64_t>(), u_sel_pairs.data_ptr<int64_t>(),
                     u_sel_count.data_ptr<int32_t>(), u_node.data_ptr<int32_t>(),
                     u_pdb.data_ptr<int32_t>(), pod_req.data_ptr<int32_t>(),
                     pod_flags.data_ptr<int32_t>(), pod_sel_hash.data_ptr<int64_t>(),
                     pod_sel_pairs.data_ptr<int64_t>(), pod_sel_count.data_ptr<int32_t>(),
                     pod_node.data_ptr<int32_t>(), pod_pdb.data_ptr<int32_t>());
}
