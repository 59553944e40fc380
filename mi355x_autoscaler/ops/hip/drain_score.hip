// Kernel D — scale-down drain-candidate scoring.
//
// Replaces the reference's per-node x per-pod drainability scan
// [R: autoscaler/cluster.py · maintain node-state loop — reconstructed]
// and adds the PDB budget check the reference lacked (BASELINE.json
// config #4). Two passes:
//
//   drain_flags_kernel: pod-parallel scatter of drain predicates into
//     per-node accumulators (integer atomics — exact, order-free, so
//     sharded ranks just all-reduce their partials):
//       blocked[j]   |= pod pins its node (BLOCKS_DRAIN)
//       pdb_bad[j]   |= pod's PDB has no disruption budget
//       nontrivial[j]+= pod is not mirror/daemonset   (idle = 0)
//       drainable[j] += pod is evictable
//
//   drain_finalize_kernel: node-parallel utilization in integer
//     per-mille of the worst axis (cpu/mem/gpu; the pod-slot axis is
//     occupancy, not utilization) + the candidate predicate.
//
// Node utilization totals (util[N, NRES] int64) come from the
// MFMA-backed segmented-sum kernel (segsum_mfma.hip) over bound pods.
#include "common.h"

extern "C" __global__ void drain_flags_kernel(
    const int32_t* __restrict__ pod_node, const int32_t* __restrict__ pod_flags,
    const int32_t* __restrict__ pod_pdb, const int32_t* __restrict__ pdb_allowed,
    int64_t P,
    int32_t* __restrict__ blocked, int32_t* __restrict__ pdb_bad,
    int32_t* __restrict__ nontrivial, int32_t* __restrict__ drainable) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < P;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int j = pod_node[i];
    if (j < 0) continue;
    const int f = pod_flags[i];
    if (f & PF_BLOCKS_DRAIN) atomicOr(&blocked[j], 1);
    if (!(f & (PF_MIRROR | PF_DAEMONSET))) {
      atomicAdd(&nontrivial[j], 1);
      if (f & PF_DRAINABLE) atomicAdd(&drainable[j], 1);
      // a drain never evicts mirror/daemonset pods, so only evictable
      // pods consume PDB budget (matches the host-side cross-drain
      // accounting and the Python oracle)
      const int d = pod_pdb[i];
      if (d >= 0 && pdb_allowed[d] < 1) atomicOr(&pdb_bad[j], 1);
    }
  }
}

extern "C" __global__ void drain_finalize_kernel(
    const int64_t* __restrict__ util,        // [N, NRES] from segsum_mfma
    const int32_t* __restrict__ node_alloc,  // [N, NRES]
    const int32_t* __restrict__ node_pool,   // [N]
    const int32_t* __restrict__ blocked, const int32_t* __restrict__ pdb_bad,
    const int32_t* __restrict__ nontrivial,
    int32_t N, int32_t util_threshold_permille,
    int32_t* __restrict__ util_permille, int32_t* __restrict__ candidate) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < N;
       j += (int64_t)gridDim.x * blockDim.x) {
    int64_t worst = 0;
#pragma unroll
    for (int r = 0; r < NRES; ++r) {
      if (r == 2) continue;  // pods axis
      const int64_t alloc = node_alloc[j * NRES + r];
      if (alloc > 0) {
        const int64_t u = util[j * NRES + r] * 1000 / alloc;
        worst = u > worst ? u : worst;
      }
    }
    util_permille[j] = (int32_t)worst;
    const bool idle = nontrivial[j] == 0;
    bool cand = !blocked[j] && !pdb_bad[j] && node_pool[j] >= 0;
    if (cand && !idle && worst >= util_threshold_permille) cand = false;
    candidate[j] = cand ? 1 : 0;
  }
}

#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

void drain_flags(torch::Tensor pod_node, torch::Tensor pod_flags,
                 torch::Tensor pod_pdb, torch::Tensor pdb_allowed,
                 torch::Tensor blocked, torch::Tensor pdb_bad,
                 torch::Tensor nontrivial, torch::Tensor drainable) {
  const int64_t P = pod_node.size(0);
  if (P == 0) return;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const int block = 256;
  const int grid = (int)std::min<int64_t>((P + block - 1) / block, 2048);
  hipLaunchKernelGGL(drain_flags_kernel, dim3(grid), dim3(block), 0,
                     stream.stream(), pod_node.data_ptr<int32_t>(),
                     pod_flags.data_ptr<int32_t>(), pod_pdb.data_ptr<int32_t>(),
                     pdb_allowed.data_ptr<int32_t>(), P,
                     blocked.data_ptr<int32_t>(), pdb_bad.data_ptr<int32_t>(),
                     nontrivial.data_ptr<int32_t>(),
                     drainable.data_ptr<int32_t>());
}

void drain_finalize(torch::Tensor util, torch::Tensor node_alloc,
                    torch::Tensor node_pool, torch::Tensor blocked,
                    torch::Tensor pdb_bad, torch::Tensor nontrivial,
                    int64_t util_threshold_permille, torch::Tensor util_permille,
                    torch::Tensor candidate) {
  const int N = (int)node_alloc.size(0);
  if (N == 0) return;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const int block = 256;
  const int grid = std::min((N + block - 1) / block, 2048);
  hipLaunchKernelGGL(drain_finalize_kernel, dim3(grid), dim3(block), 0,
                     stream.stream(), util.data_ptr<int64_t>(),
                     node_alloc.data_ptr<int32_t>(), node_pool.data_ptr<int32_t>(),
                     blocked.data_ptr<int32_t>(), pdb_bad.data_ptr<int32_t>(),
                     nontrivial.data_ptr<int32_t>(), N,
                     (int32_t)util_threshold_permille,
                     util_permille.data_ptr<int32_t>(),
                     candidate.data_ptr<int32_t>());
}
