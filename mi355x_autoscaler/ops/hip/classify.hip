// Kernel C — pod classification / sort-key building.
//
// Replaces the reference's per-cycle pod partitioning + selector-hash
// grouping [R: autoscaler/cluster.py pod partitioning, autoscaler/
// utils.py selectors_to_hash — reconstructed]. One elementwise pass
// over the pod SoA emits, for every pod, the two radix keys whose
// ascending sort order realizes the oracle's global segment order
// (selector-set hash as uint64 ASC, request key DESC):
//
//   key_sel = sel_hash ^ INT64_MIN          (signed order == uint64 order)
//             or INT64_MAX for non-pending pods (sort to the tail)
//   key_req = PACK_MAX - pack_req_key(req)  (ascending == request desc)
//
// The segment build itself (sort + unique) runs on-device via torch's
// radix sort in ops/engine_gpu.py; this kernel keeps the key math in
// one place, bit-identical to sim/oracle.py::build_segments.
#include "common.h"

#define PACK_MAX ((int64_t)1 << 56)

extern "C" __global__ void classify_keys_kernel(
    const int32_t* __restrict__ pod_flags, const int64_t* __restrict__ sel_hash,
    const int32_t* __restrict__ pod_req, int64_t P,
    int64_t* __restrict__ key_sel, int64_t* __restrict__ key_req,
    int32_t* __restrict__ pending_flag) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < P;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int f = pod_flags[i];
    const bool pending = (f & PF_PENDING) != 0;
    pending_flag[i] = pending ? 1 : 0;
    if (pending) {
      key_sel[i] = sel_hash[i] ^ (int64_t)0x8000000000000000LL;
      int32_t rq[NRES];
#pragma unroll
      for (int r = 0; r < NRES; ++r) rq[r] = pod_req[i * NRES + r];
      key_req[i] = PACK_MAX - pack_req_key(rq, f & PF_CRITICAL);
    } else {
      key_sel[i] = INT64_MAX;
      key_req[i] = INT64_MAX;
    }
  }
}

#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

void classify_keys(torch::Tensor pod_flags, torch::Tensor sel_hash,
                   torch::Tensor pod_req, torch::Tensor key_sel,
                   torch::Tensor key_req, torch::Tensor pending_flag) {
  const int64_t P = pod_flags.size(0);
  if (P == 0) return;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const int block = 256;
  const int grid = (int)std::min<int64_t>((P + block - 1) / block, 2048);
  hipLaunchKernelGGL(classify_keys_kernel, dim3(grid), dim3(block), 0,
                     stream.stream(), pod_flags.data_ptr<int32_t>(),
                     sel_hash.data_ptr<int64_t>(), pod_req.data_ptr<int32_t>(),
                     P, key_sel.data_ptr<int64_t>(), key_req.data_ptr<int64_t>(),
                     pending_flag.data_ptr<int32_t>());
}
