// Kernel B — first-fit-decreasing bin-packing as uniform placement
// steps + the virtual-new-node (units-needed) phase.
//
// Replaces the reference's O(pods x nodes) interpreted placement loop
// [R: autoscaler/cluster.py · fulfill_pending accumulation —
// reconstructed]. The oracle (sim/oracle.py) restates FFD as one
// uniform step per segment:
//
//   m_j   = min_r floor(free[j,r] / req[r])   (eligible nodes only)
//   scan  = exclusive prefix sum of m_j in node order
//   place_j = clamp(count - scan_j, 0, m_j);  free[j] -= place_j * req
//
// Segments have a loop-carried dependency through `free`, so ONE
// persistent workgroup walks the segment list in order; within a step
// all lanes work node-parallel with a block-wide int64 scan in LDS.
// `free` stays in global memory (int32, L2-resident at 16 B/node). The
// block size is dispatched on N: a 100-node cluster pays 2-wave
// barriers, not 16-wave ones (the fixed-1024 form spent half its time
// in __syncthreads at small N).
//
// Decision-critical property: identical arithmetic + identical order
// to the oracle => bit-identical placements, on 1 GPU or 8 (segment
// COUNTS are all-reduced before this kernel; the node table math is
// replicated per rank).
#include "pipeline_device.h"

template <int BLOCK>
__global__ __launch_bounds__(BLOCK) void binpack_kernel_t(
    int32_t* __restrict__ free_,                 // [N, NRES] mutated
    const uint64_t* __restrict__ elig,           // [G, W]
    const int32_t* __restrict__ seg_req,         // [G, NRES]
    const int64_t* __restrict__ seg_count,       // [G]
    int32_t N, const int32_t* __restrict__ g_ptr,
    int64_t* __restrict__ placed_out)            // [G]
{
  __shared__ int64_t lds_scan[BLOCK / WAVE + 1];
  dev_binpack<BLOCK>(free_, elig, seg_req, seg_count, N, *g_ptr, placed_out,
                     lds_scan);
}

// Units phase — see pipeline_device.h (dev_units) for the run-length
// FFD algorithm, the wave-parallel pool ownership, the inline max-new
// and the fused readback epilogue. This wrapper carves the LDS arena
// tier; the mega path calls dev_units with the global arena instead.
extern "C" __global__ __launch_bounds__(UNITS_BLOCK) void units_kernel(
    const int32_t* __restrict__ seg_pool, const int32_t* __restrict__ seg_req,
    const int64_t* __restrict__ seg_count, const int64_t* __restrict__ placed,
    const int32_t* __restrict__ pool_unit,      // [K, NRES]
    const int32_t* __restrict__ pool_max, const int32_t* __restrict__ pool_cur,
    const int32_t* __restrict__ g_ptr, int32_t K, int32_t RTOT, int32_t lds_cap,
    int64_t* __restrict__ scratch,              // [RTOT * 6] global arena
    int64_t* __restrict__ units_cnt,            // [K] out
    int64_t* __restrict__ placed_new, int64_t* __restrict__ unful,
    int64_t* __restrict__ pool_residual,        // [K, NRES] out
    const int64_t* __restrict__ seg_sel_hash, const int32_t* __restrict__ stat,
    int64_t* __restrict__ blob_mirror, int32_t* __restrict__ blob_stat,
    int32_t SM)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  dev_units(seg_pool, seg_req, seg_count, placed, pool_unit, pool_max,
            pool_cur, *g_ptr, K, RTOT, lds_cap,
            scratch, reinterpret_cast<int64_t*>(smem),
            units_cnt, placed_new, unful, pool_residual,
            seg_sel_hash, stat, blob_mirror, blob_stat, SM);
}

#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

void binpack_place(torch::Tensor free, torch::Tensor elig, torch::Tensor seg_req,
                   torch::Tensor seg_count, torch::Tensor g_dev,
                   torch::Tensor placed_out) {
  const int N = (int)free.size(0);
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  // block dispatched on N: barrier cost scales with resident waves
  void (*kfn)(int32_t*, const uint64_t*, const int32_t*, const int64_t*,
              int32_t, const int32_t*, int64_t*) = binpack_kernel_t<1024>;
  int block = 1024;
  if (N <= 128) {
    kfn = binpack_kernel_t<128>;
    block = 128;
  } else if (N <= 512) {
    kfn = binpack_kernel_t<512>;
    block = 512;
  }
  hipLaunchKernelGGL(kfn, dim3(1), dim3(block), 0, stream.stream(),
                     free.data_ptr<int32_t>(),
                     reinterpret_cast<uint64_t*>(elig.data_ptr<int64_t>()),
                     seg_req.data_ptr<int32_t>(), seg_count.data_ptr<int64_t>(),
                     N, g_dev.data_ptr<int32_t>(), placed_out.data_ptr<int64_t>());
}

void binpack_units(torch::Tensor seg_pool, torch::Tensor seg_req,
                   torch::Tensor seg_count, torch::Tensor placed,
                   torch::Tensor pool_unit, torch::Tensor pool_max,
                   torch::Tensor pool_cur, torch::Tensor g_dev,
                   torch::Tensor run_scratch, torch::Tensor units_cnt,
                   torch::Tensor placed_new, torch::Tensor unful,
                   torch::Tensor pool_residual,
                   c10::optional<torch::Tensor> seg_sel_hash,
                   c10::optional<torch::Tensor> stat,
                   c10::optional<torch::Tensor> blob) {
  const int K = (int)pool_unit.size(0);
  TORCH_CHECK(K <= UNITS_MAX_POOLS, "units_kernel supports at most 64 pools");
  const int RTOT = (int)(run_scratch.numel() / 6);
  TORCH_CHECK(RTOT >= 4 * (int)seg_req.size(0) + UNITS_PAD * UNITS_NW,
              "run arena must hold 4*G + pad*NW runs");
  const int lds_cap = 1300;  // 1300 runs * 48 B = 62.4 KiB carve
  const size_t lds_bytes = (size_t)lds_cap * 6 * sizeof(int64_t);
  const int64_t* sel_ptr = nullptr;
  const int32_t* stat_ptr = nullptr;
  int64_t* mirror_ptr = nullptr;
  int32_t* bstat_ptr = nullptr;
  int32_t SM = 0;
  if (blob.has_value()) {
    TORCH_CHECK(seg_sel_hash.has_value() && stat.has_value(),
                "epilogue needs seg_sel_hash and stat with blob");
    SM = (int32_t)seg_sel_hash->numel();
    auto* b = blob->data_ptr<int64_t>();
    const int64_t nb = blob->numel();
    mirror_ptr = b + (nb - 2 - SM);
    bstat_ptr = reinterpret_cast<int32_t*>(b + (nb - 2));
    sel_ptr = seg_sel_hash->data_ptr<int64_t>();
    stat_ptr = stat->data_ptr<int32_t>();
  }
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  hipLaunchKernelGGL(units_kernel, dim3(1), dim3(UNITS_BLOCK), lds_bytes,
                     stream.stream(),
                     seg_pool.data_ptr<int32_t>(), seg_req.data_ptr<int32_t>(),
                     seg_count.data_ptr<int64_t>(), placed.data_ptr<int64_t>(),
                     pool_unit.data_ptr<int32_t>(),
                     pool_max.data_ptr<int32_t>(), pool_cur.data_ptr<int32_t>(),
                     g_dev.data_ptr<int32_t>(), K, RTOT, lds_cap,
                     run_scratch.data_ptr<int64_t>(), units_cnt.data_ptr<int64_t>(),
                     placed_new.data_ptr<int64_t>(), unful.data_ptr<int64_t>(),
                     pool_residual.data_ptr<int64_t>(),
                     sel_ptr, stat_ptr, mirror_ptr, bstat_ptr, SM);
}
