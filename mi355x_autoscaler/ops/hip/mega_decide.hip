// Single-launch cooperative decide pipeline ("mega" path).
//
// The classic pipeline enqueues ~13 kernels + memsets per decision;
// after the C++ orchestrator removed host dispatch, the per-launch
// fixed cost on device (~4-5 us each) dominated the small/medium-P
// decision (rocprof: 141 us GPU busy at 10k pods, over half of it
// launch floors of sub-6 us kernels). This kernel runs EVERY phase in
// one hipLaunchCooperativeKernel with grid.sync() barriers:
//
//   P0  zero hash workspace; free = allocatable        (all blocks)
//   P1  hash-aggregate pending pods; subtract usage    (all blocks)
//   P2  compact + bitonic sort segment list            (block 0)
//   P3  eligibility tiles (LDS-staged labels);         (blocks over tiles)
//       segment -> pool routing                        (last block)
//   P4  FFD place + units + readback epilogue          (block 0)
//       pod -> pool map for the MFMA demand segsum     (blocks 1..)
//
// Phase bodies are the SHARED device functions of pipeline_device.h —
// the exact code the standalone kernels run — so mega and classic
// decisions are bit-identical by construction. The classic path
// remains for large P (grid-width bound by co-residency), for K==0 /
// N==0, and as the fallback when cooperative launch is unavailable.
//
// LDS: one dynamic carve, phase-exclusive — compact needs 40 KiB,
// the eligibility tile 65 KiB; by P4 that carve is idle and doubles
// as the units run arena (round 2 — the round-1 global-scratch choice
// predates the wave-parallel array run store).
#include "pipeline_device.h"

#include <hip/hip_cooperative_groups.h>

namespace cg = cooperative_groups;

#define MEGA_BLOCK 256
#define MEGA_SMEM (FIT_TILE * MAXLAB * 8 + FIT_TILE * 4)  // 66560 B

// SINGLE=true compiles the same pipeline as ONE regular (non-
// cooperative) workgroup: every grid.sync() degenerates to
// __syncthreads() and the launch skips hipLaunchCooperativeKernel's
// host-side cost. At small P (config #2: 1k pods, 8 nodes) the
// cooperative grid spends more time in its 4 grid-wide barriers and
// launch setup than in work; one block owns every phase instead.
template <bool SINGLE> struct mega_sync_t;
template <> struct mega_sync_t<false> {
  cg::grid_group g = cg::this_grid();
  __device__ void operator()() { g.sync(); }
};
template <> struct mega_sync_t<true> {
  __device__ void operator()() { __syncthreads(); }
};

template <bool SINGLE>
__global__ __launch_bounds__(MEGA_BLOCK) void mega_decide_kernel_t(
    const int32_t* __restrict__ pod_req, const int32_t* __restrict__ pod_flags,
    const int64_t* __restrict__ pod_sel_hash,
    const int64_t* __restrict__ pod_sel_pairs,
    const int32_t* __restrict__ pod_sel_count,
    const int32_t* __restrict__ pod_node,
    const int32_t* __restrict__ node_alloc,
    const int64_t* __restrict__ node_labels,
    const int32_t* __restrict__ node_label_count,
    const int32_t* __restrict__ node_unsched,
    const int32_t* __restrict__ pool_unit,
    const int64_t* __restrict__ pool_labels,
    const int32_t* __restrict__ pool_label_count,
    const int32_t* __restrict__ pool_max, const int32_t* __restrict__ pool_cur,
    int32_t* __restrict__ ws_i32, int64_t* __restrict__ slot_ks,
    int64_t* __restrict__ slot_kr, int32_t* __restrict__ slot2seg,
    int64_t* __restrict__ seg_sel_pairs, int32_t* __restrict__ seg_sel_count,
    int64_t* __restrict__ seg_sel_hash_ws, int32_t* __restrict__ seg_req,
    int64_t* __restrict__ seg_count, int64_t* __restrict__ seg_key,
    int32_t* __restrict__ seg_first, int32_t* __restrict__ slot_of_pod,
    int32_t* __restrict__ free_, uint64_t* __restrict__ elig,
    int32_t* __restrict__ seg_pool, int64_t* __restrict__ run_scratch,
    int64_t* __restrict__ blob, int32_t* __restrict__ pod_pool,
    int64_t P, int32_t N, int32_t K, int32_t H, int32_t SM, int64_t nblob,
    int32_t RTOT, int32_t want_pod_pool, uint64_t* __restrict__ demand_out)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  mega_sync_t<SINGLE> gsync;
  const int tid = threadIdx.x;
  const int64_t gsz = (int64_t)gridDim.x * blockDim.x;
  const int64_t gtid = (int64_t)blockIdx.x * blockDim.x + tid;

  int32_t* slot_state = ws_i32;
  int32_t* slot_count = ws_i32 + H;
  int32_t* slot_first = ws_i32 + 2 * H;
  int32_t* stat = ws_i32 + 3 * H;  // [G, sort_ovf, hash_ovf]

  // P0: workspace zero + free init
  for (int64_t i = gtid; i < 3 * (int64_t)H + 3; i += gsz) ws_i32[i] = 0;
  for (int64_t i = gtid; i < (int64_t)N * NRES; i += gsz) free_[i] = node_alloc[i];
  gsync();

  // P1: hash-aggregate + usage subtraction. The hash serializes on
  // per-key atomics, so only ~1 block per 1024 pods participates
  // (contention scales with wave count, the scan is bandwidth-trivial;
  // same rule as the standalone launcher); the full grid handles the
  // usage subtraction.
  {
    int hb = (int)((P + 1023) / 1024);
    if (hb < 1) hb = 1;
    if (hb > (int)gridDim.x) hb = gridDim.x;
    if ((int)blockIdx.x < hb) {
      dev_hash_build_small_ex(pod_flags, pod_sel_hash, pod_req, P, H, slot_ks,
                              slot_kr, slot_state, slot_count, slot_first,
                              slot_of_pod, stat + 2,
                              (int64_t)blockIdx.x * blockDim.x + threadIdx.x,
                              (int64_t)hb * blockDim.x);
    }
    dev_node_usage_subtract(pod_node, pod_req, P, free_);
    gsync();
  }

  // P2: compact + sort (block 0; LDS carve: 40 KiB)
  if (blockIdx.x == 0) {
    uint64_t* c_ks = reinterpret_cast<uint64_t*>(smem);
    uint64_t* c_kr = c_ks + SEG_MAX;
    int32_t* c_slot = reinterpret_cast<int32_t*>(c_kr + SEG_MAX);
    int32_t* c_n = c_slot + SEG_MAX;
    dev_compact_sort(slot_ks, slot_kr, slot_state, slot_count, slot_first, H,
                     pod_sel_pairs, pod_sel_count, pod_req, seg_sel_pairs,
                     seg_sel_count, seg_sel_hash_ws, seg_req, seg_count,
                     seg_key, seg_first, slot2seg, stat, c_ks, c_kr, c_slot,
                     c_n);
  }
  gsync();
  const int G = stat[0];

  // P3: eligibility tiles (LDS carve: 65 KiB) + routing on the last block
  {
    int64_t* f_labels = reinterpret_cast<int64_t*>(smem);
    int32_t* f_nlab = reinterpret_cast<int32_t*>(f_labels + FIT_TILE * MAXLAB);
    const int ntiles = (N + FIT_TILE - 1) / FIT_TILE;
    for (int b = blockIdx.x; b < ntiles; b += gridDim.x) {
      dev_fit_tile(b * FIT_TILE, node_labels, node_label_count, node_unsched,
                   seg_sel_pairs, seg_sel_count, N, G, elig, f_labels, f_nlab);
    }
  }
  if (blockIdx.x == gridDim.x - 1) {
    const int wv = tid / WAVE;
    for (int g = wv; g < G; g += MEGA_BLOCK / WAVE)
      dev_route_one(g, K, seg_sel_pairs, seg_sel_count, seg_req, pool_labels,
                    pool_label_count, pool_unit, seg_pool);
  }
  gsync();

  // P4: block 0 runs the serial tail (place -> units -> epilogue).
  // With demand_out set (round 2), ALL blocks first build the
  // pod -> pool map, grid.sync, and the non-zero blocks then compute
  // the MFMA per-pool demand segsum DURING the single-workgroup tail
  // (they were idle after their grid-strided map share); no separate
  // segsum launch or side stream is needed on the mega path.
  if (want_pod_pool && demand_out != nullptr) {
    for (int64_t i = gtid; i < P; i += gsz) {
      const int s = slot_of_pod[i];
      pod_pool[i] = (s >= 0) ? seg_pool[slot2seg[s]] : -1;
    }
    gsync();
    if constexpr (SINGLE) {
      // one block: demand first (tiny at small P), then the tail —
      // the segsum's LDS carve is reused by units right after
      dev_demand_segsum_fastpath(pod_req, pod_pool, P, K, demand_out, smem,
                                 /*chunk_start=*/0, /*chunk_stride=*/1);
      __syncthreads();
    } else if (blockIdx.x > 0 && gridDim.x > 1) {
      dev_demand_segsum_fastpath(
          pod_req, pod_pool, P, K, demand_out, smem,
          (int64_t)(blockIdx.x - 1) * 4, (int64_t)(gridDim.x - 1));
    }
  }
  if constexpr (SINGLE) {
    // no idle blocks exist: the lone block maps pod -> pool itself
    // before the tail when the side-stream segsum will consume it
    if (want_pod_pool && demand_out == nullptr) {
      for (int64_t i = gtid; i < P; i += gsz) {
        const int s = slot_of_pod[i];
        pod_pool[i] = (s >= 0) ? seg_pool[slot2seg[s]] : -1;
      }
      __syncthreads();
    }
  }
  if (blockIdx.x == 0) {
    __shared__ int64_t lds_scan[MEGA_BLOCK / WAVE + 1];
    dev_binpack<MEGA_BLOCK>(free_, elig, seg_req, seg_count, N, G, blob,
                            lds_scan);
    // the 65 KiB dynamic carve is idle by P4 (fit tiles done): hand
    // it to the units run arena (1386 runs) instead of global scratch
    dev_units(seg_pool, seg_req, seg_count, /*placed=*/blob, pool_unit,
              pool_max, pool_cur, G, K, RTOT,
              /*lds_cap=*/(int32_t)(MEGA_SMEM / (6 * sizeof(int64_t))),
              run_scratch,
              /*lds_arena=*/reinterpret_cast<int64_t*>(smem),
              /*units_cnt=*/blob + 3 * (int64_t)SM,
              /*placed_new=*/blob + SM,
              /*unful=*/blob + 2 * (int64_t)SM,
              /*pool_residual=*/blob + 3 * (int64_t)SM + K,
              seg_sel_hash_ws, stat,
              /*blob_mirror=*/blob + (nblob - 2 - SM),
              /*blob_stat=*/reinterpret_cast<int32_t*>(blob + (nblob - 2)), SM);
  } else if (want_pod_pool && demand_out == nullptr) {
    for (int64_t i = (int64_t)(blockIdx.x - 1) * blockDim.x + tid; i < P;
         i += (int64_t)(gridDim.x - 1) * blockDim.x) {
      const int s = slot_of_pod[i];
      pod_pool[i] = (s >= 0) ? seg_pool[slot2seg[s]] : -1;
    }
  }
}

#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

// Returns true when the launch was enqueued; false routes the caller
// to the classic multi-launch pipeline (or, for the single-block
// variant, to the cooperative mega attempt).
static bool mega_decide_launch_impl(
    bool single,
    torch::Tensor pod_req, torch::Tensor pod_flags, torch::Tensor pod_sel_hash,
    torch::Tensor pod_sel_pairs, torch::Tensor pod_sel_count,
    torch::Tensor pod_node, torch::Tensor node_alloc, torch::Tensor node_labels,
    torch::Tensor node_label_count, torch::Tensor node_unsched,
    torch::Tensor pool_unit, torch::Tensor pool_labels,
    torch::Tensor pool_label_count, torch::Tensor pool_max,
    torch::Tensor pool_cur, torch::Tensor ws_i32, torch::Tensor slot_ks,
    torch::Tensor slot_kr, torch::Tensor slot2seg, torch::Tensor seg_sel_pairs,
    torch::Tensor seg_sel_count, torch::Tensor seg_sel_hash,
    torch::Tensor seg_req, torch::Tensor seg_count, torch::Tensor seg_key,
    torch::Tensor seg_first, torch::Tensor slot_of_pod, torch::Tensor free,
    torch::Tensor elig, torch::Tensor seg_pool, torch::Tensor run_scratch,
    torch::Tensor blob, torch::Tensor pod_pool, int64_t hash_slots,
    int64_t seg_max, int64_t want_pod_pool, uint64_t* demand_out) {
  static int coop_ok = -1;
  static int max_coop_blocks = 0;
  if (!single) {
    if (coop_ok < 0) {
      int v = 0;
      if (hipDeviceGetAttribute(&v, hipDeviceAttributeCooperativeLaunch, 0) !=
          hipSuccess) v = 0;
      if (v) {
        int per_cu = 0, ncu = 0;
        if (hipOccupancyMaxActiveBlocksPerMultiprocessor(
                &per_cu,
                reinterpret_cast<const void*>(&mega_decide_kernel_t<false>),
                MEGA_BLOCK, MEGA_SMEM) != hipSuccess ||
            hipDeviceGetAttribute(&ncu, hipDeviceAttributeMultiprocessorCount,
                                  0) != hipSuccess || per_cu <= 0) {
          v = 0;
        } else {
          max_coop_blocks = per_cu * ncu;
        }
      }
      (void)hipGetLastError();  // clear any probe error
      coop_ok = v;
    }
    if (!coop_ok || max_coop_blocks < 2) return false;
  }

  const int64_t P = pod_flags.size(0);
  int32_t N = (int32_t)node_alloc.size(0);
  int32_t K = (int32_t)pool_unit.size(0);
  int32_t H = (int32_t)hash_slots;
  int32_t SM = (int32_t)seg_max;
  int64_t nblob = blob.numel();
  int32_t RTOT = (int32_t)(run_scratch.numel() / 6);
  int32_t wpp = (int32_t)want_pod_pool;

  int grid = (int)((P + MEGA_BLOCK - 1) / MEGA_BLOCK);
  const int ntiles = (N + FIT_TILE - 1) / FIT_TILE;
  if (grid < ntiles + 1) grid = ntiles + 1;
  if (grid < 8) grid = 8;
  if (!single && grid > max_coop_blocks) grid = max_coop_blocks;

  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const int32_t* p_pod_req = pod_req.data_ptr<int32_t>();
  const int32_t* p_pod_flags = pod_flags.data_ptr<int32_t>();
  const int64_t* p_sel_hash = pod_sel_hash.data_ptr<int64_t>();
  const int64_t* p_sel_pairs = pod_sel_pairs.data_ptr<int64_t>();
  const int32_t* p_sel_count = pod_sel_count.data_ptr<int32_t>();
  const int32_t* p_pod_node = pod_node.data_ptr<int32_t>();
  const int32_t* p_alloc = node_alloc.data_ptr<int32_t>();
  const int64_t* p_nlab = node_labels.data_ptr<int64_t>();
  const int32_t* p_nlabc = node_label_count.data_ptr<int32_t>();
  const int32_t* p_unsched = node_unsched.data_ptr<int32_t>();
  const int32_t* p_punit = pool_unit.data_ptr<int32_t>();
  const int64_t* p_plab = pool_labels.data_ptr<int64_t>();
  const int32_t* p_plabc = pool_label_count.data_ptr<int32_t>();
  const int32_t* p_pmax = pool_max.data_ptr<int32_t>();
  const int32_t* p_pcur = pool_cur.data_ptr<int32_t>();
  int32_t* p_ws = ws_i32.data_ptr<int32_t>();
  int64_t* p_sks = slot_ks.data_ptr<int64_t>();
  int64_t* p_skr = slot_kr.data_ptr<int64_t>();
  int32_t* p_s2s = slot2seg.data_ptr<int32_t>();
  int64_t* p_ssp = seg_sel_pairs.data_ptr<int64_t>();
  int32_t* p_ssc = seg_sel_count.data_ptr<int32_t>();
  int64_t* p_ssh = seg_sel_hash.data_ptr<int64_t>();
  int32_t* p_sreq = seg_req.data_ptr<int32_t>();
  int64_t* p_scnt = seg_count.data_ptr<int64_t>();
  int64_t* p_skey = seg_key.data_ptr<int64_t>();
  int32_t* p_sfst = seg_first.data_ptr<int32_t>();
  int32_t* p_sop = slot_of_pod.data_ptr<int32_t>();
  int32_t* p_free = free.data_ptr<int32_t>();
  uint64_t* p_elig = reinterpret_cast<uint64_t*>(elig.data_ptr<int64_t>());
  int32_t* p_spool = seg_pool.data_ptr<int32_t>();
  int64_t* p_scratch = run_scratch.data_ptr<int64_t>();
  int64_t* p_blob = blob.data_ptr<int64_t>();
  int32_t* p_ppool = wpp ? pod_pool.data_ptr<int32_t>() : nullptr;
  uint64_t* p_demand = wpp ? demand_out : nullptr;

  void* args[] = {
      (void*)&p_pod_req, (void*)&p_pod_flags, (void*)&p_sel_hash,
      (void*)&p_sel_pairs, (void*)&p_sel_count, (void*)&p_pod_node,
      (void*)&p_alloc, (void*)&p_nlab, (void*)&p_nlabc, (void*)&p_unsched,
      (void*)&p_punit, (void*)&p_plab, (void*)&p_plabc, (void*)&p_pmax,
      (void*)&p_pcur, (void*)&p_ws, (void*)&p_sks, (void*)&p_skr,
      (void*)&p_s2s, (void*)&p_ssp, (void*)&p_ssc, (void*)&p_ssh,
      (void*)&p_sreq, (void*)&p_scnt, (void*)&p_skey, (void*)&p_sfst,
      (void*)&p_sop, (void*)&p_free, (void*)&p_elig, (void*)&p_spool,
      (void*)&p_scratch, (void*)&p_blob, (void*)&p_ppool,
      (void*)&P, (void*)&N, (void*)&K, (void*)&H, (void*)&SM, (void*)&nblob,
      (void*)&RTOT, (void*)&wpp, (void*)&p_demand,
  };
  if (single) {
    static int single_ok = -1;
    if (single_ok == 0) return false;
    hipError_t e = hipLaunchKernel(
        reinterpret_cast<const void*>(&mega_decide_kernel_t<true>), dim3(1),
        dim3(MEGA_BLOCK), args, MEGA_SMEM, stream.stream());
    if (e != hipSuccess) {
      (void)hipGetLastError();
      single_ok = 0;  // don't retry a failing configuration
      return false;
    }
    single_ok = 1;
    return true;
  }
  hipError_t e = hipLaunchCooperativeKernel(
      reinterpret_cast<const void*>(&mega_decide_kernel_t<false>), dim3(grid),
      dim3(MEGA_BLOCK), args, MEGA_SMEM, stream.stream());
  if (e != hipSuccess) {
    (void)hipGetLastError();
    coop_ok = 0;  // don't retry a failing configuration
    return false;
  }
  return true;
}

#define MEGA_FWD_ARGS                                                        \
  pod_req, pod_flags, pod_sel_hash, pod_sel_pairs, pod_sel_count, pod_node,  \
      node_alloc, node_labels, node_label_count, node_unsched, pool_unit,    \
      pool_labels, pool_label_count, pool_max, pool_cur, ws_i32, slot_ks,    \
      slot_kr, slot2seg, seg_sel_pairs, seg_sel_count, seg_sel_hash,         \
      seg_req, seg_count, seg_key, seg_first, slot_of_pod, free, elig,       \
      seg_pool, run_scratch, blob, pod_pool, hash_slots, seg_max,            \
      want_pod_pool, demand_out

#define MEGA_FWD_PARAMS                                                       \
  torch::Tensor pod_req, torch::Tensor pod_flags, torch::Tensor pod_sel_hash, \
      torch::Tensor pod_sel_pairs, torch::Tensor pod_sel_count,               \
      torch::Tensor pod_node, torch::Tensor node_alloc,                       \
      torch::Tensor node_labels, torch::Tensor node_label_count,              \
      torch::Tensor node_unsched, torch::Tensor pool_unit,                    \
      torch::Tensor pool_labels, torch::Tensor pool_label_count,              \
      torch::Tensor pool_max, torch::Tensor pool_cur, torch::Tensor ws_i32,   \
      torch::Tensor slot_ks, torch::Tensor slot_kr, torch::Tensor slot2seg,   \
      torch::Tensor seg_sel_pairs, torch::Tensor seg_sel_count,               \
      torch::Tensor seg_sel_hash, torch::Tensor seg_req,                      \
      torch::Tensor seg_count, torch::Tensor seg_key, torch::Tensor seg_first,\
      torch::Tensor slot_of_pod, torch::Tensor free, torch::Tensor elig,      \
      torch::Tensor seg_pool, torch::Tensor run_scratch, torch::Tensor blob,  \
      torch::Tensor pod_pool, int64_t hash_slots, int64_t seg_max,            \
      int64_t want_pod_pool, uint64_t* demand_out

bool mega_decide_launch(MEGA_FWD_PARAMS) {
  return mega_decide_launch_impl(false, MEGA_FWD_ARGS);
}

// Single-workgroup variant: regular (non-cooperative) launch, one
// block runs every phase with __syncthreads barriers. For the small-P
// regime where grid-wide sync + cooperative-launch setup dominate.
bool mega_decide_single_launch(MEGA_FWD_PARAMS) {
  return mega_decide_launch_impl(true, MEGA_FWD_ARGS);
}
