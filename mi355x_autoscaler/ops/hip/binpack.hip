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
// persistent workgroup (1024 threads, 16 waves) walks the segment list
// in order; within a step all 1024 lanes work node-parallel with a
// block-wide int64 scan in LDS. `free` stays in global memory (int32,
// L2-resident at 16 B/node); a 10k-node, 100-segment decision is
// ~1000 chunk iterations of pure integer work.
//
// Decision-critical property: identical arithmetic + identical order
// to the oracle => bit-identical placements, on 1 GPU or 8 (segment
// COUNTS are all-reduced before this kernel; the node table math is
// replicated per rank).
#include "common.h"

#define BP_BLOCK 1024
#define UNITS_MAX_POOLS 64

extern "C" __global__ __launch_bounds__(BP_BLOCK) void binpack_kernel(
    int32_t* __restrict__ free_,                 // [N, NRES] mutated
    const uint64_t* __restrict__ elig,           // [G, W]
    const int32_t* __restrict__ seg_req,         // [G, NRES]
    const int64_t* __restrict__ seg_count,       // [G]
    int32_t N, int32_t G,
    int64_t* __restrict__ placed_out)            // [G]
{
  __shared__ int64_t lds_scan[BP_BLOCK / WAVE + 1];
  const int tid = threadIdx.x;
  const int W = (N + WAVE - 1) / WAVE;

  for (int g = 0; g < G; ++g) {
    const int64_t need = seg_count[g];
    if (need <= 0) {
      if (tid == 0) placed_out[g] = 0;
      continue;
    }
    int32_t rq[NRES];
#pragma unroll
    for (int r = 0; r < NRES; ++r) rq[r] = seg_req[g * NRES + r];
    const uint64_t* erow = elig + (int64_t)g * W;

    int64_t carry = 0;  // pods placeable on nodes [0, base)
    for (int base = 0; base < N && carry < need; base += BP_BLOCK) {
      const int j = base + tid;
      int64_t m = 0;
      int32_t f[NRES];
      if (j < N && ((erow[j / WAVE] >> (j & (WAVE - 1))) & 1ull)) {
#pragma unroll
        for (int r = 0; r < NRES; ++r) f[r] = free_[(int64_t)j * NRES + r];
        m = per_node_capacity(f, rq);
      }
      int64_t chunk_total;
      const int64_t scan = block_exclusive_scan_i64<BP_BLOCK>(m, lds_scan, &chunk_total);
      const int64_t offset = carry + scan;
      int64_t take = need - offset;
      take = take < 0 ? 0 : (take > m ? m : take);
      if (take > 0) {
#pragma unroll
        for (int r = 0; r < NRES; ++r) {
          free_[(int64_t)j * NRES + r] = f[r] - (int32_t)(take * rq[r]);
        }
      }
      carry += chunk_total;
    }
    if (tid == 0) placed_out[g] = carry < need ? carry : need;
    __syncthreads();
  }
}

// Units phase — residual pods bin-pack into virtual new nodes of the
// segment's routed pool, FFD in global segment order, capped at the
// pool's max size [R: autoscaler/cluster.py units-needed + agent_pool
// max — reconstructed]. Work is tiny (G x <=max_new units); one wave
// runs it unit-parallel with the same scan-threshold math the oracle
// uses sequentially, keeping device-resident state so the decision
// vector (units per pool) never round-trips to the host before the
// RCCL all-reduce.
//
//   units_free: [K, MAXU, NRES] int64 scratch (open virtual nodes)
//   units_cnt:  [K] int32 (doubles as units_needed output)
//   pool_residual: [K, NRES] int64 residual demand totals
extern "C" __global__ __launch_bounds__(WAVE) void units_kernel(
    const int32_t* __restrict__ seg_pool, const int32_t* __restrict__ seg_req,
    const int64_t* __restrict__ seg_count, const int64_t* __restrict__ placed,
    const int32_t* __restrict__ pool_unit,      // [K, NRES]
    const int32_t* __restrict__ pool_max_new,   // [K]
    int32_t G, int32_t K, int32_t MAXU, int32_t use_lds,
    int64_t* __restrict__ units_free, int64_t* __restrict__ units_cnt,
    int64_t* __restrict__ placed_new, int64_t* __restrict__ unful,
    int64_t* __restrict__ pool_residual)
{
  // units_free normally lives in LDS (K*MAXU*NRES int64; the engine
  // sets use_lds=0 when that exceeds the carve) — the kernel is a
  // single serial wave, so global-memory unit state was pure L2
  // round-trip latency (29 us/decide in the round-1 profile).
  extern __shared__ __attribute__((aligned(16))) char smem[];
  int64_t* U = use_lds ? reinterpret_cast<int64_t*>(smem) : units_free;
  __shared__ int64_t lds_resid[UNITS_MAX_POOLS * NRES];
  __shared__ int32_t lds_cnt[UNITS_MAX_POOLS];

  const int lane = threadIdx.x & (WAVE - 1);
  for (int k = lane; k < K; k += WAVE) lds_cnt[k] = 0;
  for (int q = lane; q < K * NRES; q += WAVE) lds_resid[q] = 0;
  __syncthreads();

  for (int g = 0; g < G; ++g) {
    int64_t rem = seg_count[g] - placed[g];
    if (lane == 0) {
      placed_new[g] = 0;
      unful[g] = 0;
    }
    if (rem <= 0) continue;
    const int k = seg_pool[g];
    if (k < 0) {
      if (lane == 0) unful[g] = rem;
      continue;
    }
    int64_t rq[NRES];
#pragma unroll
    for (int r = 0; r < NRES; ++r) rq[r] = seg_req[g * NRES + r];
    if (lane < NRES) {
      lds_resid[k * NRES + lane] += rq[lane] * rem;
    }
    const int64_t need0 = rem;

    // 1) fill already-open units in creation order (scan-threshold FFD)
    const int nopen = lds_cnt[k];
    for (int u0 = 0; u0 < nopen && rem > 0; u0 += WAVE) {
      const int u = u0 + lane;
      int64_t m = 0;
      int64_t uf[NRES];
      if (u < nopen) {
        int64_t mm = (int64_t)1 << 31;
        bool neg = false;
#pragma unroll
        for (int r = 0; r < NRES; ++r) {
          uf[r] = U[((int64_t)k * MAXU + u) * NRES + r];
          if (uf[r] < 0) neg = true;
          if (rq[r] > 0) {
            int64_t q = uf[r] / rq[r];
            mm = q < mm ? q : mm;
          }
        }
        m = neg ? 0 : mm;
      }
      int64_t inc = wave_inclusive_scan_i64(m);
      const int64_t ex = inc - m;
      int64_t take = rem - ex;
      take = take < 0 ? 0 : (take > m ? m : take);
      if (u < nopen && take > 0) {
#pragma unroll
        for (int r = 0; r < NRES; ++r) {
          U[((int64_t)k * MAXU + u) * NRES + r] = uf[r] - take * rq[r];
        }
      }
      const int64_t chunk_total =
          __shfl((long long)inc, WAVE - 1, WAVE);
      rem -= chunk_total < rem ? chunk_total : rem;
    }

    // 2) open new units while the pool cap allows
    int64_t per_unit = (int64_t)1 << 31;
    bool fits = true;
#pragma unroll
    for (int r = 0; r < NRES; ++r) {
      const int64_t cap = pool_unit[k * NRES + r];
      if (rq[r] > cap) fits = false;
      if (rq[r] > 0) {
        int64_t q = cap / rq[r];
        per_unit = q < per_unit ? q : per_unit;
      }
    }
    if (rem > 0 && fits && per_unit > 0) {
      const int room = pool_max_new[k] - nopen;
      int64_t want = (rem + per_unit - 1) / per_unit;
      const int n_new = (int)(want < room ? want : (room < 0 ? 0 : room));
      for (int u0 = 0; u0 < n_new; u0 += WAVE) {
        const int ui = u0 + lane;
        if (ui < n_new) {
          const int64_t ex = (int64_t)ui * per_unit;
          int64_t take = rem - ex;
          take = take < 0 ? 0 : (take > per_unit ? per_unit : take);
#pragma unroll
          for (int r = 0; r < NRES; ++r) {
            U[((int64_t)k * MAXU + nopen + ui) * NRES + r] =
                (int64_t)pool_unit[k * NRES + r] - take * rq[r];
          }
        }
      }
      const int64_t absorbed =
          n_new * per_unit < rem ? (int64_t)n_new * per_unit : rem;
      rem -= absorbed;
      if (lane == 0) lds_cnt[k] = nopen + n_new;
      __syncthreads();
    }
    if (lane == 0) {
      placed_new[g] = need0 - rem;
      unful[g] = rem;
    }
  }
  __syncthreads();
  for (int k = lane; k < K; k += WAVE) units_cnt[k] = (int64_t)lds_cnt[k];
  for (int q = lane; q < K * NRES; q += WAVE) pool_residual[q] = lds_resid[q];
}

#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

void binpack_place(torch::Tensor free, torch::Tensor elig, torch::Tensor seg_req,
                   torch::Tensor seg_count, torch::Tensor placed_out) {
  const int N = (int)free.size(0);
  const int G = (int)seg_req.size(0);
  if (G == 0) return;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  hipLaunchKernelGGL(binpack_kernel, dim3(1), dim3(BP_BLOCK), 0, stream.stream(),
                     free.data_ptr<int32_t>(),
                     reinterpret_cast<uint64_t*>(elig.data_ptr<int64_t>()),
                     seg_req.data_ptr<int32_t>(), seg_count.data_ptr<int64_t>(),
                     N, G, placed_out.data_ptr<int64_t>());
}

void binpack_units(torch::Tensor seg_pool, torch::Tensor seg_req,
                   torch::Tensor seg_count, torch::Tensor placed,
                   torch::Tensor pool_unit, torch::Tensor pool_max_new,
                   torch::Tensor units_free, torch::Tensor units_cnt,
                   torch::Tensor placed_new, torch::Tensor unful,
                   torch::Tensor pool_residual) {
  const int G = (int)seg_req.size(0);
  const int K = (int)pool_unit.size(0);
  const int MAXU = (int)units_free.size(1);
  if (G == 0) return;
  TORCH_CHECK(K <= UNITS_MAX_POOLS, "units_kernel supports at most 64 pools");
  const size_t lds_bytes = (size_t)K * MAXU * NRES * sizeof(int64_t);
  const int use_lds = lds_bytes <= 64 * 1024 ? 1 : 0;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  hipLaunchKernelGGL(units_kernel, dim3(1), dim3(WAVE),
                     use_lds ? lds_bytes : 0, stream.stream(),
                     seg_pool.data_ptr<int32_t>(), seg_req.data_ptr<int32_t>(),
                     seg_count.data_ptr<int64_t>(), placed.data_ptr<int64_t>(),
                     pool_unit.data_ptr<int32_t>(),
                     pool_max_new.data_ptr<int32_t>(), G, K, MAXU, use_lds,
                     units_free.data_ptr<int64_t>(), units_cnt.data_ptr<int64_t>(),
                     placed_new.data_ptr<int64_t>(), unful.data_ptr<int64_t>(),
                     pool_residual.data_ptr<int64_t>());
}
