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
#include "common.h"

#define UNITS_MAX_POOLS 64
#define UNITS_BLOCK 256
#define UNITS_NW (UNITS_BLOCK / WAVE)
#define UNITS_PAD 8  // arena slack per wave

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
  const int tid = threadIdx.x;
  const int W = (N + WAVE - 1) / WAVE;
  const int G = *g_ptr;

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
    for (int base = 0; base < N && carry < need; base += BLOCK) {
      const int j = base + tid;
      int64_t m = 0;
      int32_t f[NRES];
      if (j < N && ((erow[j / WAVE] >> (j & (WAVE - 1))) & 1ull)) {
#pragma unroll
        for (int r = 0; r < NRES; ++r) f[r] = free_[(int64_t)j * NRES + r];
        m = per_node_capacity(f, rq);
      }
      int64_t chunk_total;
      const int64_t scan = block_exclusive_scan_i64<BLOCK>(m, lds_scan, &chunk_total);
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
// max — reconstructed], keeping device-resident state so the decision
// vector (units per pool) never round-trips to the host before the
// RCCL all-reduce.
//
// Run-length representation: units opened in one batch share identical
// residual vectors, and one placement step splits at most ONE run
// (the partially-consumed one) into <=3 runs, so exact per-unit FFD
// over an UNBOUNDED unit count is a walk over <=4G+K runs per pool —
// O(G^2) total. (The per-unit dense formulation measured 15.6 ms/step
// at 1M pods / 327k units; this is the algorithmic fix.) Semantics are
// bit-identical to the oracle's unit-at-a-time loop: a unit takes
// min(capacity, remaining), units in creation order.
//
// Pools are mutually independent (a segment routes to exactly one
// pool; residual/unit state is per pool), so pool k is owned by wave
// k % NW and the serial walks run NW-way parallel — the global FFD
// order within each pool is preserved, so results are bit-identical
// to the single-lane form. The run arena (res[cap][NRES] | cnt[cap] |
// next[cap], LDS when it fits else global scratch) is carved per wave
// from a prefix scan of per-wave segment counts.
//
// max-new is computed inline (clamp(pool_max - pool_cur, 0)) — it was
// a separate 4.7 us kernel launch for K<=64 elements. When
// blob_mirror != null the kernel also runs the readback epilogue
// (segment sel-hash column + 3-word stat into the blob tail),
// replacing two device-to-device copy launches.
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
    // fused readback epilogue (all null/0 when standalone)
    const int64_t* __restrict__ seg_sel_hash, const int32_t* __restrict__ stat,
    int64_t* __restrict__ blob_mirror, int32_t* __restrict__ blob_stat,
    int32_t SM)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int G = *g_ptr;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int w = tid / WAVE;

  __shared__ int32_t head[UNITS_MAX_POOLS];
  __shared__ int32_t tailp[UNITS_MAX_POOLS];
  __shared__ int64_t opened[UNITS_MAX_POOLS];
  __shared__ int64_t resid[UNITS_MAX_POOLS * NRES];
  __shared__ int32_t wave_off[UNITS_NW + 1];

  for (int k = tid; k < K; k += UNITS_BLOCK) {
    head[k] = -1;
    tailp[k] = -1;
    opened[k] = 0;
  }
  for (int q = tid; q < K * NRES; q += UNITS_BLOCK) resid[q] = 0;

  // per-wave arena need: 4 runs per owned segment + fixed slack
  int cnt = 0;
  for (int g = lane; g < G; g += WAVE) {
    const int k = seg_pool[g];
    if (k >= 0 && (k % UNITS_NW) == w) ++cnt;
  }
#pragma unroll
  for (int off = WAVE / 2; off; off >>= 1) cnt += __shfl_down(cnt, off, WAVE);
  if (lane == 0) wave_off[w] = 4 * cnt + UNITS_PAD;
  __syncthreads();
  if (tid == 0) {
    int off = 0;
    for (int i = 0; i < UNITS_NW; ++i) {
      const int n = wave_off[i];
      wave_off[i] = off;
      off += n;
    }
    wave_off[UNITS_NW] = off;
  }
  __syncthreads();
  const int total_need = wave_off[UNITS_NW];
  const bool use_lds = total_need <= lds_cap;
  const int cap = use_lds ? lds_cap : RTOT;
  int64_t* arena = use_lds ? reinterpret_cast<int64_t*>(smem) : scratch;
  int64_t* run_res = arena;               // [cap][NRES]
  int64_t* run_cnt = arena + (int64_t)cap * NRES;
  int64_t* run_next = run_cnt + cap;
  const int a_hi = wave_off[w + 1];

  if (lane == 0 && w * WAVE < UNITS_BLOCK) {
    int alloc = wave_off[w];
    for (int g = 0; g < G; ++g) {
      const int k = seg_pool[g];
      const bool mine = (k >= 0) ? ((k % UNITS_NW) == w) : (w == 0);
      if (!mine) continue;
      int64_t rem = seg_count[g] - placed[g];
      placed_new[g] = 0;
      unful[g] = 0;
      if (rem <= 0) continue;
      if (k < 0) {
        unful[g] = rem;
        continue;
      }
      if (alloc + 4 > a_hi) {  // arena bound: a segment allocates <=4 runs
        unful[g] = rem;        // (unreachable when the scratch honors the
        continue;              //  4G+pad sizing; kept so a bound bug can
      }                        //  never corrupt the lists)
      int64_t rq[NRES];
#pragma unroll
      for (int r = 0; r < NRES; ++r) rq[r] = seg_req[g * NRES + r];
#pragma unroll
      for (int r = 0; r < NRES; ++r) resid[k * NRES + r] += rq[r] * rem;
      const int64_t need0 = rem;

      // 1) walk open runs in creation order
      for (int idx = head[k]; idx != -1 && rem > 0; idx = (int)run_next[idx]) {
        int64_t m = (int64_t)1 << 31;
        bool neg = false;
#pragma unroll
        for (int r = 0; r < NRES; ++r) {
          const int64_t fr = run_res[(int64_t)idx * NRES + r];
          if (fr < 0) neg = true;
          if (rq[r] > 0) {
            const int64_t q = fr / rq[r];
            m = q < m ? q : m;
          }
        }
        if (neg || m <= 0) continue;
        const int64_t c = run_cnt[idx];
        if (rem >= m * c) {  // every unit of the run takes m
#pragma unroll
          for (int r = 0; r < NRES; ++r) run_res[(int64_t)idx * NRES + r] -= m * rq[r];
          rem -= m * c;
          continue;
        }
        // split: nf full takers, maybe one partial, rest untouched
        const int64_t nf = rem / m;
        const int64_t part = rem - nf * m;
        int64_t untouched = c - nf - (part > 0 ? 1 : 0);
        // node idx becomes the full-takers run (or partial if nf==0)
        const int64_t nxt = run_next[idx];
        int cur = idx;
        if (nf > 0) {
          run_cnt[cur] = nf;
#pragma unroll
          for (int r = 0; r < NRES; ++r) run_res[(int64_t)cur * NRES + r] -= m * rq[r];
        }
        if (part > 0) {
          if (nf > 0) {
            const int nn = alloc++;
            run_next[cur] = nn;
            run_next[nn] = nxt;
#pragma unroll
            for (int r = 0; r < NRES; ++r)
              run_res[(int64_t)nn * NRES + r] =
                  run_res[(int64_t)idx * NRES + r] + m * rq[r] - part * rq[r];
            run_cnt[nn] = 1;
            if (tailp[k] == cur) tailp[k] = nn;
            cur = nn;
          } else {
            run_cnt[cur] = 1;
#pragma unroll
            for (int r = 0; r < NRES; ++r) run_res[(int64_t)cur * NRES + r] -= part * rq[r];
          }
        }
        if (untouched > 0) {
          const int nn = alloc++;
          run_next[cur] = nn;
          run_next[nn] = nxt;
#pragma unroll
          for (int r = 0; r < NRES; ++r)
            run_res[(int64_t)nn * NRES + r] =
                run_res[(int64_t)idx * NRES + r] +
                ((cur == idx && part > 0) ? part : m) * rq[r];
          run_cnt[nn] = untouched;
          if (tailp[k] == cur) tailp[k] = nn;
        }
        rem = 0;
        break;
      }

      // 2) open new units while the pool cap allows
      int64_t per_unit = (int64_t)1 << 31;
      bool fits = true;
#pragma unroll
      for (int r = 0; r < NRES; ++r) {
        const int64_t ucap = pool_unit[k * NRES + r];  // not the arena cap
        if (rq[r] > ucap) fits = false;
        if (rq[r] > 0) {
          const int64_t q = ucap / rq[r];
          per_unit = q < per_unit ? q : per_unit;
        }
      }
      if (rem > 0 && fits && per_unit > 0) {
        int32_t mn = pool_max[k] - pool_cur[k];  // inline max-new
        if (mn < 0) mn = 0;
        int64_t room = (int64_t)mn - opened[k];
        if (room < 0) room = 0;
        int64_t nf = rem / per_unit;
        nf = nf < room ? nf : room;
        if (nf > 0) {
          const int nn = alloc++;
#pragma unroll
          for (int r = 0; r < NRES; ++r)
            run_res[(int64_t)nn * NRES + r] = (int64_t)pool_unit[k * NRES + r] - per_unit * rq[r];
          run_cnt[nn] = nf;
          run_next[nn] = -1;
          if (tailp[k] == -1) head[k] = nn; else run_next[tailp[k]] = nn;
          tailp[k] = nn;
          opened[k] += nf;
          rem -= nf * per_unit;
          room -= nf;
        }
        if (rem > 0 && room > 0) {  // one partial unit
          const int nn = alloc++;
#pragma unroll
          for (int r = 0; r < NRES; ++r)
            run_res[(int64_t)nn * NRES + r] = (int64_t)pool_unit[k * NRES + r] - rem * rq[r];
          run_cnt[nn] = 1;
          run_next[nn] = -1;
          if (tailp[k] == -1) head[k] = nn; else run_next[tailp[k]] = nn;
          tailp[k] = nn;
          opened[k] += 1;
          rem = 0;
        }
      }
      placed_new[g] = need0 - rem;
      unful[g] = rem;
    }
  }
  __syncthreads();
  for (int k = tid; k < K; k += UNITS_BLOCK) units_cnt[k] = opened[k];
  for (int q = tid; q < K * NRES; q += UNITS_BLOCK) pool_residual[q] = resid[q];
  if (blob_mirror != nullptr) {
    for (int i = tid; i < SM; i += UNITS_BLOCK) blob_mirror[i] = seg_sel_hash[i];
    if (tid < 3) blob_stat[tid] = stat[tid];
  }
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
