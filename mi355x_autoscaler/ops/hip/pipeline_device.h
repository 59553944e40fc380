// Shared device bodies of the scale-up decision pipeline.
//
// Each phase of the decide (hash-aggregate -> compact+sort -> free ->
// eligibility -> route -> FFD place -> units) lives here as a
// __device__ function with the block/grid mapping passed in. Two
// callers instantiate them:
//
//   * the standalone kernels (segment_hash.hip, binpack.hip,
//     fit_predicates.hip, node_free.hip) — one launch per phase,
//     used by the classic orchestrated pipeline, the op-by-op Python
//     reference path, and the large-P regime;
//   * mega_decide.hip — ONE cooperative launch running every phase
//     with grid.sync() barriers, for the small/medium-P regime where
//     the ~13 per-launch fixed costs dominate the decision.
//
// Sharing the bodies keeps the two paths bit-identical by
// construction (same arithmetic, same order); tests assert equality
// against the CPU oracle on hardware.
#pragma once

#include "common.h"

// ---------------------------------------------------------------------------
// hash-aggregate constants (shared by segment_hash.hip and the mega path)
#define SEG_MAX 2048
#define ST_EMPTY 0
#define ST_CLAIMED 1
#define ST_READY 2
#define MAX_SPIN 100000
#define RLX_AGENT __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT

__device__ inline uint32_t mix_hash(int64_t ks, int64_t kr) {
  uint64_t x = (uint64_t)ks * 0x9E3779B97F4A7C15ull ^
               (uint64_t)kr * 0xC2B2AE3D27D4EB4Full;
  x ^= x >> 29;
  x *= 0xBF58476D1CE4E5B9ull;
  x ^= x >> 32;
  return (uint32_t)x;
}

// Probe/claim one key; returns the slot or -1 (overflow). Agent-scope
// release/ready-flag discipline (guide §6 G16); spins bounded.
__device__ inline int probe_claim(int64_t ks, int64_t kr, int32_t H,
                                  int64_t* slot_ks, int64_t* slot_kr,
                                  int32_t* slot_state, int32_t* overflow) {
  const uint32_t mask = (uint32_t)H - 1u;
  uint32_t h = mix_hash(ks, kr) & mask;
  for (int probe = 0; probe < H; ++probe) {
    int st = __hip_atomic_load(&slot_state[h], RLX_AGENT);
    if (st == ST_EMPTY) {
      int expected = ST_EMPTY;
      if (__hip_atomic_compare_exchange_strong(
              &slot_state[h], &expected, ST_CLAIMED, __ATOMIC_RELAXED,
              __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT)) {
        __hip_atomic_store(&slot_ks[h], ks, RLX_AGENT);
        __hip_atomic_store(&slot_kr[h], kr, RLX_AGENT);
        __hip_atomic_store(&slot_state[h], ST_READY,
                           __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_AGENT);
        return (int)h;
      }
      st = expected;  // observed state; handle CLAIMED/READY below
    }
    if (st == ST_CLAIMED) {
      // claimer is between CAS and READY: spin briefly on this slot
      int spins = 0;
      while ((st = __hip_atomic_load(&slot_state[h], RLX_AGENT)) == ST_CLAIMED) {
        __builtin_amdgcn_s_sleep(1);
        if (++spins > MAX_SPIN) {
          atomicOr(overflow, 2);
          st = ST_READY;  // bail; comparison below will mismatch or not
          break;
        }
      }
    }
    if (st == ST_READY) {
      if (__hip_atomic_load(&slot_ks[h], RLX_AGENT) == ks &&
          __hip_atomic_load(&slot_kr[h], RLX_AGENT) == kr) {
        return (int)h;
      }
    }
    h = (h + 1u) & mask;
  }
  return -1;
}

// Wave-aggregated direct-global hash build (the small-P form).
// Round 2: per-lane 1-entry memo with LOCAL run counting — a burst is
// runs of identical (selector, request) keys, so the steady state per
// pod is two register compares + an increment; counts/min-ids flush
// once per key RUN (memo switch or loop end) instead of one global
// atomic pair per distinct key per wave-iteration. Grid-strided over
// pods; safe at any grid shape.
__device__ inline void dev_hash_build_small_ex(
    const int32_t* __restrict__ pod_flags, const int64_t* __restrict__ sel_hash,
    const int32_t* __restrict__ pod_req, int64_t P, int32_t H,
    int64_t* __restrict__ slot_ks, int64_t* __restrict__ slot_kr,
    int32_t* __restrict__ slot_state, int32_t* __restrict__ slot_count,
    int32_t* __restrict__ slot_first, int32_t* __restrict__ slot_of_pod,
    int32_t* __restrict__ overflow, int64_t start, int64_t stride) {
  const int lane = threadIdx.x & (WAVE - 1);
  int64_t memo_ks = 0, memo_kr = 0;
  int memo_slot = INT32_MIN;
  int32_t memo_cnt = 0;
  int32_t memo_min_i0 = 0;
  auto memo_flush = [&]() {
    if (memo_slot >= 0 && memo_cnt > 0) {
      atomicAdd(&slot_count[memo_slot], memo_cnt);
      atomicMax(&slot_first[memo_slot], INT32_MAX - memo_min_i0);
    }
    memo_cnt = 0;
  };
  for (int64_t i0 = start; ; i0 += stride) {
    if (__all(i0 >= P)) break;
    const int pf = i0 < P ? pod_flags[i0] : 0;
    const bool live = i0 < P && (pf & PF_PENDING);
    int64_t ks = 0, kr = 0;
    if (live) {
      ks = sel_hash[i0];
      int32_t rq[NRES];
#pragma unroll
      for (int r = 0; r < NRES; ++r) rq[r] = pod_req[i0 * NRES + r];
      kr = pack_req_key(rq, pf & PF_CRITICAL);
    } else if (i0 < P) {
      slot_of_pod[i0] = -1;
    }

    int my_slot = -1;
    bool hit = false;
    if (live && memo_slot != INT32_MIN && ks == memo_ks && kr == memo_kr) {
      my_slot = memo_slot;
      hit = true;
      if (memo_slot >= 0) {
        if (memo_cnt == 0) memo_min_i0 = (int32_t)i0;
        ++memo_cnt;
      }
    }

    uint64_t pending_lanes = __ballot(live && !hit);
    while (pending_lanes) {
      const int leader = __ffsll((unsigned long long)pending_lanes) - 1;
      const int64_t k0 = __shfl(ks, leader, WAVE);
      const int64_t r0 = __shfl(kr, leader, WAVE);
      const bool match = live && !hit && ks == k0 && kr == r0;
      const uint64_t match_mask = __ballot(match);
      int slot = -1;
      if (lane == leader) {
        slot = probe_claim(k0, r0, H, slot_ks, slot_kr, slot_state, overflow);
        if (slot >= 0) {
          atomicAdd(&slot_count[slot], __popcll((unsigned long long)match_mask));
          atomicMax(&slot_first[slot], INT32_MAX - (int32_t)i0);
        } else {
          atomicOr(overflow, 1);
        }
      }
      slot = __shfl(slot, leader, WAVE);
      if (match) {
        my_slot = slot;
        // counted by the leader already: seed the memo with cnt 0 so
        // the run continues on the register fast path
        memo_flush();
        memo_ks = ks;
        memo_kr = kr;
        memo_slot = slot;
      }
      pending_lanes &= ~match_mask;
    }
    if (i0 < P && live) slot_of_pod[i0] = my_slot;
  }
  memo_flush();
}

__device__ inline void dev_hash_build_small(
    const int32_t* __restrict__ pod_flags, const int64_t* __restrict__ sel_hash,
    const int32_t* __restrict__ pod_req, int64_t P, int32_t H,
    int64_t* __restrict__ slot_ks, int64_t* __restrict__ slot_kr,
    int32_t* __restrict__ slot_state, int32_t* __restrict__ slot_count,
    int32_t* __restrict__ slot_first, int32_t* __restrict__ slot_of_pod,
    int32_t* __restrict__ overflow) {
  dev_hash_build_small_ex(pod_flags, sel_hash, pod_req, P, H, slot_ks, slot_kr,
                          slot_state, slot_count, slot_first, slot_of_pod,
                          overflow,
                          (int64_t)blockIdx.x * blockDim.x + threadIdx.x,
                          (int64_t)gridDim.x * blockDim.x);
}

// Single-block compact + bitonic sort into the oracle's global segment
// order (sel-hash as uint64 asc, request key desc). LDS arrays are
// caller-provided: lds_ks/lds_kr [SEG_MAX] u64, lds_slot [SEG_MAX]
// i32, lds_n one i32. Runs at any block size.
__device__ inline void dev_compact_sort(
    const int64_t* __restrict__ slot_ks, const int64_t* __restrict__ slot_kr,
    const int32_t* __restrict__ slot_state, const int32_t* __restrict__ slot_count,
    const int32_t* __restrict__ slot_first, int32_t H,
    const int64_t* __restrict__ pod_sel_pairs, const int32_t* __restrict__ pod_sel_count,
    const int32_t* __restrict__ pod_req,
    int64_t* __restrict__ seg_sel_pairs, int32_t* __restrict__ seg_sel_count,
    int64_t* __restrict__ seg_sel_hash, int32_t* __restrict__ seg_req,
    int64_t* __restrict__ seg_count, int64_t* __restrict__ seg_key,
    int32_t* __restrict__ seg_first_pod, int32_t* __restrict__ slot2seg,
    int32_t* __restrict__ g_out,
    uint64_t* lds_ks, uint64_t* lds_kr, int32_t* lds_slot, int32_t* lds_n,
    int64_t* __restrict__ rows_out = nullptr) {  // [SEG_MAX, MAXSEL+8] wire rows
  const int tid = threadIdx.x;
  const int B = blockDim.x;
  if (tid == 0) *lds_n = 0;
  __syncthreads();

  // compact occupied slots into LDS (order arbitrary; sort fixes it)
  for (int h = tid; h < H; h += B) {
    if (slot_state[h] == ST_READY && slot_count[h] > 0) {
      const int g = atomicAdd(lds_n, 1);
      if (g < SEG_MAX) {
        // sel asc as uint64: flip the sign bit so int compare == uint
        lds_ks[g] = (uint64_t)slot_ks[h] ^ 0x8000000000000000ull;
        // req desc: invert the pack key
        lds_kr[g] = ~(uint64_t)slot_kr[h];
        lds_slot[g] = h;
      }
    }
  }
  __syncthreads();
  int G = *lds_n;
  if (G > SEG_MAX) {
    if (tid == 0) {
      g_out[0] = 0;
      g_out[1] = 1;
    }
    return;
  }
  // pad to pow2 with +inf keys
  int NP = 1;
  while (NP < G) NP <<= 1;
  if (NP < 2) NP = 2;
  for (int g = G + tid; g < NP; g += B) {
    lds_ks[g] = ~0ull;
    lds_kr[g] = ~0ull;
    lds_slot[g] = -1;
  }
  __syncthreads();

  // bitonic sort ascending by (ks, kr)
  for (int k = 2; k <= NP; k <<= 1) {
    for (int j = k >> 1; j > 0; j >>= 1) {
      for (int t = tid; t < NP; t += B) {
        const int ixj = t ^ j;
        if (ixj > t) {
          const bool up = ((t & k) == 0);
          const bool gt = (lds_ks[t] > lds_ks[ixj]) ||
                          (lds_ks[t] == lds_ks[ixj] && lds_kr[t] > lds_kr[ixj]);
          if (gt == up) {
            uint64_t a = lds_ks[t]; lds_ks[t] = lds_ks[ixj]; lds_ks[ixj] = a;
            uint64_t b = lds_kr[t]; lds_kr[t] = lds_kr[ixj]; lds_kr[ixj] = b;
            int32_t c = lds_slot[t]; lds_slot[t] = lds_slot[ixj]; lds_slot[ixj] = c;
          }
        }
      }
      __syncthreads();
    }
  }

  // emit segment SoA in sorted order + slot->segment LUT
  for (int g = tid; g < G; g += B) {
    const int h = lds_slot[g];
    slot2seg[h] = g;
    // slot_first stores INT32_MAX - min(pod id) under atomicMax so the
    // workspace zeroes with one memset; decode here.
    const int fp = INT32_MAX - slot_first[h];
    seg_sel_hash[g] = (int64_t)(lds_ks[g] ^ 0x8000000000000000ull);
    seg_count[g] = (int64_t)slot_count[h];
    seg_key[g] = (int64_t)~lds_kr[g];  // un-invert: the raw pack key
    seg_first_pod[g] = fp;
    seg_sel_count[g] = pod_sel_count[fp];
#pragma unroll
    for (int q = 0; q < MAXSEL; ++q)
      seg_sel_pairs[(int64_t)g * MAXSEL + q] = pod_sel_pairs[(int64_t)fp * MAXSEL + q];
#pragma unroll
    for (int r = 0; r < NRES; ++r)
      seg_req[g * NRES + r] = pod_req[(int64_t)fp * NRES + r];
    if (rows_out != nullptr) {
      // sharded wire format (engine_sharded._ROW_COLS): emitted here so
      // the all-gather source needs no torch-op assembly
      int64_t* row = rows_out + (int64_t)g * (MAXSEL + 8);
#pragma unroll
      for (int q = 0; q < MAXSEL; ++q) row[q] = pod_sel_pairs[(int64_t)fp * MAXSEL + q];
      row[MAXSEL] = pod_sel_count[fp];
#pragma unroll
      for (int r = 0; r < NRES; ++r) row[MAXSEL + 1 + r] = pod_req[(int64_t)fp * NRES + r];
      row[MAXSEL + 5] = (int64_t)slot_count[h];
      row[MAXSEL + 6] = seg_sel_hash[g];
      row[MAXSEL + 7] = seg_key[g];
    }
  }
  if (rows_out != nullptr) {
    // stale rows from earlier cycles must read as padding (count 0) so
    // a FIXED-size gather needs no host-visible G
    for (int g = G + tid; g < SEG_MAX; g += B)
      rows_out[(int64_t)g * (MAXSEL + 8) + MAXSEL + 5] = 0;
  }
  if (tid == 0) {
    g_out[0] = G;
    g_out[1] = 0;
  }
}

// free-capacity usage subtraction, grid-strided over pods.
__device__ inline void dev_node_usage_subtract(
    const int32_t* __restrict__ pod_node, const int32_t* __restrict__ pod_req,
    int64_t P, int32_t* __restrict__ node_free) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < P;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int j = pod_node[i];
    if (j < 0) continue;
#pragma unroll
    for (int r = 0; r < NRES; ++r) {
      const int32_t v = pod_req[i * NRES + r];
      // zero-valued atomics still serialize on the line; skip them
      // (tombstone/padding rows and sparse request vectors)
      if (v != 0) atomicSub(&node_free[(int64_t)j * NRES + r], v);
    }
  }
}

// One 256-node eligibility tile: stage the tile's label sets into LDS,
// then emit 64-bit ballots for every segment. lds_labels holds
// FIT_TILE*MAXLAB i64, lds_nlab FIT_TILE i32. Block size must be
// FIT_TILE (== 256).
#define FIT_TILE 256

__device__ inline void dev_fit_tile(
    int node0,
    const int64_t* __restrict__ node_labels, const int32_t* __restrict__ node_label_count,
    const int32_t* __restrict__ node_unsched,
    const int64_t* __restrict__ seg_sel, const int32_t* __restrict__ seg_sel_count,
    int32_t N, int G, uint64_t* __restrict__ elig,
    int64_t* lds_labels, int32_t* lds_nlab) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;
  const int my_node = node0 + tid;
  const int W = (N + WAVE - 1) / WAVE;

  // stage this tile's node labels into LDS: thread t copies node t's row
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
  __syncthreads();  // tile reuse: LDS free for the next tile
}

// Segment -> pool routing for ONE segment; a full wave cooperates
// (lanes fan out over pools). Matches the oracle's first-fit pool
// choice (ascending pool index == sorted pool name).
__device__ inline void dev_route_one(
    int g, int32_t K,
    const int64_t* __restrict__ seg_sel, const int32_t* __restrict__ seg_sel_count,
    const int32_t* __restrict__ seg_req,
    const int64_t* __restrict__ pool_labels, const int32_t* __restrict__ pool_label_count,
    const int32_t* __restrict__ pool_unit, int32_t* __restrict__ seg_pool) {
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

// pod -> routed-pool map (for the MFMA demand segsum), grid-strided.
__device__ inline void dev_pod_pool_map(
    const int32_t* __restrict__ slot_of_pod, const int32_t* __restrict__ slot2seg,
    const int32_t* __restrict__ seg_pool, int64_t P,
    int32_t* __restrict__ pod_pool) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < P;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int s = slot_of_pod[i];
    pod_pool[i] = (s >= 0) ? seg_pool[slot2seg[s]] : -1;
  }
}

// Single-block FFD placement over existing nodes (kernel B core).
// lds_scan must hold BLOCK/WAVE + 1 int64 slots.
//
// The serial segment loop is LATENCY-bound, not bandwidth-bound
// (rocprof: 17.8 us for 32 segments x 100 nodes): every iteration
// chained global loads of the segment metadata and the free table.
// Two fixes: (a) segment count/req preload into LDS once, (b) when
// one thread per node covers the cluster (N <= BLOCK — the common
// case), the free table lives in REGISTERS across the whole segment
// walk and is written back once at the end.
#define BP_PRELOAD 512

template <int BLOCK>
__device__ inline void dev_binpack(
    int32_t* __restrict__ free_, const uint64_t* __restrict__ elig,
    const int32_t* __restrict__ seg_req, const int64_t* __restrict__ seg_count,
    int32_t N, int G, int64_t* __restrict__ placed_out, int64_t* lds_scan) {
  const int tid = threadIdx.x;
  const int W = (N + WAVE - 1) / WAVE;

  __shared__ int64_t s_cnt[BP_PRELOAD];
  __shared__ int32_t s_req[BP_PRELOAD * NRES];
  for (int g = tid; g < G && g < BP_PRELOAD; g += BLOCK) {
    s_cnt[g] = seg_count[g];
#pragma unroll
    for (int r = 0; r < NRES; ++r) s_req[g * NRES + r] = seg_req[g * NRES + r];
  }

  const bool reg_free = N <= BLOCK;
  int32_t fr[NRES];
  if (reg_free && tid < N) {
#pragma unroll
    for (int r = 0; r < NRES; ++r) fr[r] = free_[(int64_t)tid * NRES + r];
  }
  __syncthreads();

  if (N <= WAVE) {
    // single-wave fast path: one wave covers every node, so the
    // G-serial walk needs NO block barriers — each iteration is a
    // wave shuffle-scan instead of a 2-barrier LDS scan. At tiny N
    // (the 1k-pod / 8-node BASELINE config #2 shape) the segment loop
    // is latency-bound on exactly those barriers.
    if (tid < WAVE) {
      const int lane = tid;
      for (int g = 0; g < G; ++g) {
        const int64_t need = g < BP_PRELOAD ? s_cnt[g] : seg_count[g];
        if (need <= 0) {
          if (lane == 0) placed_out[g] = 0;
          continue;
        }
        int32_t rq[NRES];
#pragma unroll
        for (int r = 0; r < NRES; ++r)
          rq[r] = g < BP_PRELOAD ? s_req[g * NRES + r] : seg_req[g * NRES + r];
        int64_t m = 0;
        if (lane < N && ((elig[(int64_t)g * W] >> lane) & 1ull)) {
          m = per_node_capacity(fr, rq);
        }
        const int64_t inc = wave_inclusive_scan_i64(m);
        const int64_t chunk_total = __shfl(inc, WAVE - 1);
        int64_t take = need - (inc - m);
        take = take < 0 ? 0 : (take > m ? m : take);
        if (take > 0) {
#pragma unroll
          for (int r = 0; r < NRES; ++r) fr[r] -= (int32_t)(take * rq[r]);
        }
        if (lane == 0) placed_out[g] = chunk_total < need ? chunk_total : need;
      }
      if (lane < N) {  // keep the mutated-free contract
#pragma unroll
        for (int r = 0; r < NRES; ++r) free_[(int64_t)lane * NRES + r] = fr[r];
      }
    }
    __syncthreads();  // other waves rejoin; wave 0's writes visible
    return;
  }

  // Large-N fast path (round 2, the 1M-config N=4000 regime): wave-
  // owned node chunks with the free table REGISTER-resident across
  // the whole segment walk and ONE block barrier per segment (the
  // round-1 chunked loop paid 2 barriers per chunk per segment —
  // ~208 barriers at N=4000/G=26 — plus a free reload per chunk).
  // Wave w owns chunks {w, w+NW, ...}; only the per-chunk capacity
  // totals cross waves (parity-double-buffered LDS row), so the FFD
  // scan-threshold math is bit-identical to the serial node order.
  const int nchunks = (N + WAVE - 1) / WAVE;
  if constexpr (BLOCK >= 1024) {
   if (N > BLOCK && nchunks <= WAVE) {
    constexpr int NW_ = BLOCK / WAVE;
    constexpr int CPW = (WAVE + NW_ - 1) / NW_;  // chunks per wave
    const int lane = tid & (WAVE - 1);
    const int wv = tid / WAVE;
    __shared__ int64_t s_ctot[2][WAVE];
    // stage the whole eligibility bitmask in LDS when it fits: the
    // per-chunk erow loads were dependent L2/HBM round trips on the
    // single resident workgroup (PMC: 692k VALU yet ~70 us — pure
    // latency), and every segment touches every chunk word
#define BP_ELIG_LDS 2048
    __shared__ uint64_t s_elig[BP_ELIG_LDS];
    const bool stage_elig = (int64_t)G * W <= BP_ELIG_LDS;
    if (stage_elig) {
      for (int idx = tid; idx < G * W; idx += BLOCK) s_elig[idx] = elig[idx];
    }
    __syncthreads();
    // my chunk ids: wv, wv+NW_, ... ; node j = chunk*WAVE + lane
    int32_t fr2[CPW][NRES];
    int cid[CPW];
#pragma unroll
    for (int q = 0; q < CPW; ++q) {
      cid[q] = wv + q * NW_;
      const int j = cid[q] * WAVE + lane;
      if (cid[q] < nchunks && j < N) {
#pragma unroll
        for (int r = 0; r < NRES; ++r)
          fr2[q][r] = free_[(int64_t)j * NRES + r];
      }
    }
    for (int g = 0; g < G; ++g) {
      const int64_t need = g < BP_PRELOAD ? s_cnt[g] : seg_count[g];
      const int par = g & 1;
      if (need <= 0) {
        if (tid == 0) placed_out[g] = 0;
        // keep the barrier structure uniform across the block
        __syncthreads();
        continue;
      }
      int32_t rq[NRES];
#pragma unroll
      for (int r = 0; r < NRES; ++r)
        rq[r] = g < BP_PRELOAD ? s_req[g * NRES + r] : seg_req[g * NRES + r];
      const uint64_t* erow =
          stage_elig ? s_elig + (int64_t)g * W : elig + (int64_t)g * W;
      int64_t m_q[CPW], inc_q[CPW];
#pragma unroll
      for (int q = 0; q < CPW; ++q) {
        const int j = cid[q] * WAVE + lane;
        int64_t m = 0;
        if (cid[q] < nchunks && j < N &&
            ((erow[cid[q]] >> lane) & 1ull)) {
          m = per_node_capacity(fr2[q], rq);
        }
        m_q[q] = m;
        inc_q[q] = wave_inclusive_scan_i64(m);
        if (cid[q] < nchunks && lane == WAVE - 1)
          s_ctot[par][cid[q]] = inc_q[q];
      }
      __syncthreads();  // publish chunk totals
      // every wave redundantly scans the totals row (<= 64 entries)
      const int64_t v = lane < nchunks ? s_ctot[par][lane] : 0;
      const int64_t cinc = wave_inclusive_scan_i64(v);
      const int64_t total_all = __shfl(cinc, nchunks - 1, WAVE);
#pragma unroll
      for (int q = 0; q < CPW; ++q) {
        if (cid[q] >= nchunks) continue;
        const int64_t cpref = __shfl(cinc - v, cid[q], WAVE);
        const int64_t S = cpref + inc_q[q] - m_q[q];
        int64_t take = need - S;
        take = take < 0 ? 0 : (take > m_q[q] ? m_q[q] : take);
        if (take > 0) {
#pragma unroll
          for (int r = 0; r < NRES; ++r)
            fr2[q][r] -= (int32_t)(take * rq[r]);
        }
      }
      if (tid == 0) placed_out[g] = total_all < need ? total_all : need;
      // no second barrier: the next segment's totals go to the other
      // parity row, and free is only touched by its owner lane
    }
#pragma unroll
    for (int q = 0; q < CPW; ++q) {  // keep the mutated-free contract
      const int j = cid[q] * WAVE + lane;
      if (cid[q] < nchunks && j < N) {
#pragma unroll
        for (int r = 0; r < NRES; ++r)
          free_[(int64_t)j * NRES + r] = fr2[q][r];
      }
    }
    __syncthreads();
    return;
   }
  }

  for (int g = 0; g < G; ++g) {
    const int64_t need = g < BP_PRELOAD ? s_cnt[g] : seg_count[g];
    if (need <= 0) {
      if (tid == 0) placed_out[g] = 0;
      continue;
    }
    int32_t rq[NRES];
#pragma unroll
    for (int r = 0; r < NRES; ++r)
      rq[r] = g < BP_PRELOAD ? s_req[g * NRES + r] : seg_req[g * NRES + r];
    const uint64_t* erow = elig + (int64_t)g * W;

    if (reg_free) {
      // one chunk covers every node; free stays in registers
      const int j = tid;
      int64_t m = 0;
      if (j < N && ((erow[j / WAVE] >> (j & (WAVE - 1))) & 1ull)) {
        m = per_node_capacity(fr, rq);
      }
      int64_t chunk_total;
      const int64_t scan = block_exclusive_scan_i64<BLOCK>(m, lds_scan, &chunk_total);
      int64_t take = need - scan;
      take = take < 0 ? 0 : (take > m ? m : take);
      if (take > 0) {
#pragma unroll
        for (int r = 0; r < NRES; ++r) fr[r] -= (int32_t)(take * rq[r]);
      }
      if (tid == 0) placed_out[g] = chunk_total < need ? chunk_total : need;
      __syncthreads();
      continue;
    }

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

  if (reg_free && tid < N) {  // keep the mutated-free contract
#pragma unroll
    for (int r = 0; r < NRES; ++r) free_[(int64_t)tid * NRES + r] = fr[r];
  }
}

// Single-window MFMA demand segsum as a DEVICE function (round 2):
// the mega kernel's non-zero blocks run it over pod_pool during the
// single-workgroup FFD tail (which otherwise leaves them idle), so
// the per-pool demand needs no separate launch or side stream on the
// mega path. Same exact-f32 limb math as segsum_mfma.hip (chunks of
// 256, partials < 2^24); S <= 16 (one window); out accumulated with
// device atomics (commutative integer adds — deterministic).
typedef float bp_f32x4 __attribute__((ext_vector_type(4)));

__device__ inline void dev_demand_segsum_fastpath(
    const int32_t* __restrict__ values,  // [M, NRES]
    const int32_t* __restrict__ seg_of,  // [M] in [-1, S)
    int64_t M, int32_t S,
    uint64_t* __restrict__ out,          // [S, NRES] zeroed by caller
    char* smem_base,                     // >= 4*(256*NRES+256)*4 + 16*NRES*8
    int64_t chunk_start, int64_t chunk_stride) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wv = tid / WAVE;  // 4 waves per 256-thread block
  int32_t* l_vals = reinterpret_cast<int32_t*>(smem_base) + wv * 256 * NRES;
  int32_t* l_segs = reinterpret_cast<int32_t*>(smem_base) + 4 * 256 * NRES +
                    wv * 256;
  unsigned long long* s_acc = reinterpret_cast<unsigned long long*>(
      smem_base + 4 * (256 * NRES + 256) * sizeof(int32_t));

  for (int idx = tid; idx < 16 * NRES; idx += blockDim.x) s_acc[idx] = 0ull;
  __syncthreads();

  const int col = lane & 15;
  const int kidx = lane >> 4;
  const int row = col;
  const int res = row & 3;
  const int limb = (row >> 2) & 1;
  const bool live_row = row < 8;
  const int64_t T = (M + 255) >> 8;

  long long p64[4] = {0, 0, 0, 0};
  for (int64_t t = chunk_start + wv; t < T; t += chunk_stride * 4) {
    const int64_t p_begin = t << 8;
    const int64_t p_end = p_begin + 256 < M ? p_begin + 256 : M;
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      const int pl = lane + 64 * q;
      const int64_t pp = p_begin + pl;
      if (pp < p_end) {
        const int4 v = reinterpret_cast<const int4*>(values)[pp];
        reinterpret_cast<int4*>(l_vals)[pl] = v;
        l_segs[pl] = seg_of[pp];
      } else {
        reinterpret_cast<int4*>(l_vals)[pl] = int4{0, 0, 0, 0};
        l_segs[pl] = -1;
      }
    }
    __builtin_amdgcn_wave_barrier();
    const int steps = (int)((p_end - p_begin + 3) >> 2);
    bp_f32x4 acc[4] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f},
                       {0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
    int sstep = 0;
    for (; sstep + 4 <= steps; sstep += 4) {
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const int pl = (sstep + u) * 4 + kidx;
        const float b = (l_segs[pl] == col) ? 1.f : 0.f;
        float a = 0.f;
        if (live_row) {
          const uint32_t v = (uint32_t)l_vals[pl * NRES + res];
          a = (float)(limb ? (v >> 16) : (v & 0xffffu));
        }
        acc[u] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc[u], 0, 0, 0);
      }
    }
    for (; sstep < steps; ++sstep) {
      const int pl = sstep * 4 + kidx;
      const float b = (l_segs[pl] == col) ? 1.f : 0.f;
      float a = 0.f;
      if (live_row) {
        const uint32_t v = (uint32_t)l_vals[pl * NRES + res];
        a = (float)(limb ? (v >> 16) : (v & 0xffffu));
      }
      acc[0] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc[0], 0, 0, 0);
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) p64[reg] += (long long)acc[u][reg];
    }
    __builtin_amdgcn_wave_barrier();
  }

  // flush window partials into block LDS, then one global add set
  const int s = col;
  if (s < S) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int orow = (lane >> 4) * 4 + reg;
      if (orow < 8) {
        const int ores = orow & 3;
        const int olimb = (orow >> 2) & 1;
        const unsigned long long partial = (unsigned long long)p64[reg];
        if (partial) {
          atomicAdd(&s_acc[s * NRES + ores],
                    olimb ? (partial << 16) : partial);
        }
      }
    }
  }
  __syncthreads();
  for (int idx = tid; idx < S * NRES; idx += blockDim.x) {
    const unsigned long long v = s_acc[idx];
    if (v) atomicAdd(reinterpret_cast<unsigned long long*>(out) + idx, v);
  }
}

// Units phase (kernel B'): run-length FFD into virtual new nodes,
// pools wave-parallel, inline max-new, optional fused readback
// epilogue. Block size must be UNITS_BLOCK (== 256). `lds_arena`
// (lds_cap runs * 6 i64) may be null when lds_cap == 0 — then the
// global scratch is used unconditionally.
//
// Run store (round 2): runs live in an ARRAY indexed
// by an explicit per-pool visit-order list `ord` (replacing the
// round-1 linked list whose single-lane walk dominated config #2).
// Per segment the whole wave evaluates run capacities in visit order,
// a wave scan turns FFD into the same scan-threshold uniform step the
// node FFD uses, full runs update in parallel, and the one boundary
// run splits with a wave-parallel ord shift. Semantics are the
// oracle's exactly: units visited in creation order, splits keep the
// [full-takers, partial, untouched] adjacency (sim/oracle.py
// fulfill_pending phase 2).
#define UNITS_MAX_POOLS 64
#define UNITS_BLOCK 256
#define UNITS_NW (UNITS_BLOCK / WAVE)
#define UNITS_PAD 8  // arena slack per wave

__device__ inline void dev_units(
    const int32_t* __restrict__ seg_pool, const int32_t* __restrict__ seg_req,
    const int64_t* __restrict__ seg_count, const int64_t* __restrict__ placed,
    const int32_t* __restrict__ pool_unit,
    const int32_t* __restrict__ pool_max, const int32_t* __restrict__ pool_cur,
    int G, int32_t K, int32_t RTOT, int32_t lds_cap,
    int64_t* __restrict__ scratch, int64_t* lds_arena,
    int64_t* __restrict__ units_cnt,
    int64_t* __restrict__ placed_new, int64_t* __restrict__ unful,
    int64_t* __restrict__ pool_residual,
    const int64_t* __restrict__ seg_sel_hash, const int32_t* __restrict__ stat,
    int64_t* __restrict__ blob_mirror, int32_t* __restrict__ blob_stat,
    int32_t SM) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int w = tid / WAVE;

  __shared__ int32_t pool_nseg[UNITS_MAX_POOLS];   // segments per pool
  __shared__ int32_t ord_base_s[UNITS_MAX_POOLS];  // per-pool ord region
  __shared__ int32_t pool_R[UNITS_MAX_POOLS];      // live runs per pool
  __shared__ int64_t opened[UNITS_MAX_POOLS];
  __shared__ int64_t resid[UNITS_MAX_POOLS * NRES];
  __shared__ int32_t wave_off[UNITS_NW + 1];
  __shared__ int32_t wave_alloc[UNITS_NW];         // run-slot allocator
  // the metadata loads are latency-bound; stage them in LDS first
#define U_PRELOAD 512
  __shared__ int32_t u_pool[U_PRELOAD];
  __shared__ int64_t u_cnt[U_PRELOAD];
  __shared__ int64_t u_placed[U_PRELOAD];
  __shared__ int32_t u_req[U_PRELOAD * NRES];

  for (int k = tid; k < K; k += UNITS_BLOCK) {
    pool_nseg[k] = 0;
    pool_R[k] = 0;
    opened[k] = 0;
  }
  for (int q = tid; q < K * NRES; q += UNITS_BLOCK) resid[q] = 0;
  for (int g = tid; g < G && g < U_PRELOAD; g += UNITS_BLOCK) {
    u_pool[g] = seg_pool[g];
    u_cnt[g] = seg_count[g];
    u_placed[g] = placed[g];
#pragma unroll
    for (int r = 0; r < NRES; ++r) u_req[g * NRES + r] = seg_req[g * NRES + r];
  }
  __syncthreads();
  for (int g = tid; g < G; g += UNITS_BLOCK) {
    const int k = g < U_PRELOAD ? u_pool[g] : seg_pool[g];
    if (k >= 0) atomicAdd(&pool_nseg[k], 1);
  }
  __syncthreads();
  if (tid == 0) {
    // wave regions sized 4 runs per owned segment + fixed slack; the
    // per-pool ord (visit-order) regions are carved inside them —
    // runs per pool <= 4 * its segments (<=2 split + <=2 new-unit
    // runs each), so ord and run-slot usage both fit the region.
    int woff[UNITS_NW];
    for (int i = 0; i < UNITS_NW; ++i) woff[i] = 0;
    for (int k = 0; k < K; ++k) woff[k % UNITS_NW] += 4 * pool_nseg[k];
    int off = 0;
    for (int i = 0; i < UNITS_NW; ++i) {
      wave_off[i] = off;
      wave_alloc[i] = off;
      off += woff[i] + UNITS_PAD;
    }
    wave_off[UNITS_NW] = off;
    int cur[UNITS_NW];
    for (int i = 0; i < UNITS_NW; ++i) cur[i] = wave_off[i];
    for (int k = 0; k < K; ++k) {
      const int i = k % UNITS_NW;
      ord_base_s[k] = cur[i];
      cur[i] += 4 * pool_nseg[k];
    }
  }
  __syncthreads();
  const int total_need = wave_off[UNITS_NW];
  const bool use_lds = lds_arena != nullptr && total_need <= lds_cap;
  const int cap = use_lds ? lds_cap : RTOT;
  int64_t* arena = use_lds ? lds_arena : scratch;
  int64_t* run_res = arena;                        // [cap][NRES]
  int64_t* run_cnt = arena + (int64_t)cap * NRES;  // [cap]
  int64_t* ord = run_cnt + cap;                    // [cap] visit order
  const int a_hi = wave_off[w + 1];

  for (int g = 0; g < G; ++g) {
    const bool pre = g < U_PRELOAD;
    const int k = pre ? u_pool[g] : seg_pool[g];
    if (k < 0) {
      if (w == 0 && lane == 0) {
        const int64_t rem0 = (pre ? u_cnt[g] : seg_count[g]) -
                             (pre ? u_placed[g] : placed[g]);
        placed_new[g] = 0;
        unful[g] = rem0 > 0 ? rem0 : 0;
      }
      continue;
    }
    if ((k % UNITS_NW) != w) continue;
    int64_t rem = (pre ? u_cnt[g] : seg_count[g]) -
                  (pre ? u_placed[g] : placed[g]);
    if (lane == 0) {
      placed_new[g] = 0;
      unful[g] = 0;
    }
    if (rem <= 0) continue;
    if (wave_alloc[w] + 4 > a_hi) {  // arena bound: <=4 new runs/segment
      if (lane == 0) unful[g] = rem; // (unreachable when the scratch
      continue;                      //  honors 4G+pad; kept as a fence)
    }
    int64_t rq[NRES];
#pragma unroll
    for (int r = 0; r < NRES; ++r)
      rq[r] = pre ? u_req[g * NRES + r] : seg_req[g * NRES + r];
    if (lane == 0) {
#pragma unroll
      for (int r = 0; r < NRES; ++r) resid[k * NRES + r] += rq[r] * rem;
    }
    const int64_t need0 = rem;
    const int ob = ord_base_s[k];
    int R = pool_R[k];

    // 1) wave-parallel FFD over open runs in visit order: capacity
    //    scan -> threshold take (the same uniform step as the node
    //    FFD); full runs update in parallel, the single boundary run
    //    records itself for the split below.
    int64_t carry = 0;
    int b_have = 0, b_slot = -1, b_pos = -1;
    int64_t b_take = 0, b_m = 0, b_cnt = 0, b_res[NRES] = {0};
    for (int base = 0; base < R && carry < rem; base += WAVE) {
      const int p = base + lane;
      int slot = -1;
      int64_t m = 0, c = 0, fr[NRES] = {0};
      if (p < R) {
        slot = (int)ord[ob + p];
        c = run_cnt[slot];
        bool neg = false;
        m = (int64_t)1 << 31;
#pragma unroll
        for (int r = 0; r < NRES; ++r) {
          fr[r] = run_res[(int64_t)slot * NRES + r];
          if (fr[r] < 0) neg = true;
          if (rq[r] > 0) {
            const int64_t q = fr[r] / rq[r];
            m = q < m ? q : m;
          }
        }
        if (neg || m < 0) m = 0;
      }
      // run capacity, clamped to rem+1 so the 64-lane scan cannot
      // overflow i64; take/boundary math is unchanged (a clamped run
      // can never appear fully consumed: take <= rem < rem+1)
      const int64_t t_true = m * c;
      const int64_t t = t_true > rem + 1 ? rem + 1 : t_true;
      const int64_t inc = wave_inclusive_scan_i64(t);
      const int64_t S = carry + inc - t;
      int64_t take = rem - S;
      take = take < 0 ? 0 : (take > t ? t : take);
      if (t > 0 && take == t && t == t_true) {
        // fully consumed: every node of the run takes m
#pragma unroll
        for (int r = 0; r < NRES; ++r)
          run_res[(int64_t)slot * NRES + r] = fr[r] - m * rq[r];
      } else if (take > 0) {
        b_have = 1;
        b_slot = slot;
        b_pos = p;
        b_take = take;
        b_m = m;
        b_cnt = c;
#pragma unroll
        for (int r = 0; r < NRES; ++r) b_res[r] = fr[r];
      }
      carry += __shfl(inc, WAVE - 1);
    }
    const int64_t placed_runs = carry < rem ? carry : rem;

    // 2) boundary split (at most one run): nf full takers keep the
    //    slot; partial/untouched allocate new runs inserted right
    //    after it in the visit order (wave-parallel ord shift) —
    //    exactly the oracle's unit adjacency.
    const uint64_t bmask = __ballot(b_have);
    if (bmask) {
      const int bl = __ffsll((unsigned long long)bmask) - 1;
      const int slot = __shfl(b_slot, bl);
      const int pos = __shfl(b_pos, bl);
      const int64_t take = __shfl(b_take, bl);
      const int64_t m = __shfl(b_m, bl);
      const int64_t c = __shfl(b_cnt, bl);
      int64_t res0[NRES];
#pragma unroll
      for (int r = 0; r < NRES; ++r) res0[r] = __shfl(b_res[r], bl);
      const int64_t nf = m > 0 ? take / m : 0;
      const int64_t part = take - nf * m;
      const int64_t untouched = c - nf - (part > 0 ? 1 : 0);
      const int ins =
          ((nf > 0 && part > 0) ? 1 : 0) + (untouched > 0 ? 1 : 0);
      int s1 = -1, s2 = -1;
      if (lane == 0) {
        int a = wave_alloc[w];
        if (nf > 0) {
          run_cnt[slot] = nf;
#pragma unroll
          for (int r = 0; r < NRES; ++r)
            run_res[(int64_t)slot * NRES + r] = res0[r] - m * rq[r];
          if (part > 0) {
            s1 = a++;
            run_cnt[s1] = 1;
#pragma unroll
            for (int r = 0; r < NRES; ++r)
              run_res[(int64_t)s1 * NRES + r] = res0[r] - part * rq[r];
          }
        } else if (part > 0) {
          run_cnt[slot] = 1;
#pragma unroll
          for (int r = 0; r < NRES; ++r)
            run_res[(int64_t)slot * NRES + r] = res0[r] - part * rq[r];
        }
        if (untouched > 0) {
          s2 = a++;
          run_cnt[s2] = untouched;
#pragma unroll
          for (int r = 0; r < NRES; ++r)
            run_res[(int64_t)s2 * NRES + r] = res0[r];
        }
        wave_alloc[w] = a;
      }
      s1 = __shfl(s1, 0);
      s2 = __shfl(s2, 0);
      if (ins > 0) {
        // shift ord[pos+1 .. R-1] right by ins. High-to-low chunks:
        // within a chunk the wave's lockstep load completes for every
        // lane before the store issues; across chunks the higher one
        // finished first and the ranges don't collide.
        const int tail = R - 1 - (pos + 1);
        for (int base = (tail / WAVE) * WAVE; base >= 0; base -= WAVE) {
          const int i = pos + 1 + base + lane;
          int64_t v = 0;
          const bool act = i >= pos + 1 && i < R;
          if (act) v = ord[ob + i];
          __builtin_amdgcn_wave_barrier();
          if (act) ord[ob + i + ins] = v;
          __builtin_amdgcn_wave_barrier();
        }
        if (lane == 0) {
          int qp = pos + 1;
          if (s1 >= 0) ord[ob + qp++] = s1;
          if (s2 >= 0) ord[ob + qp++] = s2;
        }
        R += ins;
      }
    }
    rem -= placed_runs;

    // 3) open new virtual units while the pool cap allows (O(1):
    //    one full-takers run + at most one partial run)
    if (lane == 0) {
      if (rem > 0) {
        int64_t per_unit = (int64_t)1 << 31;
        bool fits = true;
#pragma unroll
        for (int r = 0; r < NRES; ++r) {
          const int64_t ucap = pool_unit[k * NRES + r];
          if (rq[r] > ucap) fits = false;
          if (rq[r] > 0) {
            const int64_t q = ucap / rq[r];
            per_unit = q < per_unit ? q : per_unit;
          }
        }
        if (fits && per_unit > 0) {
          int32_t mn = pool_max[k] - pool_cur[k];  // inline max-new
          if (mn < 0) mn = 0;
          int64_t room = (int64_t)mn - opened[k];
          if (room < 0) room = 0;
          int64_t nf = rem / per_unit;
          nf = nf < room ? nf : room;
          int a = wave_alloc[w];
          if (nf > 0) {
            const int nn = a++;
#pragma unroll
            for (int r = 0; r < NRES; ++r)
              run_res[(int64_t)nn * NRES + r] =
                  (int64_t)pool_unit[k * NRES + r] - per_unit * rq[r];
            run_cnt[nn] = nf;
            ord[ob + R++] = nn;
            opened[k] += nf;
            rem -= nf * per_unit;
            room -= nf;
          }
          if (rem > 0 && room > 0) {  // one partial unit
            const int nn = a++;
#pragma unroll
            for (int r = 0; r < NRES; ++r)
              run_res[(int64_t)nn * NRES + r] =
                  (int64_t)pool_unit[k * NRES + r] - rem * rq[r];
            run_cnt[nn] = 1;
            ord[ob + R++] = nn;
            opened[k] += 1;
            rem = 0;
          }
          wave_alloc[w] = a;
        }
      }
      placed_new[g] = need0 - rem;
      unful[g] = rem;
      pool_R[k] = R;
    }
    __builtin_amdgcn_wave_barrier();
  }
  __syncthreads();
  for (int k = tid; k < K; k += UNITS_BLOCK) units_cnt[k] = opened[k];
  for (int q = tid; q < K * NRES; q += UNITS_BLOCK) pool_residual[q] = resid[q];
  if (blob_mirror != nullptr) {
    for (int i = tid; i < SM; i += UNITS_BLOCK) blob_mirror[i] = seg_sel_hash[i];
    if (tid < 3) blob_stat[tid] = stat[tid];
  }
}
