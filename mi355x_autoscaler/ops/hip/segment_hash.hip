// Device-side segment aggregation — the "radix-hash pass" of SURVEY
// §2.2 row C, as a lock-free hash table instead of a sort.
//
// Replaces the torch sort/unique/nonzero chain (~15 rocprim kernel
// launches per cycle, 92% of the decision's GPU time in the round-1
// profile) with two launches:
//
//   segment_hash_build: one pass over the pod SoA; pending pods CAS-
//     claim open-addressed slots keyed by (selector-set hash, request
//     pack key) and accumulate {count, min pod id} with device-scope
//     integer atomics (exact, order-free). Inter-workgroup publication
//     follows the agent-scope release/ready-flag discipline (guide §6
//     G16): the claimer stores both keys with relaxed agent-scope
//     atomics (L2-direct), then release-stores READY; probers read the
//     keys with relaxed agent-scope atomic loads, so no per-probe
//     acquire fence (no L1 involvement at all on the shared words).
//     Spins are bounded; on bound/overflow a flag routes the host to
//     the torch fallback path.
//
//   segment_compact_sort: ONE 1024-thread block compacts occupied
//     slots into LDS and bitonic-sorts them by (sel-hash as uint64
//     asc, request key desc) — the oracle's global segment order —
//     then writes the segment SoA + the slot->segment LUT.
//
// Capacity: H slots (pow2), <= SEG_MAX segments. Real bursts have
// O(10..100) distinct (selector, shape) pairs; 2048 is generous.
#include "pipeline_device.h"

// Two-level build: per-block LDS sub-table -> one global flush per
// distinct key per block. A burst has few distinct (selector, request)
// keys, so per-wave global atomics still serialized on the hot L2
// lines (413 us at 1M pods); aggregating in LDS first cuts the global
// atomic count from (waves x distinct) to (blocks x distinct).
//
// Intra-wave deadlock discipline: a lane spinning on another lane of
// the SAME wave never terminates (lockstep exec mask), so every
// claim/probe section runs leader-at-a-time via ballot/ffs — only one
// lane of a wave is ever inside a probe loop.
#define LDS_SLOTS 256

extern "C" __global__ __launch_bounds__(256) void segment_hash_build(
    const int32_t* __restrict__ pod_flags, const int64_t* __restrict__ sel_hash,
    const int32_t* __restrict__ pod_req, int64_t P, int32_t H,
    int64_t* __restrict__ slot_ks, int64_t* __restrict__ slot_kr,
    int32_t* __restrict__ slot_state, int32_t* __restrict__ slot_count,
    int32_t* __restrict__ slot_first, int32_t* __restrict__ slot_of_pod,
    int32_t* __restrict__ overflow) {
  __shared__ int64_t l_ks[LDS_SLOTS];
  __shared__ int64_t l_kr[LDS_SLOTS];
  __shared__ int32_t l_state[LDS_SLOTS];
  __shared__ int32_t l_count[LDS_SLOTS];
  __shared__ int32_t l_first[LDS_SLOTS];
  __shared__ int32_t l_gslot[LDS_SLOTS];

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  for (int q = tid; q < LDS_SLOTS; q += blockDim.x) {
    l_state[q] = ST_EMPTY;
    l_count[q] = 0;
    l_first[q] = 0;  // stores INT32_MAX - min(pod id) under atomicMax
  }
  __syncthreads();

  // pass 1: aggregate this block's pods into the LDS table; stash the
  // LDS slot id (or direct-global fallback) in slot_of_pod for pass 2.
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  // per-lane 1-entry memo WITH local run counting: bursts are runs of
  // identical (selector, request) keys, so the steady state per pod is
  // two register compares + a register increment — no LDS probe and NO
  // LDS atomic at all; counts/min-ids flush once per key RUN (memo
  // switch or loop end). Round-1 profile: the probe chain + per-pod
  // LDS atomics were 74.7M VALU / 218k bank conflicts at 1M pods.
  int64_t memo_ks = 0, memo_kr = 0;
  int memo_enc = INT32_MIN;
  int32_t memo_cnt = 0;
  int32_t memo_min_i0 = 0;
  auto memo_flush = [&]() {
    if (memo_enc >= 0 && memo_cnt > 0) {
      __hip_atomic_fetch_add(&l_count[memo_enc], memo_cnt, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_WORKGROUP);
      atomicMax(&l_first[memo_enc], INT32_MAX - memo_min_i0);
    }
    memo_cnt = 0;
  };
  for (int64_t i0 = (int64_t)blockIdx.x * blockDim.x + tid; ; i0 += stride) {
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

    // fast path: memo hit (count locally), else per-lane DIRECT probe
    // of already-claimed LDS slots — reads are deadlock-free (only
    // claims need the leader-at-a-time discipline); after the first
    // occurrence of each key per block every pod hits here
    int my_enc = -1;
    if (live) {
      if (memo_enc != INT32_MIN && ks == memo_ks && kr == memo_kr) {
        my_enc = memo_enc;
        if (memo_enc >= 0) {
          if (memo_cnt == 0) memo_min_i0 = (int32_t)i0;
          ++memo_cnt;
        }
      } else {
        uint32_t h = mix_hash(ks, kr) & (LDS_SLOTS - 1);
        for (int probe = 0; probe < 8; ++probe) {
          const int st = __hip_atomic_load(&l_state[h], __ATOMIC_RELAXED,
                                           __HIP_MEMORY_SCOPE_WORKGROUP);
          if (st == ST_EMPTY) break;  // key not claimed yet: slow path
          if (st == ST_READY) {
            __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "workgroup");
            if (l_ks[h] == ks && l_kr[h] == kr) {
              my_enc = (int)h;
              break;
            }
          }
          // CLAIMED or mismatch: next slot (never spin — claimer may
          // be a lane of this very wave)
          h = (h + 1u) & (LDS_SLOTS - 1);
        }
        if (my_enc >= 0) {
          memo_flush();
          memo_ks = ks;
          memo_kr = kr;
          memo_enc = my_enc;
          memo_cnt = 1;
          memo_min_i0 = (int32_t)i0;
        }
      }
    }

    uint64_t pending_lanes = __ballot(live && my_enc < 0);
    while (pending_lanes) {
      const int leader = __ffsll((unsigned long long)pending_lanes) - 1;
      const int64_t k0 = __shfl(ks, leader, WAVE);
      const int64_t r0 = __shfl(kr, leader, WAVE);
      const bool match = live && my_enc < 0 && ks == k0 && kr == r0;
      const uint64_t match_mask = __ballot(match);
      int enc = -1;
      if (lane == leader) {
        // LDS find-or-claim (leader-at-a-time => no intra-wave spin)
        uint32_t h = mix_hash(k0, r0) & (LDS_SLOTS - 1);
        for (int probe = 0; probe < LDS_SLOTS; ++probe) {
          int st = __hip_atomic_load(&l_state[h], __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_WORKGROUP);
          if (st == ST_EMPTY) {
            int expected = ST_EMPTY;
            if (__hip_atomic_compare_exchange_strong(
                    &l_state[h], &expected, ST_CLAIMED, __ATOMIC_RELAXED,
                    __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_WORKGROUP)) {
              l_ks[h] = k0;
              l_kr[h] = r0;
              __hip_atomic_store(&l_state[h], ST_READY, __ATOMIC_RELEASE,
                                 __HIP_MEMORY_SCOPE_WORKGROUP);
              enc = (int)h;
              break;
            }
            st = expected;
          }
          if (st == ST_CLAIMED) {
            while (__hip_atomic_load(&l_state[h], __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_WORKGROUP) == ST_CLAIMED) {
              __builtin_amdgcn_s_sleep(1);
            }
            st = ST_READY;
          }
          if (st == ST_READY) {
            __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "workgroup");
            if (l_ks[h] == k0 && l_kr[h] == r0) {
              enc = (int)h;
              break;
            }
          }
          h = (h + 1u) & (LDS_SLOTS - 1);
        }
        if (enc < 0) {
          // LDS table full: direct global path for this key
          const int g = probe_claim(k0, r0, H, slot_ks, slot_kr, slot_state, overflow);
          if (g >= 0) {
            atomicAdd(&slot_count[g], __popcll((unsigned long long)match_mask));
            atomicMax(&slot_first[g], INT32_MAX - (int32_t)i0);
            enc = -(g + 2);
          } else {
            atomicOr(overflow, 1);
            enc = INT32_MIN;  // marks unresolvable
          }
        } else {
          atomicAdd(&l_count[enc], __popcll((unsigned long long)match_mask));
          atomicMax(&l_first[enc], INT32_MAX - (int32_t)i0);
        }
      }
      enc = __shfl(enc, leader, WAVE);
      if (match) {
        my_enc = enc;
        if (enc >= 0) {
          // already counted by the leader; seed the memo so the run
          // continues on the fast path (cnt starts at 0)
          memo_flush();
          memo_ks = ks;
          memo_kr = kr;
          memo_enc = enc;
        }
      }
      pending_lanes &= ~match_mask;
    }
    if (i0 < P && live) slot_of_pod[i0] = my_enc;
  }
  memo_flush();
  __syncthreads();

  // flush: wave 0 resolves each occupied LDS slot to a global slot,
  // leader-at-a-time (same intra-wave discipline).
  if (tid < WAVE) {
    for (int base = 0; base < LDS_SLOTS; base += WAVE) {
      const int q = base + lane;
      const bool occupied = l_state[q] == ST_READY && l_count[q] > 0;
      uint64_t mask = __ballot(occupied);
      int gres = -1;
      while (mask) {
        const int leader = __ffsll((unsigned long long)mask) - 1;
        if (lane == leader) {
          const int g = probe_claim(l_ks[q], l_kr[q], H, slot_ks, slot_kr,
                                    slot_state, overflow);
          if (g >= 0) {
            atomicAdd(&slot_count[g], l_count[q]);
            atomicMax(&slot_first[g], l_first[q]);
            gres = g;
          } else {
            atomicOr(overflow, 1);
            gres = INT32_MIN;
          }
        }
        mask &= mask - 1;
      }
      if (occupied) l_gslot[q] = gres;
    }
  }
  __syncthreads();

  // pass 2: remap stashed LDS slot ids to global slot ids.
  for (int64_t i0 = (int64_t)blockIdx.x * blockDim.x + tid; i0 < P; i0 += stride) {
    const int v = slot_of_pod[i0];
    if (v >= 0) {
      const int g = l_gslot[v];
      slot_of_pod[i0] = g == INT32_MIN ? -1 : g;
    } else if (v != -1) {
      slot_of_pod[i0] = v == INT32_MIN ? -1 : -(v + 2);
    }
  }
}

// Small-P variant: below ~64k pods the LDS sub-table's fixed costs
// (table init, per-block flush) exceed the contention they avoid, so
// the wave-aggregated direct-global form wins; the host dispatches on
// P (segment_hash launcher below).
extern "C" __global__ void segment_hash_build_small(
    const int32_t* __restrict__ pod_flags, const int64_t* __restrict__ sel_hash,
    const int32_t* __restrict__ pod_req, int64_t P, int32_t H,
    int64_t* __restrict__ slot_ks, int64_t* __restrict__ slot_kr,
    int32_t* __restrict__ slot_state, int32_t* __restrict__ slot_count,
    int32_t* __restrict__ slot_first, int32_t* __restrict__ slot_of_pod,
    int32_t* __restrict__ overflow) {
  dev_hash_build_small(pod_flags, sel_hash, pod_req, P, H, slot_ks, slot_kr,
                       slot_state, slot_count, slot_first, slot_of_pod,
                       overflow);
}

// Sharded merge: hash-accumulate GATHERED SEGMENT ROWS (one row per
// (selector-set, request) segment per rank, counts pre-aggregated)
// into the same table the pod build uses — the host-side numpy merge
// cost ~180 us/rank of serialize + merge + workspace reload at the
// 8-GPU weak-scaling point. Padded rows (count <= 0) are skipped, so
// the all-gather needs no size compaction. Same wave-leader discipline;
// matched lanes' counts reduce in-wave before ONE atomic pair.
extern "C" __global__ void rows_hash_kernel(
    const int64_t* __restrict__ rows,    // [R, MAXSEL+8] gathered wire rows
    int64_t R, int32_t H,
    int64_t* __restrict__ slot_ks, int64_t* __restrict__ slot_kr,
    int32_t* __restrict__ slot_state, int32_t* __restrict__ slot_count,
    int32_t* __restrict__ slot_first, int32_t* __restrict__ overflow) {
  const int C = MAXSEL + 8;
  const int lane = threadIdx.x & (WAVE - 1);
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; ; i0 += stride) {
    if (__all(i0 >= R)) break;
    const int64_t* row = rows + i0 * C;
    if (i0 < R && row[MAXSEL + 5] < 0) {
      // poison row: some rank's LOCAL hash build overflowed; the
      // gather is symmetric so every rank sees it and falls back
      // together (no extra collective)
      atomicOr(overflow, 4);
    }
    const bool live = i0 < R && row[MAXSEL + 5] > 0;
    const int64_t ks = live ? row[MAXSEL + 6] : 0;
    const int64_t kr = live ? row[MAXSEL + 7] : 0;
    const int cnt = live ? (int)row[MAXSEL + 5] : 0;

    uint64_t pending = __ballot(live);
    while (pending) {
      const int leader = __ffsll((unsigned long long)pending) - 1;
      const int64_t k0 = __shfl(ks, leader, WAVE);
      const int64_t r0 = __shfl(kr, leader, WAVE);
      const bool match = live && ks == k0 && kr == r0;
      const uint64_t mm = __ballot(match);
      int contrib = match ? cnt : 0;
#pragma unroll
      for (int off = WAVE / 2; off; off >>= 1)
        contrib += __shfl_down(contrib, off, WAVE);
      const int total = __shfl(contrib, 0, WAVE);
      if (lane == leader) {
        const int slot = probe_claim(k0, r0, H, slot_ks, slot_kr, slot_state,
                                     overflow);
        if (slot >= 0) {
          atomicAdd(&slot_count[slot], total);
          // any matched row is a valid representative (identical
          // pairs/req for the same key); keep the min index
          atomicMax(&slot_first[slot], INT32_MAX - (int32_t)i0);
        } else {
          atomicOr(overflow, 1);
        }
      }
      pending &= ~mm;
    }
  }
}

// ---------------------------------------------------------------------------

#define CS_BLOCK 1024

extern "C" __global__ __launch_bounds__(CS_BLOCK) void segment_compact_sort(
    const int64_t* __restrict__ slot_ks, const int64_t* __restrict__ slot_kr,
    const int32_t* __restrict__ slot_state, const int32_t* __restrict__ slot_count,
    const int32_t* __restrict__ slot_first, int32_t H,
    const int64_t* __restrict__ pod_sel_pairs,  // [P, MAXSEL]
    const int32_t* __restrict__ pod_sel_count,
    const int32_t* __restrict__ pod_req,
    // outputs
    int64_t* __restrict__ seg_sel_pairs,  // [SEG_MAX, MAXSEL]
    int32_t* __restrict__ seg_sel_count, int64_t* __restrict__ seg_sel_hash,
    int32_t* __restrict__ seg_req, int64_t* __restrict__ seg_count,
    int64_t* __restrict__ seg_key, int32_t* __restrict__ seg_first_pod,
    int32_t* __restrict__ slot2seg,       // [H]
    int32_t* __restrict__ g_out,          // [2]: G, overflow(G>SEG_MAX)
    int64_t* __restrict__ rows_out)       // optional [SEG_MAX, MAXSEL+8]
{
  __shared__ uint64_t lds_ks[SEG_MAX];
  __shared__ uint64_t lds_kr[SEG_MAX];
  __shared__ int32_t lds_slot[SEG_MAX];
  __shared__ int32_t lds_n;
  dev_compact_sort(slot_ks, slot_kr, slot_state, slot_count, slot_first, H,
                   pod_sel_pairs, pod_sel_count, pod_req, seg_sel_pairs,
                   seg_sel_count, seg_sel_hash, seg_req, seg_count, seg_key,
                   seg_first_pod, slot2seg, g_out, lds_ks, lds_kr, lds_slot,
                   &lds_n, rows_out);
}

// pod -> routed-pool map (for the MFMA demand segsum): pool id of the
// pod's segment, or -1 for non-pending pods.
extern "C" __global__ void pod_pool_map_kernel(
    const int32_t* __restrict__ slot_of_pod, const int32_t* __restrict__ slot2seg,
    const int32_t* __restrict__ seg_pool, int64_t P,
    int32_t* __restrict__ pod_pool) {
  dev_pod_pool_map(slot_of_pod, slot2seg, seg_pool, P, pod_pool);
}

#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

void segment_hash(torch::Tensor pod_flags, torch::Tensor sel_hash,
                  torch::Tensor pod_req, torch::Tensor slot_ks,
                  torch::Tensor slot_kr, torch::Tensor slot_state,
                  torch::Tensor slot_count, torch::Tensor slot_first,
                  torch::Tensor slot_of_pod, torch::Tensor overflow) {
  const int64_t P = pod_flags.size(0);
  const int H = (int)slot_ks.size(0);
  if (P == 0) return;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const int block = 256;
  int grid = (int)std::min<int64_t>((P + block - 1) / block, 2048);
  // crossover between the direct wave-aggregated form and the LDS
  // sub-table form; measured ~64k on MI355X, overridable for tuning
  static int small_max = -1;
  if (small_max < 0) {
    const char* e = getenv("MI355X_AS_HASH_SMALL_MAX");
    small_max = e ? atoi(e) : 65536;
  }
  auto kfn = P < small_max ? segment_hash_build_small : segment_hash_build;
  if (P < small_max) {
    // the direct form serializes on per-key atomics: contention scales
    // with wave count, the scan itself is bandwidth-trivial, so run
    // ~4 pods per thread instead of 1 (18.2 -> measured lower at 10k)
    grid = (int)std::max<int64_t>(1, std::min<int64_t>((P + 1023) / 1024, 2048));
  } else {
    // LDS-table form: the per-BLOCK flush serializes on the global
    // slots (2048 blocks x ~distinct keys = 26k same-line atomics at
    // 1M pods); ~16 pods per thread cuts the flush count 8x while the
    // grid-strided scan stays bandwidth-bound
    grid = (int)std::max<int64_t>(8, std::min<int64_t>((P + 4095) / 4096, 2048));
  }
  hipLaunchKernelGGL(kfn, dim3(grid), dim3(block), 0,
                     stream.stream(), pod_flags.data_ptr<int32_t>(),
                     sel_hash.data_ptr<int64_t>(), pod_req.data_ptr<int32_t>(),
                     P, H, slot_ks.data_ptr<int64_t>(), slot_kr.data_ptr<int64_t>(),
                     slot_state.data_ptr<int32_t>(), slot_count.data_ptr<int32_t>(),
                     slot_first.data_ptr<int32_t>(), slot_of_pod.data_ptr<int32_t>(),
                     overflow.data_ptr<int32_t>());
}

void segment_compact(torch::Tensor slot_ks, torch::Tensor slot_kr,
                     torch::Tensor slot_state, torch::Tensor slot_count,
                     torch::Tensor slot_first, torch::Tensor pod_sel_pairs,
                     torch::Tensor pod_sel_count, torch::Tensor pod_req,
                     torch::Tensor seg_sel_pairs, torch::Tensor seg_sel_count,
                     torch::Tensor seg_sel_hash, torch::Tensor seg_req,
                     torch::Tensor seg_count, torch::Tensor seg_key,
                     torch::Tensor seg_first_pod,
                     torch::Tensor slot2seg, torch::Tensor g_out,
                     c10::optional<torch::Tensor> rows_out) {
  const int H = (int)slot_ks.size(0);
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  int64_t* rows_ptr = rows_out.has_value() ? rows_out->data_ptr<int64_t>() : nullptr;
  hipLaunchKernelGGL(segment_compact_sort, dim3(1), dim3(CS_BLOCK), 0,
                     stream.stream(), slot_ks.data_ptr<int64_t>(),
                     slot_kr.data_ptr<int64_t>(), slot_state.data_ptr<int32_t>(),
                     slot_count.data_ptr<int32_t>(), slot_first.data_ptr<int32_t>(),
                     H, pod_sel_pairs.data_ptr<int64_t>(),
                     pod_sel_count.data_ptr<int32_t>(), pod_req.data_ptr<int32_t>(),
                     seg_sel_pairs.data_ptr<int64_t>(),
                     seg_sel_count.data_ptr<int32_t>(),
                     seg_sel_hash.data_ptr<int64_t>(), seg_req.data_ptr<int32_t>(),
                     seg_count.data_ptr<int64_t>(), seg_key.data_ptr<int64_t>(),
                     seg_first_pod.data_ptr<int32_t>(),
                     slot2seg.data_ptr<int32_t>(), g_out.data_ptr<int32_t>(),
                     rows_ptr);
}

void pod_pool_map(torch::Tensor slot_of_pod, torch::Tensor slot2seg,
                  torch::Tensor seg_pool, torch::Tensor pod_pool) {
  const int64_t P = slot_of_pod.size(0);
  if (P == 0) return;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const int block = 256;
  const int grid = (int)std::min<int64_t>((P + block - 1) / block, 2048);
  hipLaunchKernelGGL(pod_pool_map_kernel, dim3(grid), dim3(block), 0,
                     stream.stream(), slot_of_pod.data_ptr<int32_t>(),
                     slot2seg.data_ptr<int32_t>(), seg_pool.data_ptr<int32_t>(),
                     P, pod_pool.data_ptr<int32_t>());
}

void rows_hash(torch::Tensor rows, torch::Tensor slot_ks, torch::Tensor slot_kr,
               torch::Tensor slot_state, torch::Tensor slot_count,
               torch::Tensor slot_first, torch::Tensor overflow) {
  const int64_t R = rows.size(0);
  const int H = (int)slot_ks.size(0);
  if (R == 0) return;
  TORCH_CHECK(rows.size(1) == MAXSEL + 8, "rows must be [R, MAXSEL+8]");
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const int block = 256;
  const int grid = (int)std::max<int64_t>(1, std::min<int64_t>((R + 1023) / 1024, 256));
  hipLaunchKernelGGL(rows_hash_kernel, dim3(grid), dim3(block), 0,
                     stream.stream(), rows.data_ptr<int64_t>(), R, H,
                     slot_ks.data_ptr<int64_t>(), slot_kr.data_ptr<int64_t>(),
                     slot_state.data_ptr<int32_t>(), slot_count.data_ptr<int32_t>(),
                     slot_first.data_ptr<int32_t>(), overflow.data_ptr<int32_t>());
}

// gathered wire rows -> the typed "pod" columns segment_compact reads
// (ONE kernel instead of ~8 torch slice/cast dispatches per step)
extern "C" __global__ void rows_unpack_kernel(
    const int64_t* __restrict__ rows, int64_t R,
    int64_t* __restrict__ pairs, int32_t* __restrict__ selc,
    int32_t* __restrict__ req) {
  const int C = MAXSEL + 8;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < R;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t* row = rows + i * C;
#pragma unroll
    for (int q = 0; q < MAXSEL; ++q) pairs[i * MAXSEL + q] = row[q];
    selc[i] = (int32_t)row[MAXSEL];
#pragma unroll
    for (int r = 0; r < NRES; ++r) req[i * NRES + r] = (int32_t)row[MAXSEL + 1 + r];
  }
}

void rows_unpack(torch::Tensor rows, torch::Tensor pairs, torch::Tensor selc,
                 torch::Tensor req) {
  const int64_t R = rows.size(0);
  if (R == 0) return;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const int block = 256;
  const int grid = (int)std::max<int64_t>(1, std::min<int64_t>((R + 255) / 256, 1024));
  hipLaunchKernelGGL(rows_unpack_kernel, dim3(grid), dim3(block), 0,
                     stream.stream(), rows.data_ptr<int64_t>(), R,
                     pairs.data_ptr<int64_t>(), selc.data_ptr<int32_t>(),
                     req.data_ptr<int32_t>());
}
