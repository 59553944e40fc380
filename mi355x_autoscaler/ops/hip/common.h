// Shared device-side definitions for the MI355X autoscaler kernels.
//
// Everything here mirrors the Python lattice in snapshot/encode.py and
// sim/oracle.py exactly — resource-row order, flag bits, selector-hash
// matching — so CPU-oracle and device decisions are bit-identical.
// Target: gfx950 (CDNA4), wave64. No CUDA-compat paths.
#pragma once

#include <hip/hip_runtime.h>
#include <stdint.h>

#define NRES 4            // cpu_milli, mem_mib, pods, gpu — fixed row order
#define MAXSEL 32         // padded selector pair-hash width (pods); ==
                          // MAXLAB so subset matching is exact for every
                          // representable selector (>32 pairs: poisoned
                          // with 0xFFFF…F, matches nothing — encode.py)
#define MAXLAB 32         // padded label pair-hash width (nodes/pools)
#define WAVE 64

// PodFlags — must match snapshot/encode.py::PodFlags
#define PF_PENDING (1 << 0)
#define PF_MIRROR (1 << 1)
#define PF_DAEMONSET (1 << 2)
#define PF_CRITICAL (1 << 3)
#define PF_REPLICATED (1 << 4)
#define PF_DRAINABLE (1 << 5)
#define PF_BLOCKS_DRAIN (1 << 6)
#define PF_SEL_OVERFLOW (1 << 7)
#define PF_BOUND (1 << 8)

// Request sort-key packing — must match sim/oracle.py::_pack_req_key.
// Critical pods [R: reference critical-pod annotation] outrank every
// non-critical shape within their selector group: bit 55 sits above
// the resource fields, so descending key order places them first.
__device__ __host__ inline int64_t pack_req_key(const int32_t req[NRES],
                                                int critical) {
  int64_t cpu = req[0] < 0 ? 0 : (int64_t)req[0];
  int64_t mem = req[1] < 0 ? 0 : (int64_t)req[1];
  int64_t gpu = req[3] < 0 ? 0 : (int64_t)req[3];
  if (cpu > (1 << 21) - 1) cpu = (1 << 21) - 1;
  if (mem > (1 << 23) - 1) mem = (1 << 23) - 1;
  if (gpu > (1 << 8) - 1) gpu = (1 << 8) - 1;
  return ((int64_t)(critical ? 1 : 0) << 55) | (cpu << 31) | (mem << 8) | gpu;
}

// Selector-subset predicate over sorted pair-hash arrays — device form of
// snapshot/encode.py::selector_subset_match. `sel` is wave-uniform in
// every caller; `lab` may live in LDS or global. The producer sorts
// the raw 64-bit FNV hashes UNSIGNED before the two's-complement int64
// cast, so the walk compares unsigned (a signed walk silently fails
// whenever a selector mixes positive- and negative-cast hashes).
__device__ inline bool sel_subset(const int64_t* __restrict__ sel, int nsel,
                                  const int64_t* __restrict__ lab, int nlab) {
  int li = 0;
  for (int si = 0; si < nsel; ++si) {
    const uint64_t s = (uint64_t)sel[si];
    while (li < nlab && (uint64_t)lab[li] < s) ++li;
    if (li >= nlab || (uint64_t)lab[li] != s) return false;
  }
  return true;
}

// min over resources of floor(free_r / req_r); 0 if ineligible or any
// axis negative. Device form of OracleSimulator._per_node_capacity.
__device__ inline int64_t per_node_capacity(const int32_t f[NRES],
                                            const int32_t rq[NRES]) {
  int64_t m = (int64_t)1 << 31;
#pragma unroll
  for (int r = 0; r < NRES; ++r) {
    if (f[r] < 0) return 0;
    if (rq[r] > 0) {
      // both operands proven non-negative: 32-bit unsigned divide
      // (the i64 form was the hot scalar op of the single-CU FFD walk)
      int64_t q = (int64_t)((uint32_t)f[r] / (uint32_t)rq[r]);
      m = q < m ? q : m;
    }
  }
  return m;
}

// ---- wave/block scan helpers (int64) ------------------------------------

__device__ inline int64_t wave_inclusive_scan_i64(int64_t v) {
#pragma unroll
  for (int off = 1; off < WAVE; off <<= 1) {
    int64_t other = __shfl_up((long long)v, off, WAVE);
    if ((threadIdx.x & (WAVE - 1)) >= off) v += other;
  }
  return v;
}

// Block-wide EXCLUSIVE scan over one value per thread. `lds` must hold
// blockDim.x/WAVE (+1) int64 slots. Returns the exclusive prefix; the
// block total is written to *total (all threads see it after return).
template <int BLOCK>
__device__ inline int64_t block_exclusive_scan_i64(int64_t v, int64_t* lds,
                                                   int64_t* total) {
  constexpr int NWAVES = BLOCK / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  int64_t inc = wave_inclusive_scan_i64(v);
  if (lane == WAVE - 1) lds[wid] = inc;
  __syncthreads();
  if (wid == 0) {
    int64_t w = (lane < NWAVES) ? lds[lane] : 0;
    int64_t winc = wave_inclusive_scan_i64(w);
    if (lane < NWAVES) lds[lane] = winc - w;  // exclusive wave offsets
    if (lane == NWAVES - 1) lds[NWAVES] = winc;
  }
  __syncthreads();
  int64_t ex = lds[wid] + inc - v;
  *total = lds[NWAVES];
  __syncthreads();
  return ex;
}

#define HIP_CHECK_LAST()                                        \
  do {                                                          \
    hipError_t e_ = hipGetLastError();                          \
    if (e_ != hipSuccess) {                                     \
      return e_;                                                \
    }                                                           \
  } while (0)
