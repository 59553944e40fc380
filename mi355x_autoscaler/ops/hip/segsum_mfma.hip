// MFMA-backed segmented sums over resource vectors (gfx950).
//
// The north-star op [SURVEY §2.2 row B]: per-segment totals of int32
// resource vectors (segments = nodes for utilization, = pools for
// demand), computed on the matrix cores as
//
//   D[row][s] = Σ_p A[row][p] · B[p][s]
//
// with `v_mfma_f32_16x16x4_f32` (exact f32 MFMA at the vector rate —
// guide §3: bit-exact fmaf chain, no TF32 on gfx950). Rows encode
// (limb, resource): each int32 value is split into 16-bit limbs
// (lo = v & 0xffff, hi = v >> 16), 8 of the 16 A-rows carry data. B is
// the segment-indicator built in-register from the sorted segment ids.
// Exactness: a 256-element accumulation chain keeps every partial
// < 2^24 (lo ≤ 256·(2^16−1) < 2^24, hi ≤ 256·(2^15−1) < 2^23), so the
// f32 MFMA result is the exact integer sum; limbs recombine into int64
// with a device-scope atomicAdd per (segment, resource) per chunk.
// Integer atomics commute exactly => deterministic for any chunk/shard
// decomposition (the RCCL agreement requirement).
//
// Work decomposition: elements must arrive sorted by segment id.
// Segments are tiled 16 per MFMA *window*; a work item = (window,
// 256-element chunk of that window's element range). Each wave walks
// work items grid-strided, accumulating the exact per-chunk integer
// partials in int64 REGISTERS across chunks of the same window and
// flushing once per window — the flush-per-chunk form serialized
// ~30k atomics onto K*NRES hot words (202 us at 1M pods, the whole
// kernel; the MFMA math itself is microseconds).
#include "common.h"

typedef float f32x4 __attribute__((ext_vector_type(4)));

#define SS_WAVES 4  // waves per block, each owning its own chunk + LDS

extern "C" __global__ __launch_bounds__(WAVE * SS_WAVES) void segsum_mfma_kernel(
    const int32_t* __restrict__ values,   // [M, NRES] sorted by seg
    const int32_t* __restrict__ seg_of,   // [M] ascending
    const int32_t* __restrict__ wi_win,   // [T] window per work item
    const int64_t* __restrict__ wi_start, // [T] element start
    const int64_t* __restrict__ wi_end,   // [T] element end (window bound)
    int32_t S,                            // number of segments
    int64_t T,
    uint64_t* __restrict__ out)           // [S, NRES] (non-negative i64)
{
  // The whole 256-element chunk is staged cooperatively into LDS first
  // (4 coalesced dwordx4 loads + 1 dword load per lane, all in flight
  // together), then the MFMA loop feeds from LDS broadcasts — the
  // fragment-at-a-time global loads of v1 serialized on L2 latency
  // (220 us at 1M pods).
  //
  // Round 2: 4 waves per block (each with a private LDS region and its
  // own grid-strided work-item walk — no cross-wave barriers) and FOUR
  // independent MFMA accumulator chains per chunk. The v2 single-wave
  // blocks left 3/4 of the SIMDs idle and the lone dependent MFMA
  // chain exposed the full 16x16x4 latency every step (74.9 us at 1M);
  // exactness is unchanged — each chain sums <= 64 limb values < 2^22,
  // recombined through the same int64 window partials.
  __shared__ int32_t l_vals[SS_WAVES][256 * NRES];
  __shared__ int32_t l_segs[SS_WAVES][256];
  // single-window case (S <= 16, the per-pool demand sum): waves
  // accumulate into BLOCK LDS first and one wave flushes per block —
  // direct per-wave global atomics serialized ~8k adds onto S*NRES
  // hot words at 1M pods (the whole kernel was flush-bound)
  __shared__ unsigned long long s_acc[16 * NRES];

  const bool single_win = S <= 16;
  if (single_win) {
    for (int idx = threadIdx.x; idx < 16 * NRES; idx += blockDim.x)
      s_acc[idx] = 0ull;
    __syncthreads();
  }

  const int lane = threadIdx.x & (WAVE - 1);
  const int wv = threadIdx.x / WAVE;
  const int col = lane & 15;      // A row index i / B,D column j
  const int kidx = lane >> 4;     // K slot 0..3 (one pod each)

  // A-row decode for this lane: rows 0..7 = (limb = row>>2, res = row&3)
  const int row = col;
  const int res = row & 3;
  const int limb = (row >> 2) & 1;
  const bool live_row = row < 8;

  int cur_win = -1;
  long long p64[4] = {0, 0, 0, 0};  // exact int64 window partials

  auto flush = [&](int win) {
    const int s = win * 16 + col;
    if (s < S) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int orow = (lane >> 4) * 4 + reg;
        if (orow < 8) {
          const int ores = orow & 3;
          const int olimb = (orow >> 2) & 1;
          const unsigned long long partial = (unsigned long long)p64[reg];
          if (partial) {
            const unsigned long long v = olimb ? (partial << 16) : partial;
            if (single_win) {
              atomicAdd(&s_acc[s * NRES + ores], v);  // LDS, block-local
            } else {
              atomicAdd(reinterpret_cast<unsigned long long*>(
                            &out[(int64_t)s * NRES + ores]), v);
            }
          }
        }
      }
    }
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) p64[reg] = 0;
  };

  int32_t* lv = l_vals[wv];
  int32_t* ls = l_segs[wv];
  const int64_t wstride = (int64_t)gridDim.x * SS_WAVES;
  for (int64_t t = (int64_t)blockIdx.x * SS_WAVES + wv; t < T; t += wstride) {
    const int win = wi_win[t];
    if (win != cur_win) {
      if (cur_win >= 0) flush(cur_win);
      cur_win = win;
    }
    const int64_t p_begin = wi_start[t];
    const int64_t p_end = wi_end[t];
    const int seg_base = win * 16;

    // stage: lane covers pods {lane, lane+64, lane+128, lane+192}
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      const int pl = lane + 64 * q;
      const int64_t p = p_begin + pl;
      if (p < p_end) {
        const int4 v = reinterpret_cast<const int4*>(values)[p];
        reinterpret_cast<int4*>(lv)[pl] = v;
        ls[pl] = seg_of[p] - seg_base;
      } else {
        reinterpret_cast<int4*>(lv)[pl] = int4{0, 0, 0, 0};
        ls[pl] = -1;  // no column matches: contributes zero
      }
    }
    // private LDS region: only this wave's in-order LDS ops touch it
    __builtin_amdgcn_wave_barrier();

    const int steps = (int)((p_end - p_begin + 3) >> 2);
    f32x4 acc[4] = {{0.f, 0.f, 0.f, 0.f},
                    {0.f, 0.f, 0.f, 0.f},
                    {0.f, 0.f, 0.f, 0.f},
                    {0.f, 0.f, 0.f, 0.f}};
    int sstep = 0;
    for (; sstep + 4 <= steps; sstep += 4) {
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const int pl = (sstep + u) * 4 + kidx;
        const float b = (ls[pl] == col) ? 1.f : 0.f;
        float a = 0.f;
        if (live_row) {
          const uint32_t v = (uint32_t)lv[pl * NRES + res];
          a = (float)(limb ? (v >> 16) : (v & 0xffffu));
        }
        acc[u] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc[u], 0, 0, 0);
      }
    }
    for (; sstep < steps; ++sstep) {
      const int pl = sstep * 4 + kidx;
      const float b = (ls[pl] == col) ? 1.f : 0.f;
      float a = 0.f;
      if (live_row) {
        const uint32_t v = (uint32_t)lv[pl * NRES + res];
        a = (float)(limb ? (v >> 16) : (v & 0xffffu));
      }
      acc[0] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc[0], 0, 0, 0);
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) p64[reg] += (long long)acc[u][reg];
    }
    __builtin_amdgcn_wave_barrier();  // LDS region reused next chunk
  }
  if (cur_win >= 0) flush(cur_win);
  if (single_win) {
    __syncthreads();  // all waves' LDS partials in
    for (int idx = threadIdx.x; idx < S * NRES; idx += blockDim.x) {
      const unsigned long long v = s_acc[idx];
      if (v) {
        atomicAdd(reinterpret_cast<unsigned long long*>(out) + idx, v);
      }
    }
  }
}

#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

// values/seg_of sorted by segment; work items precomputed on device by
// ops/engine_gpu.py (window = seg // 16, chunks of <=256 elements that
// never cross a window boundary). out must be zeroed.
void segsum_mfma_on(torch::Tensor values, torch::Tensor seg_of,
                    torch::Tensor wi_win, torch::Tensor wi_start,
                    torch::Tensor wi_end, int64_t num_segments,
                    torch::Tensor out, hipStream_t stream) {
  const int64_t T = wi_win.size(0);
  if (T == 0) return;
  // SS_WAVES waves per block; cap at 256 blocks = 1024 waves (one per
  // SIMD). MORE waves would hand each wave ~1 chunk and collapse the
  // flush-once-per-window amortization back into flush-per-chunk
  // device atomics on K*NRES hot words — measured 202 us at 1M pods
  // (vs 75 us) when the grid ran 3908 waves.
  const uint32_t grid =
      (uint32_t)std::min<int64_t>((T + SS_WAVES - 1) / SS_WAVES, 256);
  hipLaunchKernelGGL(segsum_mfma_kernel, dim3(grid), dim3(WAVE * SS_WAVES), 0,
                     stream, values.data_ptr<int32_t>(),
                     seg_of.data_ptr<int32_t>(), wi_win.data_ptr<int32_t>(),
                     wi_start.data_ptr<int64_t>(), wi_end.data_ptr<int64_t>(),
                     (int32_t)num_segments, T,
                     reinterpret_cast<uint64_t*>(out.data_ptr<int64_t>()));
}

void segsum_mfma(torch::Tensor values, torch::Tensor seg_of,
                 torch::Tensor wi_win, torch::Tensor wi_start,
                 torch::Tensor wi_end, int64_t num_segments,
                 torch::Tensor out) {
  segsum_mfma_on(values, seg_of, wi_win, wi_start, wi_end, num_segments, out,
                 at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream());
}
