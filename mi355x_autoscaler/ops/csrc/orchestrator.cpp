// Native decision orchestrator: enqueues the ENTIRE fused scale-up
// pipeline in one Python->C++ call. The Python fused path issues ~30
// individual op calls per decide (~5 us of dispatcher each — a large
// share of the 0.35 ms wall at 10k pods); here the per-call cost is
// paid once and every kernel launch is back-to-back on the stream.
// The Python implementation (engine_gpu._decide_fused) remains as the
// reference/fallback and the two are differential-tested.
#include <torch/extension.h>

#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <hip/hip_runtime_api.h>

// launchers defined in the .hip translation units
void segment_hash(torch::Tensor pod_flags, torch::Tensor sel_hash,
                  torch::Tensor pod_req, torch::Tensor slot_ks,
                  torch::Tensor slot_kr, torch::Tensor slot_state,
                  torch::Tensor slot_count, torch::Tensor slot_first,
                  torch::Tensor slot_of_pod, torch::Tensor overflow);
void segment_compact(torch::Tensor slot_ks, torch::Tensor slot_kr,
                     torch::Tensor slot_state, torch::Tensor slot_count,
                     torch::Tensor slot_first, torch::Tensor pod_sel_pairs,
                     torch::Tensor pod_sel_count, torch::Tensor pod_req,
                     torch::Tensor seg_sel_pairs, torch::Tensor seg_sel_count,
                     torch::Tensor seg_sel_hash, torch::Tensor seg_req,
                     torch::Tensor seg_count, torch::Tensor seg_key,
                     torch::Tensor seg_first_pod,
                     torch::Tensor slot2seg, torch::Tensor g_out,
                     c10::optional<torch::Tensor> rows_out);
void node_free_subtract(torch::Tensor pod_node, torch::Tensor pod_req,
                        torch::Tensor node_free);
void fit_eligibility(torch::Tensor node_labels, torch::Tensor node_label_count,
                     torch::Tensor node_unsched, torch::Tensor seg_sel,
                     torch::Tensor seg_sel_count, torch::Tensor g_dev,
                     torch::Tensor elig);
void seg_pool_route(torch::Tensor seg_sel, torch::Tensor seg_sel_count,
                    torch::Tensor seg_req, torch::Tensor pool_labels,
                    torch::Tensor pool_label_count, torch::Tensor pool_unit,
                    torch::Tensor g_dev, torch::Tensor seg_pool);
void binpack_place(torch::Tensor free, torch::Tensor elig, torch::Tensor seg_req,
                   torch::Tensor seg_count, torch::Tensor g_dev,
                   torch::Tensor placed_out);
void binpack_units(torch::Tensor seg_pool, torch::Tensor seg_req,
                   torch::Tensor seg_count, torch::Tensor placed,
                   torch::Tensor pool_unit, torch::Tensor pool_max,
                   torch::Tensor pool_cur, torch::Tensor g_dev,
                   torch::Tensor run_scratch, torch::Tensor units_cnt,
                   torch::Tensor placed_new, torch::Tensor unful,
                   torch::Tensor pool_residual,
                   c10::optional<torch::Tensor> seg_sel_hash,
                   c10::optional<torch::Tensor> stat,
                   c10::optional<torch::Tensor> blob);
void pod_pool_map(torch::Tensor slot_of_pod, torch::Tensor slot2seg,
                  torch::Tensor seg_pool, torch::Tensor pod_pool);
bool mega_decide_launch(
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
    int64_t seg_max, int64_t want_pod_pool, uint64_t* demand_out);
bool mega_decide_single_launch(
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
    int64_t seg_max, int64_t want_pod_pool, uint64_t* demand_out);

// set by run_decide_plan_auto for the next fused launch: the mega
// kernel computes the per-pool demand segsum in-kernel (idle blocks
// during the serial tail) instead of a side-stream launch. Cleared by
// fused_decide_launch; g_mega_demand_used reports consumption.
static uint64_t* g_mega_demand = nullptr;
static bool g_mega_demand_used = false;

void fused_classic_phase(
    torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor,
    torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor,
    torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor,
    torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor,
    torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor,
    torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor,
    torch::Tensor, torch::Tensor, torch::Tensor, int64_t, int64_t, int64_t,
    int64_t);

void fused_decide_launch(
    // snapshot (device views)
    torch::Tensor pod_req, torch::Tensor pod_flags, torch::Tensor pod_sel_hash,
    torch::Tensor pod_sel_pairs, torch::Tensor pod_sel_count,
    torch::Tensor pod_node, torch::Tensor node_alloc, torch::Tensor node_labels,
    torch::Tensor node_label_count, torch::Tensor node_unsched,
    torch::Tensor pool_unit, torch::Tensor pool_labels,
    torch::Tensor pool_label_count, torch::Tensor pool_max, torch::Tensor pool_cur,
    // hash workspace (ws_i32 = state|count|first|stat contiguous)
    torch::Tensor ws_i32, torch::Tensor slot_ks, torch::Tensor slot_kr,
    torch::Tensor slot2seg,
    // segment workspace
    torch::Tensor seg_sel_pairs, torch::Tensor seg_sel_count,
    torch::Tensor seg_sel_hash, torch::Tensor seg_req, torch::Tensor seg_count,
    torch::Tensor seg_key, torch::Tensor seg_first,
    // per-decide buffers
    torch::Tensor slot_of_pod, torch::Tensor free, torch::Tensor elig,
    torch::Tensor seg_pool, torch::Tensor run_scratch,
    torch::Tensor blob, torch::Tensor pod_pool, int64_t hash_slots,
    int64_t seg_max, int64_t want_pod_pool) {
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  const int64_t H = hash_slots;
  const int64_t SM = seg_max;
  const int64_t K = pool_unit.size(0);
  const int64_t N = node_alloc.size(0);
  const int64_t P = pod_flags.size(0);
  // dev_units keeps per-pool state in fixed LDS (UNITS_MAX_POOLS=64);
  // without this check K>64 would write out of LDS bounds. The Python
  // engine falls back to the CPU oracle before reaching here.
  TORCH_CHECK(K <= 64, "fused decide supports at most 64 agent pools, got ", K);

  // small/medium-P regime: ONE cooperative launch runs every phase
  // (mega_decide.hip); falls through to the classic multi-launch
  // pipeline on large P, degenerate shapes, or when cooperative
  // launch is unavailable. MI355X_AS_NO_MEGA=1 forces classic.
  static int mega_env = -1;
  static int64_t mega_max_p = 0;
  static int64_t single_max_p = 0;
  if (mega_env < 0) {
    mega_env = getenv("MI355X_AS_NO_MEGA") ? 0 : 1;
    const char* mp = getenv("MI355X_AS_MEGA_MAX_P");
    mega_max_p = mp ? atoll(mp) : 131072;
    // single-workgroup regime: regular launch, __syncthreads-only
    // barriers. 0 disables. Crossover measured on MI355X (see
    // profiles/r02_single_*): ~P 1000-1500, earlier on node-heavy
    // shapes — single wins at (1000 pods, 8 nodes) and (500, 100)
    // but loses at (1000, 100) and (2000, any), so the gate is
    // (P <= max && N <= 32) || P <= max/2.
    const char* sp = getenv("MI355X_AS_SINGLE_MAX_P");
    single_max_p = sp ? atoll(sp) : 1024;
  }
  uint64_t* demand = g_mega_demand;
  g_mega_demand = nullptr;
  g_mega_demand_used = false;
  if (mega_env && single_max_p > 0 && N > 0 && K > 0 && P > 0 &&
      ((P <= single_max_p && N <= 32) || P <= single_max_p / 2) && N <= 2048) {
    if (mega_decide_single_launch(
            pod_req, pod_flags, pod_sel_hash, pod_sel_pairs, pod_sel_count,
            pod_node, node_alloc, node_labels, node_label_count, node_unsched,
            pool_unit, pool_labels, pool_label_count, pool_max, pool_cur,
            ws_i32, slot_ks, slot_kr, slot2seg, seg_sel_pairs, seg_sel_count,
            seg_sel_hash, seg_req, seg_count, seg_key, seg_first, slot_of_pod,
            free, elig, seg_pool, run_scratch, blob, pod_pool, hash_slots,
            seg_max, want_pod_pool, demand)) {
      g_mega_demand_used = (demand != nullptr && want_pod_pool);
      return;
    }
  }
  if (mega_env && N > 0 && K > 0 && P > 0 && P <= mega_max_p && N <= 65536) {
    if (mega_decide_launch(pod_req, pod_flags, pod_sel_hash, pod_sel_pairs,
                           pod_sel_count, pod_node, node_alloc, node_labels,
                           node_label_count, node_unsched, pool_unit,
                           pool_labels, pool_label_count, pool_max, pool_cur,
                           ws_i32, slot_ks, slot_kr, slot2seg, seg_sel_pairs,
                           seg_sel_count, seg_sel_hash, seg_req, seg_count,
                           seg_key, seg_first, slot_of_pod, free, elig,
                           seg_pool, run_scratch, blob, pod_pool, hash_slots,
                           seg_max, want_pod_pool, demand)) {
      g_mega_demand_used = (demand != nullptr && want_pod_pool);
      return;
    }
  }

  fused_classic_phase(pod_req, pod_flags, pod_sel_hash, pod_sel_pairs,
                      pod_sel_count, pod_node, node_alloc, node_labels,
                      node_label_count, node_unsched, pool_unit, pool_labels,
                      pool_label_count, pool_max, pool_cur, ws_i32, slot_ks,
                      slot_kr, slot2seg, seg_sel_pairs, seg_sel_count,
                      seg_sel_hash, seg_req, seg_count, seg_key, seg_first,
                      slot_of_pod, free, elig, seg_pool, run_scratch, blob,
                      pod_pool, hash_slots, seg_max, want_pod_pool, 3);
}

// Classic multi-launch pipeline with a PHASE mask: 1 = head (hash ->
// compact -> free -> eligibility -> route -> pod_pool_map), 2 = tail
// (FFD place + units/epilogue), 3 = both. The 1M resident path runs
// head, forks the side-stream MFMA demand segsum off pod_pool, then
// runs the tail — the single-workgroup place/units kernels leave
// 255/256 CUs idle, exactly where the segsum overlaps.
void fused_classic_phase(
    torch::Tensor pod_req, torch::Tensor pod_flags, torch::Tensor pod_sel_hash,
    torch::Tensor pod_sel_pairs, torch::Tensor pod_sel_count,
    torch::Tensor pod_node, torch::Tensor node_alloc, torch::Tensor node_labels,
    torch::Tensor node_label_count, torch::Tensor node_unsched,
    torch::Tensor pool_unit, torch::Tensor pool_labels,
    torch::Tensor pool_label_count, torch::Tensor pool_max, torch::Tensor pool_cur,
    torch::Tensor ws_i32, torch::Tensor slot_ks, torch::Tensor slot_kr,
    torch::Tensor slot2seg, torch::Tensor seg_sel_pairs,
    torch::Tensor seg_sel_count, torch::Tensor seg_sel_hash,
    torch::Tensor seg_req, torch::Tensor seg_count, torch::Tensor seg_key,
    torch::Tensor seg_first, torch::Tensor slot_of_pod, torch::Tensor free,
    torch::Tensor elig, torch::Tensor seg_pool, torch::Tensor run_scratch,
    torch::Tensor blob, torch::Tensor pod_pool, int64_t hash_slots,
    int64_t seg_max, int64_t want_pod_pool, int64_t phase) {
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  const int64_t H = hash_slots;
  const int64_t SM = seg_max;
  const int64_t K = pool_unit.size(0);
  const int64_t N = node_alloc.size(0);

  auto slot_state = ws_i32.narrow(0, 0, H);
  auto slot_count = ws_i32.narrow(0, H, H);
  auto slot_first = ws_i32.narrow(0, 2 * H, H);
  auto stat = ws_i32.narrow(0, 3 * H, 3);
  auto g_dev = stat.narrow(0, 0, 1);
  auto overflow = stat.narrow(0, 2, 1);

  if (phase & 1) {
    // workspace reinit; the blob needs zeroing only when a producer
    // kernel is skipped (binpack writes placed[:G], units writes the
    // rest + the readback epilogue)
    hipMemsetAsync(ws_i32.data_ptr<int32_t>(), 0,
                   (size_t)ws_i32.numel() * sizeof(int32_t), stream);
    if (N == 0 || K == 0) {
      hipMemsetAsync(blob.data_ptr<int64_t>(), 0,
                     (size_t)blob.numel() * sizeof(int64_t), stream);
    }

    segment_hash(pod_flags, pod_sel_hash, pod_req, slot_ks, slot_kr, slot_state,
                 slot_count, slot_first, slot_of_pod, overflow);
    segment_compact(slot_ks, slot_kr, slot_state, slot_count, slot_first,
                    pod_sel_pairs, pod_sel_count, pod_req, seg_sel_pairs,
                    seg_sel_count, seg_sel_hash, seg_req, seg_count, seg_key,
                    seg_first, slot2seg, stat.narrow(0, 0, 2), c10::nullopt);

    // free = alloc - bound requests
    hipMemcpyAsync(free.data_ptr<int32_t>(), node_alloc.data_ptr<int32_t>(),
                   (size_t)node_alloc.numel() * sizeof(int32_t),
                   hipMemcpyDeviceToDevice, stream);
    node_free_subtract(pod_node, pod_req, free);

    if (N > 0) {
      fit_eligibility(node_labels, node_label_count, node_unsched, seg_sel_pairs,
                      seg_sel_count, g_dev, elig);
    }
    seg_pool_route(seg_sel_pairs, seg_sel_count, seg_req, pool_labels,
                   pool_label_count, pool_unit, g_dev, seg_pool);
    if (want_pod_pool) {
      pod_pool_map(slot_of_pod, slot2seg, seg_pool, pod_pool);
    }
  }

  if (phase & 2) {
    auto placed = blob.narrow(0, 0, SM);
    auto placed_new = blob.narrow(0, SM, SM);
    auto unful = blob.narrow(0, 2 * SM, SM);
    if (N > 0) {
      binpack_place(free, elig, seg_req, seg_count, g_dev, placed);
    }
    if (K > 0) {
      auto units_cnt = blob.narrow(0, 3 * SM, K);
      auto pool_residual = blob.narrow(0, 3 * SM + K, 4 * K);
      // units also runs the readback epilogue: segment sel-hash column +
      // 3-word stat (G, sort-ovf, hash-ovf) land on the blob tail so the
      // host needs exactly ONE readback (the sel-hash feeds
      // unfulfillable_detail — without the mirror that path cost a
      // second D2H sync per decide)
      binpack_units(seg_pool, seg_req, seg_count, placed, pool_unit, pool_max,
                    pool_cur, g_dev, run_scratch, units_cnt, placed_new, unful,
                    pool_residual, seg_sel_hash.narrow(0, 0, SM), stat, blob);
    } else {
      hipMemcpyAsync(blob.data_ptr<int64_t>() + (blob.numel() - 2 - SM),
                     seg_sel_hash.data_ptr<int64_t>(), (size_t)SM * sizeof(int64_t),
                     hipMemcpyDeviceToDevice, stream);
      hipMemcpyAsync(blob.data_ptr<int64_t>() + (blob.numel() - 2),
                     stat.data_ptr<int32_t>(), 3 * sizeof(int32_t),
                     hipMemcpyDeviceToDevice, stream);
    }
  }
}


// Tail-only variant for the SHARDED path: the merged global segment
// list (all-gathered + merged by the host) is already in the seg
// workspace and stat[0] = G; this enqueues everything after the
// segment build. want_free=0 means the caller already produced `free`
// (the cross-rank usage all-reduce sits between subtract and place).
void fused_tail_launch(
    torch::Tensor pod_req, torch::Tensor pod_node, torch::Tensor node_alloc,
    torch::Tensor node_labels, torch::Tensor node_label_count,
    torch::Tensor node_unsched, torch::Tensor pool_unit,
    torch::Tensor pool_labels, torch::Tensor pool_label_count,
    torch::Tensor pool_max, torch::Tensor pool_cur,
    torch::Tensor stat,            // [3] i32: G, 0, 0 (host-set)
    torch::Tensor seg_sel_pairs, torch::Tensor seg_sel_count,
    torch::Tensor seg_sel_hash,
    torch::Tensor seg_req, torch::Tensor seg_count,
    torch::Tensor free, torch::Tensor elig, torch::Tensor seg_pool,
    torch::Tensor run_scratch, torch::Tensor blob,
    int64_t seg_max, int64_t want_free) {
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  const int64_t SM = seg_max;
  const int64_t K = pool_unit.size(0);
  const int64_t N = node_alloc.size(0);
  auto g_dev = stat.narrow(0, 0, 1);

  if (N == 0 || K == 0) {
    hipMemsetAsync(blob.data_ptr<int64_t>(), 0,
                   (size_t)blob.numel() * sizeof(int64_t), stream);
  }
  if (want_free) {
    hipMemcpyAsync(free.data_ptr<int32_t>(), node_alloc.data_ptr<int32_t>(),
                   (size_t)node_alloc.numel() * sizeof(int32_t),
                   hipMemcpyDeviceToDevice, stream);
    node_free_subtract(pod_node, pod_req, free);
  }
  if (N > 0) {
    fit_eligibility(node_labels, node_label_count, node_unsched, seg_sel_pairs,
                    seg_sel_count, g_dev, elig);
  }
  seg_pool_route(seg_sel_pairs, seg_sel_count, seg_req, pool_labels,
                 pool_label_count, pool_unit, g_dev, seg_pool);
  auto placed = blob.narrow(0, 0, SM);
  auto placed_new = blob.narrow(0, SM, SM);
  auto unful = blob.narrow(0, 2 * SM, SM);
  if (N > 0) {
    binpack_place(free, elig, seg_req, seg_count, g_dev, placed);
  }
  if (K > 0) {
    auto units_cnt = blob.narrow(0, 3 * SM, K);
    auto pool_residual = blob.narrow(0, 3 * SM + K, 4 * K);
    binpack_units(seg_pool, seg_req, seg_count, placed, pool_unit, pool_max,
                  pool_cur, g_dev, run_scratch, units_cnt, placed_new, unful,
                  pool_residual, seg_sel_hash.narrow(0, 0, SM), stat, blob);
  } else {
    hipMemcpyAsync(blob.data_ptr<int64_t>() + (blob.numel() - 2),
                   stat.data_ptr<int32_t>(), 3 * sizeof(int32_t),
                   hipMemcpyDeviceToDevice, stream);
  }
}

// Host tail of the fused decide: one pass over the
// pinned blob mirror replaces ~30 us of numpy sums / nonzero / Python
// pool loop per decision. Pure host code — no stream work; the caller
// already synchronized via the pinned copy_.
// Returns (G, sort_ovf, hash_ovf, placed_existing, placed_new,
//          unfulfillable, units[K] (over-provision applied),
//          detail [(sel_hash_u64, count)...], residual i64 [K,4]).
pybind11::tuple finish_decision(torch::Tensor blob_host, int64_t K,
                                int64_t seg_max, int64_t over_provision,
                                torch::Tensor max_new,    // i64 [K] host
                                torch::Tensor pool_cur) { // i32 [K] host
  TORCH_CHECK(!blob_host.is_cuda() && blob_host.is_contiguous());
  const int64_t* b = blob_host.data_ptr<int64_t>();
  const int64_t SM = seg_max;
  const int64_t nb = blob_host.numel();
  const int32_t* st = reinterpret_cast<const int32_t*>(b + nb - 2);
  const int64_t G = st[0];
  const int64_t sort_ovf = st[1], hash_ovf = st[2];
  namespace py = pybind11;
  auto units = py::list();
  auto detail = py::list();
  auto residual = torch::zeros({K, 4}, torch::kInt64);
  if (sort_ovf || hash_ovf || G < 0 || G > SM) {
    return py::make_tuple(G, sort_ovf, hash_ovf, 0, 0, 0, units, detail,
                          residual);
  }
  int64_t placed_e = 0, placed_n = 0, unful_t = 0;
  for (int64_t g = 0; g < G; ++g) placed_e += b[g];
  for (int64_t g = 0; g < G; ++g) placed_n += b[SM + g];
  const int64_t* unful = b + 2 * SM;
  const int64_t* selh = b + (nb - 2 - SM);
  for (int64_t g = 0; g < G; ++g) {
    if (unful[g] > 0) {
      unful_t += unful[g];
      detail.append(py::make_tuple((uint64_t)selh[g], unful[g]));
    }
  }
  const int64_t* mn = max_new.data_ptr<int64_t>();
  const int32_t* pc = pool_cur.data_ptr<int32_t>();
  for (int64_t k = 0; k < K; ++k) {
    int64_t n = b[3 * SM + k];
    if (n > 0 && over_provision > 0) {
      n = std::min(n + over_provision, mn[k]);
    }
    units.append(n);
  }
  std::memcpy(residual.data_ptr<int64_t>(), b + 3 * SM + K,
              (size_t)(4 * K) * sizeof(int64_t));
  (void)pc;
  return py::make_tuple(G, sort_ovf, hash_ovf, placed_e, placed_n, unful_t,
                        units, detail, residual);
}


// Pre-bound launch plan (host-tail work, round 2): converting 33
// torch::Tensor arguments through pybind costs tens of microseconds
// per decide; a plan binds them once and run() re-enqueues the whole
// pipeline with ZERO per-call argument conversion. The Python side
// caches plans keyed by the tensor pointer set (same discipline as
// the hipGraph cache).
#include "decide_plan.h"

std::shared_ptr<DecidePlan> create_decide_plan(
    torch::Tensor pod_req, torch::Tensor pod_flags, torch::Tensor pod_sel_hash,
    torch::Tensor pod_sel_pairs, torch::Tensor pod_sel_count,
    torch::Tensor pod_node, torch::Tensor node_alloc, torch::Tensor node_labels,
    torch::Tensor node_label_count, torch::Tensor node_unsched,
    torch::Tensor pool_unit, torch::Tensor pool_labels,
    torch::Tensor pool_label_count, torch::Tensor pool_max, torch::Tensor pool_cur,
    torch::Tensor ws_i32, torch::Tensor slot_ks, torch::Tensor slot_kr,
    torch::Tensor slot2seg, torch::Tensor seg_sel_pairs,
    torch::Tensor seg_sel_count, torch::Tensor seg_sel_hash,
    torch::Tensor seg_req, torch::Tensor seg_count, torch::Tensor seg_key,
    torch::Tensor seg_first, torch::Tensor slot_of_pod, torch::Tensor free,
    torch::Tensor elig, torch::Tensor seg_pool, torch::Tensor run_scratch,
    torch::Tensor blob, torch::Tensor pod_pool, int64_t hash_slots,
    int64_t seg_max, int64_t want_pod_pool) {
  auto p = std::make_shared<DecidePlan>();
  p->t = {pod_req, pod_flags, pod_sel_hash, pod_sel_pairs, pod_sel_count,
          pod_node, node_alloc, node_labels, node_label_count, node_unsched,
          pool_unit, pool_labels, pool_label_count, pool_max, pool_cur,
          ws_i32, slot_ks, slot_kr, slot2seg, seg_sel_pairs, seg_sel_count,
          seg_sel_hash, seg_req, seg_count, seg_key, seg_first, slot_of_pod,
          free, elig, seg_pool, run_scratch, blob, pod_pool};
  p->hash_slots = hash_slots;
  p->seg_max = seg_max;
  p->want_pod_pool = want_pod_pool;
  return p;
}

void run_decide_plan_phase(const std::shared_ptr<DecidePlan>& p,
                           int64_t phase) {
  auto& t = p->t;
  fused_classic_phase(t[0], t[1], t[2], t[3], t[4], t[5], t[6], t[7], t[8],
                      t[9], t[10], t[11], t[12], t[13], t[14], t[15], t[16],
                      t[17], t[18], t[19], t[20], t[21], t[22], t[23], t[24],
                      t[25], t[26], t[27], t[28], t[29], t[30], t[31], t[32],
                      p->hash_slots, p->seg_max, p->want_pod_pool, phase);
}

void run_decide_plan(const std::shared_ptr<DecidePlan>& p) {
  auto& t = p->t;
  fused_decide_launch(t[0], t[1], t[2], t[3], t[4], t[5], t[6], t[7], t[8],
                      t[9], t[10], t[11], t[12], t[13], t[14], t[15], t[16],
                      t[17], t[18], t[19], t[20], t[21], t[22], t[23], t[24],
                      t[25], t[26], t[27], t[28], t[29], t[30], t[31], t[32],
                      p->hash_slots, p->seg_max, p->want_pod_pool);
}


// ---------------------------------------------------------------------------
// Orchestrator-owned side stream for the demand segsum. Statics are
// per process (one GPU per process in every deployment shape); the
// wait-on-unrecorded-event is a defined no-op, so the first call needs
// no special casing.
void segsum_mfma_on(torch::Tensor values, torch::Tensor seg_of,
                    torch::Tensor wi_win, torch::Tensor wi_start,
                    torch::Tensor wi_end, int64_t num_segments,
                    torch::Tensor out, hipStream_t stream);

static hipStream_t g_side_stream = nullptr;
static hipEvent_t g_fork_ev = nullptr;
static hipEvent_t g_join_ev = nullptr;

static void side_init() {
  if (g_side_stream == nullptr) {
    hipStreamCreateWithFlags(&g_side_stream, hipStreamNonBlocking);
    hipEventCreateWithFlags(&g_fork_ev, hipEventDisableTiming);
    hipEventCreateWithFlags(&g_join_ev, hipEventDisableTiming);
  }
}

void set_plan_side(const std::shared_ptr<DecidePlan>& p,
                   torch::Tensor wi_win, torch::Tensor wi_start,
                   torch::Tensor wi_end, torch::Tensor demand_a,
                   torch::Tensor demand_b, int64_t K) {
  p->wi_win = wi_win;
  p->wi_start = wi_start;
  p->wi_end = wi_end;
  p->demand_a = demand_a;
  p->demand_b = demand_b;
  p->K = K;
  p->has_side = true;
}

// Run the whole decide INCLUDING the side-stream demand segsum in one
// Python->C++ call. `split` selects the classic head/fork/tail overlap
// (large P); otherwise the full pipeline (mega at small P) runs first
// and the segsum forks after it. Returns which demand buffer was
// written (0 = a, 1 = b), or -1 when the plan has no side work.
int64_t run_decide_plan_auto(const std::shared_ptr<DecidePlan>& p,
                             int64_t split) {
  auto& t = p->t;
  if (!p->has_side) {
    run_decide_plan(p);
    return -1;
  }
  side_init();
  auto main_stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  // order THIS pipeline (which rewrites pod_pool / pod columns) after
  // the previous decide's segsum — stream-level, normally long past
  hipStreamWaitEvent(main_stream, g_join_ev, 0);
  const int use_b = p->flip;
  p->flip ^= 1;
  torch::Tensor out = use_b ? p->demand_b : p->demand_a;
  if (split) {
    run_decide_plan_phase(p, 1);  // classic head: through pod_pool_map
  } else {
    // mega path: hand the demand buffer to the kernel — its idle
    // blocks compute the segsum during the serial tail (no side
    // stream, no extra launch). Classic fallback leaves it unused.
    hipMemsetAsync(out.data_ptr<int64_t>(), 0,
                   (size_t)out.numel() * sizeof(int64_t), main_stream);
    g_mega_demand = reinterpret_cast<uint64_t*>(out.data_ptr<int64_t>());
    run_decide_plan(p);           // mega or full classic
    if (g_mega_demand_used) {
      // demand produced on the MAIN stream; keep the join-event
      // contract for pool_demand()/resident-scatter ordering
      hipEventRecord(g_join_ev, main_stream);
      return use_b;
    }
  }
  hipEventRecord(g_fork_ev, main_stream);
  hipStreamWaitEvent(g_side_stream, g_fork_ev, 0);
  hipMemsetAsync(out.data_ptr<int64_t>(), 0,
                 (size_t)out.numel() * sizeof(int64_t), g_side_stream);
  segsum_mfma_on(t[0], t[32], p->wi_win, p->wi_start, p->wi_end, p->K, out,
                 g_side_stream);
  hipEventRecord(g_join_ev, g_side_stream);
  if (split) {
    run_decide_plan_phase(p, 2);  // classic tail overlaps the segsum
  }
  return use_b;
}

// main stream orders after the last side segsum (resident scatter
// rewrites pod columns the segsum may still read)
void main_wait_side() {
  side_init();
  hipStreamWaitEvent(at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream(),
                     g_join_ev, 0);
}

// host-blocking: pool_demand() hands the tensor out
void side_sync() {
  side_init();
  hipEventSynchronize(g_join_ev);
}
