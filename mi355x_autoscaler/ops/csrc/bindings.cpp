// Python bindings for the MI355X autoscaler HIP kernels.
// Kernels live in ops/hip/*.hip (CDNA4-native; no CUDA-compat paths).
#include <torch/extension.h>
#include <pybind11/numpy.h>

void classify_keys(torch::Tensor pod_flags, torch::Tensor sel_hash,
                   torch::Tensor pod_req, torch::Tensor key_sel,
                   torch::Tensor key_req, torch::Tensor pending_flag);
void node_free_subtract(torch::Tensor pod_node, torch::Tensor pod_req,
                        torch::Tensor node_free);
void node_usage_sum(torch::Tensor pod_node, torch::Tensor pod_req,
                    torch::Tensor usage);
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
void drain_flags(torch::Tensor pod_node, torch::Tensor pod_flags,
                 torch::Tensor pod_pdb, torch::Tensor pdb_allowed,
                 torch::Tensor blocked, torch::Tensor pdb_bad,
                 torch::Tensor nontrivial, torch::Tensor drainable);
void drain_finalize(torch::Tensor util, torch::Tensor node_alloc,
                    torch::Tensor node_pool, torch::Tensor blocked,
                    torch::Tensor pdb_bad, torch::Tensor nontrivial,
                    int64_t util_threshold_permille, torch::Tensor util_permille,
                    torch::Tensor candidate);
void segsum_mfma(torch::Tensor values, torch::Tensor seg_of,
                 torch::Tensor wi_win, torch::Tensor wi_start,
                 torch::Tensor wi_end, int64_t num_segments, torch::Tensor out);
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
void pod_pool_map(torch::Tensor slot_of_pod, torch::Tensor slot2seg,
                  torch::Tensor seg_pool, torch::Tensor pod_pool);
void rows_hash(torch::Tensor rows, torch::Tensor slot_ks, torch::Tensor slot_kr,
               torch::Tensor slot_state, torch::Tensor slot_count,
               torch::Tensor slot_first, torch::Tensor overflow);
void rows_unpack(torch::Tensor rows, torch::Tensor pairs, torch::Tensor selc,
                 torch::Tensor req);
pybind11::dict encode_pods_json(pybind11::bytes body, pybind11::dict node_index,
                                pybind11::list pdb_selectors);
void pack_into(pybind11::buffer buf, const std::vector<int64_t>& offsets,
               pybind11::list arrays);
pybind11::dict decode_watch_events(pybind11::bytes body,
                                   pybind11::dict node_index,
                                   pybind11::list pdb_selectors);
pybind11::tuple resolve_batch_rows(
    pybind11::array_t<int8_t, pybind11::array::c_style | pybind11::array::forcecast> types,
    pybind11::list uids, pybind11::list refs, pybind11::dict uid_row,
    pybind11::list free_rows, int64_t size, int64_t cap,
    pybind11::list pod_uids, pybind11::list pod_refs);
void fused_tail_launch(
    torch::Tensor pod_req, torch::Tensor pod_node, torch::Tensor node_alloc,
    torch::Tensor node_labels, torch::Tensor node_label_count,
    torch::Tensor node_unsched, torch::Tensor pool_unit,
    torch::Tensor pool_labels, torch::Tensor pool_label_count,
    torch::Tensor pool_max, torch::Tensor pool_cur,
    torch::Tensor stat,
    torch::Tensor seg_sel_pairs, torch::Tensor seg_sel_count,
    torch::Tensor seg_sel_hash,
    torch::Tensor seg_req, torch::Tensor seg_count,
    torch::Tensor free, torch::Tensor elig, torch::Tensor seg_pool,
    torch::Tensor run_scratch, torch::Tensor blob,
    int64_t seg_max, int64_t want_free);
void fused_decide_launch(
    torch::Tensor pod_req, torch::Tensor pod_flags, torch::Tensor pod_sel_hash,
    torch::Tensor pod_sel_pairs, torch::Tensor pod_sel_count,
    torch::Tensor pod_node, torch::Tensor node_alloc, torch::Tensor node_labels,
    torch::Tensor node_label_count, torch::Tensor node_unsched,
    torch::Tensor pool_unit, torch::Tensor pool_labels,
    torch::Tensor pool_label_count, torch::Tensor pool_max, torch::Tensor pool_cur,
    torch::Tensor ws_i32, torch::Tensor slot_ks, torch::Tensor slot_kr,
    torch::Tensor slot2seg,
    torch::Tensor seg_sel_pairs, torch::Tensor seg_sel_count,
    torch::Tensor seg_sel_hash, torch::Tensor seg_req, torch::Tensor seg_count,
    torch::Tensor seg_key, torch::Tensor seg_first,
    torch::Tensor slot_of_pod, torch::Tensor free, torch::Tensor elig,
    torch::Tensor seg_pool, torch::Tensor run_scratch,
    torch::Tensor blob, torch::Tensor pod_pool, int64_t hash_slots,
    int64_t seg_max, int64_t want_pod_pool);
#include "decide_plan.h"
std::shared_ptr<DecidePlan> create_decide_plan(
    torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor,
    torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor,
    torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor,
    torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor,
    torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor,
    torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor,
    torch::Tensor, torch::Tensor, torch::Tensor, int64_t, int64_t, int64_t);
void run_decide_plan(const std::shared_ptr<DecidePlan>&);
void run_decide_plan_phase(const std::shared_ptr<DecidePlan>&, int64_t);
void set_plan_side(const std::shared_ptr<DecidePlan>&, torch::Tensor,
                   torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor,
                   int64_t);
int64_t run_decide_plan_auto(const std::shared_ptr<DecidePlan>&, int64_t);
void main_wait_side();
void side_sync();
pybind11::tuple finish_decision(torch::Tensor blob_host, int64_t K,
                                int64_t seg_max, int64_t over_provision,
                                torch::Tensor max_new, torch::Tensor pool_cur);
void snapshot_scatter(torch::Tensor rows, torch::Tensor u_req,
                      torch::Tensor u_flags, torch::Tensor u_sel_hash,
                      torch::Tensor u_sel_pairs, torch::Tensor u_sel_count,
                      torch::Tensor u_node, torch::Tensor u_pdb,
                      torch::Tensor pod_req, torch::Tensor pod_flags,
                      torch::Tensor pod_sel_hash, torch::Tensor pod_sel_pairs,
                      torch::Tensor pod_sel_count, torch::Tensor pod_node,
                      torch::Tensor pod_pdb);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "MI355X-native autoscaler scheduling kernels (gfx950 HIP)";
  m.def("classify_keys", &classify_keys, "kernel C: pod radix keys");
  m.def("node_free_subtract", &node_free_subtract, "node free capacity");
  m.def("node_usage_sum", &node_usage_sum, "per-node int64 usage totals");
  m.def("fit_eligibility", &fit_eligibility, "kernel A: segment x node eligibility");
  m.def("seg_pool_route", &seg_pool_route, "segment -> pool routing");
  m.def("binpack_place", &binpack_place, "kernel B: FFD placement steps");
  m.def("binpack_units", &binpack_units, "kernel B: units-needed phase",
        pybind11::arg("seg_pool"), pybind11::arg("seg_req"),
        pybind11::arg("seg_count"), pybind11::arg("placed"),
        pybind11::arg("pool_unit"), pybind11::arg("pool_max"),
        pybind11::arg("pool_cur"), pybind11::arg("g_dev"),
        pybind11::arg("run_scratch"), pybind11::arg("units_cnt"),
        pybind11::arg("placed_new"), pybind11::arg("unful"),
        pybind11::arg("pool_residual"),
        pybind11::arg("seg_sel_hash") = pybind11::none(),
        pybind11::arg("stat") = pybind11::none(),
        pybind11::arg("blob") = pybind11::none());
  m.def("drain_flags", &drain_flags, "kernel D: drain predicates");
  m.def("drain_finalize", &drain_finalize, "kernel D: utilization + candidates");
  m.def("segsum_mfma", &segsum_mfma, "MFMA segmented resource sums");
  m.def("segment_hash", &segment_hash, "kernel C: hash-aggregate pending pods");
  m.def("segment_compact", &segment_compact, "kernel C: compact+sort segments",
        pybind11::arg("slot_ks"), pybind11::arg("slot_kr"),
        pybind11::arg("slot_state"), pybind11::arg("slot_count"),
        pybind11::arg("slot_first"), pybind11::arg("pod_sel_pairs"),
        pybind11::arg("pod_sel_count"), pybind11::arg("pod_req"),
        pybind11::arg("seg_sel_pairs"), pybind11::arg("seg_sel_count"),
        pybind11::arg("seg_sel_hash"), pybind11::arg("seg_req"),
        pybind11::arg("seg_count"), pybind11::arg("seg_key"),
        pybind11::arg("seg_first_pod"), pybind11::arg("slot2seg"),
        pybind11::arg("g_out"), pybind11::arg("rows_out") = pybind11::none());
  m.def("pod_pool_map", &pod_pool_map, "pod -> routed pool map");
  m.def("rows_hash", &rows_hash, "hash-accumulate gathered segment rows");
  m.def("rows_unpack", &rows_unpack, "wire rows -> typed segment columns");
  m.def("encode_pods_json", &encode_pods_json,
        "PodList JSON -> pod SoA columns (native snapshot codec)");
  m.def("snapshot_scatter", &snapshot_scatter,
        "kernel E: delta-apply rows into the resident snapshot");
  m.def("pack_into", &pack_into, "memcpy arrays into the staging buffer");
  m.def("decode_watch_events", &decode_watch_events,
        "JSONL watch stream -> event types + pod SoA columns");
  m.def("resolve_batch_rows", &resolve_batch_rows,
        "watch batch -> target rows (C-speed uid/row bookkeeping)");
  m.def("compiled_constants", [] {
    pybind11::dict d;
    d["MAXSEL"] = 32;   // must match ops/hip/common.h and encode.py
    d["MAXLAB"] = 32;
    d["SEG_MAX"] = 2048;
    return d;
  }, "lattice constants this extension was compiled with");
  m.def("fused_decide_launch", &fused_decide_launch,
        "enqueue the whole scale-up pipeline in one call");
  m.def("fused_tail_launch", &fused_tail_launch,
        "enqueue the post-segment-build pipeline (sharded merged path)");
  m.def("finish_decision", &finish_decision,
        "host tail: pinned blob -> packed decision numbers (one C pass)");
  pybind11::class_<DecidePlan, std::shared_ptr<DecidePlan>>(m, "DecidePlan");
  m.def("create_decide_plan", &create_decide_plan,
        "bind the fused-decide tensor set once");
  m.def("run_decide_plan", &run_decide_plan,
        "re-enqueue the fused decide with zero argument conversion");
  m.def("run_decide_plan_phase", &run_decide_plan_phase,
        "classic pipeline phase: 1=head (through pod_pool_map), 2=tail");
  m.def("set_plan_side", &set_plan_side,
        "bind the side-stream demand segsum (work items + double buffers)");
  m.def("run_decide_plan_auto", &run_decide_plan_auto,
        "pipeline + side-stream demand segsum in ONE call; returns buffer idx");
  m.def("main_wait_side", &main_wait_side,
        "main stream waits for the last side segsum (stream-level)");
  m.def("side_sync", &side_sync, "host-block until the last side segsum");
}
