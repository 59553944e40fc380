"""Per-op GPU phase timing of the CLASSIC decide pipeline via
torch.cuda events — the rocprof replacement while rocprofv3 segfaults
on the current box image. Mirrors GpuEngine._enqueue_py's sequence and
prints averaged per-op times for the 10k (config #3) and gpu1k
(config #2) shapes, to attribute the small-shape residual."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.ops.engine_gpu import SEG_MAX, GpuEngine, segsum
from mi355x_autoscaler.sim.synth import burst_trace, wrap
from mi355x_autoscaler.snapshot.encode import encode_snapshot

REPS = 200


def build(pods, sizes, pools, gpufrac):
    pods_json, nodes_json = burst_trace(
        pods, n_nodes_per_pool=sizes, pools=pools, gpu_fraction=gpufrac,
        seed=0, bound_pods_per_node=2)
    ps, ns = wrap(pods_json, nodes_json)
    return encode_snapshot(ps, ns, group_nodes_into_pools(ns))


def probe(label, snap, eng):
    ds = eng.packer.pack(snap, eng.device)
    N, K = ds.N, ds.K
    want_pod_pool = bool(K and K <= 16)
    bufs = eng._decide_buffers(ds.P, K, want_pod_pool)
    slot_of_pod, blob = bufs["slot_of_pod"], bufs["blob"]
    pod_pool = bufs["pod_pool"] if want_pod_pool else None
    ops = eng.ops
    g_dev = eng.ws_stat[0:1]
    free = eng._free_buffer(N)
    W = (N + 63) // 64
    elig = eng._elig_buffer(W)

    steps = [
        ("ws zero", lambda: (eng._ws_i32.zero_(), blob.zero_())),
        ("segment_hash", lambda: ops.segment_hash(
            ds.pod_flags, ds.pod_sel_hash, ds.pod_req, eng.ws_slot_ks,
            eng.ws_slot_kr, eng.ws_slot_state, eng.ws_slot_count,
            eng.ws_slot_first, slot_of_pod, eng.ws_stat[2:3])),
        ("segment_compact", lambda: ops.segment_compact(
            eng.ws_slot_ks, eng.ws_slot_kr, eng.ws_slot_state,
            eng.ws_slot_count, eng.ws_slot_first, ds.pod_sel_pairs,
            ds.pod_sel_count, ds.pod_req, eng.ws_seg_sel_pairs,
            eng.ws_seg_sel_count, eng.ws_seg_sel_hash, eng.ws_seg_req,
            eng.ws_seg_count, eng.ws_seg_key, eng.ws_seg_first,
            eng.ws_slot2seg, eng.ws_stat[0:2])),
        ("node_free", lambda: (free.copy_(ds.node_alloc),
                               ops.node_free_subtract(ds.pod_node, ds.pod_req, free))),
        ("fit_eligibility", lambda: ops.fit_eligibility(
            ds.node_labels, ds.node_label_count, ds.node_unsched,
            eng.ws_seg_sel_pairs, eng.ws_seg_sel_count, g_dev, elig)),
        ("seg_pool_route", lambda: ops.seg_pool_route(
            eng.ws_seg_sel_pairs, eng.ws_seg_sel_count, eng.ws_seg_req,
            ds.pool_labels, ds.pool_label_count, ds.pool_unit, g_dev,
            eng.ws_seg_pool)),
        ("pod_pool_map", lambda: ops.pod_pool_map(
            slot_of_pod, eng.ws_slot2seg, eng.ws_seg_pool, pod_pool)
            if want_pod_pool else None),
        ("binpack_place", lambda: ops.binpack_place(
            free, elig, eng.ws_seg_req, eng.ws_seg_count, g_dev,
            blob[0:SEG_MAX])),
        ("binpack_units", lambda: ops.binpack_units(
            eng.ws_seg_pool, eng.ws_seg_req, eng.ws_seg_count, blob[0:SEG_MAX],
            ds.pool_unit, ds.pool_max, ds.pool_cur, g_dev, eng.ws_run_scratch,
            blob[3 * SEG_MAX : 3 * SEG_MAX + K], blob[SEG_MAX : 2 * SEG_MAX],
            blob[2 * SEG_MAX : 3 * SEG_MAX],
            blob[3 * SEG_MAX + K : 3 * SEG_MAX + 5 * K], eng.ws_seg_sel_hash,
            eng.ws_stat, blob) if K else None),
        ("segsum demand", lambda: segsum(ds.pod_req, pod_pool, K, ops)
            if want_pod_pool else None),
        ("blob D2H", lambda: bufs["blob_host"].copy_(blob)),
    ]
    # warm
    for _ in range(5):
        for _, fn in steps:
            fn()
    torch.cuda.synchronize()
    evs = [(name, torch.cuda.Event(True), torch.cuda.Event(True))
           for name, _ in steps]
    acc = {name: 0.0 for name, _ in steps}
    for _ in range(REPS):
        for (name, fn), (_, e0, e1) in zip(steps, evs):
            e0.record()
            fn()
            e1.record()
        torch.cuda.synchronize()
        for name, e0, e1 in evs:
            acc[name] += e0.elapsed_time(e1)
    print(f"\n[{label}] P={ds.P} N={N} K={K} (us/op over {REPS} reps, "
          "event-timed, serialized)")
    total = 0.0
    for name, _ in steps:
        us = acc[name] / REPS * 1000.0
        total += us
        print(f"  {name:16s} {us:8.1f} us")
    print(f"  {'TOTAL':16s} {total:8.1f} us")
    stat = eng.ws_stat.cpu().numpy()
    print(f"  G={int(stat[0])}")


def main():
    eng = GpuEngine("cuda:0")
    snap10k = build(10000, (40, 30, 30),
                    (("agentpool1", "Standard_D2_v2"),
                     ("bigpool", "Standard_D4_v2"),
                     ("gpupool", "Standard_ND96_MI355X_v1")), 0.35)
    probe("burst10k", snap10k, eng)
    snap1k = build(1000, (8,), (("gpupool", "Standard_ND96_MI355X_v1"),), 1.0)
    probe("gpu1k", snap1k, eng)


if __name__ == "__main__":
    main()
