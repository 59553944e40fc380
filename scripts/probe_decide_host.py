"""Fine-grained host-side timing of _decide_fused at 10k (mega path):
where do the ~90 us between GPU-busy and wall go?"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.ops import engine_gpu as eg
from mi355x_autoscaler.ops.engine_gpu import SEG_MAX, GpuEngine
from mi355x_autoscaler.sim.synth import burst_trace, wrap
from mi355x_autoscaler.snapshot.encode import encode_snapshot


def main():
    pods_json, nodes_json = burst_trace(
        10000, n_nodes_per_pool=(40, 30, 30),
        pools=(("agentpool1", "Standard_D2_v2"), ("bigpool", "Standard_D4_v2"),
               ("gpupool", "Standard_ND96_MI355X_v1")),
        gpu_fraction=0.35, seed=0, bound_pods_per_node=3)
    pods, nodes = wrap(pods_json, nodes_json)
    snap = encode_snapshot(pods, nodes, group_nodes_into_pools(nodes))
    eng = GpuEngine("cuda:0")
    for _ in range(5):
        eng.decide(snap)
    ds = eng.packer.pack(snap, eng.device)
    K = ds.K

    # instrument one _decide_fused manually (mirrors the real path)
    N = 200
    marks = {k: 0.0 for k in ("bufs", "launch", "fork", "sync", "finish")}
    torch.cuda.synchronize()
    t_total0 = time.perf_counter()
    for _ in range(N):
        t0 = time.perf_counter()
        W = (ds.N + 63) // 64
        elig = eng._elig_buffer(W)
        want = True
        bufs = eng._decide_buffers(ds.P, K, want)
        blob = bufs["blob"]
        pod_pool = bufs["pod_pool"]
        tensors = (
            ds.pod_req, ds.pod_flags, ds.pod_sel_hash, ds.pod_sel_pairs,
            ds.pod_sel_count, ds.pod_node, ds.node_alloc, ds.node_labels,
            ds.node_label_count, ds.node_unsched, ds.pool_unit,
            ds.pool_labels, ds.pool_label_count, ds.pool_max, ds.pool_cur,
            eng._ws_i32, eng.ws_slot_ks, eng.ws_slot_kr, eng.ws_slot2seg,
            eng.ws_seg_sel_pairs, eng.ws_seg_sel_count, eng.ws_seg_sel_hash,
            eng.ws_seg_req, eng.ws_seg_count, eng.ws_seg_key, eng.ws_seg_first,
            bufs["slot_of_pod"], eng._free_buffer(ds.N), elig, eng.ws_seg_pool,
            eng.ws_run_scratch, blob, pod_pool)
        scalars = (eg.HASH_SLOTS, SEG_MAX, 1)
        pkey = scalars + tuple(t.data_ptr() for t in tensors)
        plan = eng._plan_cache.get(pkey)
        if plan is None:
            plan = eng.ops.create_decide_plan(*tensors, *scalars)
            eng._plan_cache[pkey] = plan
        t1 = time.perf_counter()
        eng.ops.run_decide_plan(plan)
        t2 = time.perf_counter()
        eng._fork_demand_segsum(ds, pod_pool, K, False)
        t3 = time.perf_counter()
        bufs["blob_host"].copy_(blob)
        t4 = time.perf_counter()
        blob_h = bufs["blob_host"]
        max_new_h = np.maximum(
            ds.pool_max_h.astype(np.int64) - ds.pool_cur_h, 0)
        out = eng.ops.finish_decision(
            blob_h, K, SEG_MAX, 0,
            torch.from_numpy(np.ascontiguousarray(max_new_h)),
            torch.from_numpy(ds.pool_cur_h))
        t5 = time.perf_counter()
        marks["bufs"] += t1 - t0
        marks["launch"] += t2 - t1
        marks["fork"] += t3 - t2
        marks["sync"] += t4 - t3
        marks["finish"] += t5 - t4
    torch.cuda.synchronize()
    total = (time.perf_counter() - t_total0) / N * 1e3
    for k, v in marks.items():
        print(f"{k:8s}: {v / N * 1e3:7.3f} ms")
    print(f"total   : {total:7.3f} ms")
    # reference: the real call
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(N):
        eng._decide_fused(ds, snap.pool_names)
    torch.cuda.synchronize()
    print(f"_decide_fused: {(time.perf_counter() - t0) / N * 1e3:7.3f} ms")


if __name__ == "__main__":
    main()
