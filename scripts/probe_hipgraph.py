"""Isolate hipGraph replay cost vs eager enqueue of the fused pipeline.

The first integration of graph capture REGRESSED the 10k decide 0.24 ->
0.94 ms/step; this probe separates (a) hipGraphLaunch overhead, (b)
replay-side sync behavior, (c) capture cost, so the engine default can
be set from data rather than the CUDA-era assumption that graphs are
always cheaper.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.ops.engine_gpu import GpuEngine
from mi355x_autoscaler.sim.synth import burst_trace, wrap
from mi355x_autoscaler.snapshot.encode import encode_snapshot


def ms(fn, n=60):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


def main():
    pods_json, nodes_json = burst_trace(
        10000, n_nodes_per_pool=(40, 30, 30),
        pools=(("agentpool1", "Standard_D2_v2"), ("bigpool", "Standard_D4_v2"),
               ("gpupool", "Standard_ND96_MI355X_v1")),
        gpu_fraction=0.35, seed=0, bound_pods_per_node=3)
    pods, nodes = wrap(pods_json, nodes_json)
    pools = group_nodes_into_pools(nodes)
    snap = encode_snapshot(pods, nodes, pools)

    eng = GpuEngine("cuda:0")
    eng._graph_ok = False
    for _ in range(3):
        eng.decide(snap)
    print(f"decide eager          : {ms(lambda: eng.decide(snap)):8.3f} ms")

    # raw launch cost without readback
    ds = eng.packer.pack(snap, eng.device)
    import mi355x_autoscaler.ops.engine_gpu as eg

    K = ds.K
    W = (ds.N + 63) // 64
    elig = eng._elig_buffer(W)
    bufs = eng._decide_buffers(ds.P, K, True)
    args = (ds.pod_req, ds.pod_flags, ds.pod_sel_hash, ds.pod_sel_pairs,
            ds.pod_sel_count, ds.pod_node, ds.node_alloc, ds.node_labels,
            ds.node_label_count, ds.node_unsched, ds.pool_unit,
            ds.pool_labels, ds.pool_label_count, ds.pool_max, ds.pool_cur,
            eng._ws_i32, eng.ws_slot_ks, eng.ws_slot_kr, eng.ws_slot2seg,
            eng.ws_seg_sel_pairs, eng.ws_seg_sel_count, eng.ws_seg_sel_hash,
            eng.ws_seg_req, eng.ws_seg_count, eng.ws_seg_key, eng.ws_seg_first,
            bufs["slot_of_pod"], eng._free_buffer(ds.N), elig, eng.ws_seg_pool,
            bufs["max_new"], eng.ws_run_scratch, bufs["blob"], bufs["pod_pool"])
    scal = (eg.HASH_SLOTS, eg.SEG_MAX, 1)
    fn = eng.ops.fused_decide_launch
    fn(*args, *scal)
    print(f"pipeline eager sync   : {ms(lambda: fn(*args, *scal)):8.3f} ms")

    t0 = time.perf_counter()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        fn(*args, *scal)
    torch.cuda.synchronize()
    print(f"capture (one-time)    : {(time.perf_counter() - t0) * 1e3:8.3f} ms")
    g.replay()
    torch.cuda.synchronize()
    print(f"graph replay sync     : {ms(lambda: g.replay()):8.3f} ms")

    # enqueue-only (no sync in the loop): dispatch cost comparison
    def burst_eager():
        for _ in range(20):
            fn(*args, *scal)
    def burst_graph():
        for _ in range(20):
            g.replay()
    print(f"eager enqueue+run /20 : {ms(burst_eager, 10) / 20:8.3f} ms")
    print(f"replay enqueue+run /20: {ms(burst_graph, 10) / 20:8.3f} ms")




def probe_decide_graph():
    """Reproduce the bench regression: decide() with graphs enabled,
    instrumenting the cache to see captures vs replays per step."""
    pods_json, nodes_json = burst_trace(
        10000, n_nodes_per_pool=(40, 30, 30),
        pools=(("agentpool1", "Standard_D2_v2"), ("bigpool", "Standard_D4_v2"),
               ("gpupool", "Standard_ND96_MI355X_v1")),
        gpu_fraction=0.35, seed=0, bound_pods_per_node=3)
    pods, nodes = wrap(pods_json, nodes_json)
    pools = group_nodes_into_pools(nodes)
    snap = encode_snapshot(pods, nodes, pools)

    eng = GpuEngine("cuda:0")
    assert eng._graph_ok
    for i in range(6):
        t0 = time.perf_counter()
        eng.decide(snap)
        dt = (time.perf_counter() - t0) * 1e3
        kinds = sorted(type(v).__name__ for v in eng._graph_cache.values())
        print(f"decide[{i}]: {dt:7.3f} ms  cache={len(eng._graph_cache)} {kinds}")
    print(f"decide graph steady   : {ms(lambda: eng.decide(snap)):8.3f} ms")


if __name__ == "__main__":
    main()
    probe_decide_graph()
