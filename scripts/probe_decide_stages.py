"""Stage-level timing of the GPU decide: pack / fused launch / segsum /
readback / host finish — drives the next latency target."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.ops.engine_gpu import GpuEngine, segsum
from mi355x_autoscaler.sim.synth import burst_trace, wrap
from mi355x_autoscaler.snapshot.encode import encode_snapshot


def ms(fn, n=100):
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
    for _ in range(3):
        eng.decide(snap)

    print(f"full decide        : {ms(lambda: eng.decide(snap)):7.3f} ms")
    print(f"pack (H2D in)      : {ms(lambda: eng.packer.pack(snap, eng.device)):7.3f} ms")

    ds = eng.packer.pack(snap, eng.device)
    K = ds.K
    print(f"decide from ds     : {ms(lambda: eng._decide_fused(ds, snap.pool_names)):7.3f} ms")

    # segsum alone (the MFMA demand path)
    pod_pool = eng._decide_buffers(ds.P, K, True)["pod_pool"]
    print(f"segsum (sync)      : {ms(lambda: (segsum(ds.pod_req, pod_pool, K, eng.ops), torch.cuda.synchronize())):7.3f} ms")

    # readback alone
    bufs = eng._decide_buffers(ds.P, K, True)
    blob, blob_host = bufs["blob"], bufs["blob_host"]
    print(f"blob pinned D2H    : {ms(lambda: blob_host.copy_(blob)):7.3f} ms")
    torch.cuda.synchronize()
    t = ms(lambda: None, 1000)
    print(f"timer overhead     : {t:7.3f} ms")

    # gpu1k shape (BASELINE config #2): 1k pods, 1 pool, 8 nodes — why
    # is the tiny burst SLOWER per step than the 10k one?
    pods_json, nodes_json = burst_trace(
        1000, n_nodes_per_pool=(8,),
        pools=(("gpupool", "Standard_ND96_MI355X_v1"),),
        gpu_fraction=1.0, seed=0, bound_pods_per_node=2)
    pods, nodes = wrap(pods_json, nodes_json)
    snap1k = encode_snapshot(pods, nodes, group_nodes_into_pools(nodes))
    for _ in range(3):
        eng.decide(snap1k)
    print(f"[gpu1k] full decide: {ms(lambda: eng.decide(snap1k)):7.3f} ms")
    print(f"[gpu1k] pack       : {ms(lambda: eng.packer.pack(snap1k, eng.device)):7.3f} ms")
    ds1 = eng.packer.pack(snap1k, eng.device)
    print(f"[gpu1k] from ds    : {ms(lambda: eng._decide_fused(ds1, snap1k.pool_names)):7.3f} ms")


if __name__ == "__main__":
    main()
