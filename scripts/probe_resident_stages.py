"""Where does the 1M resident cycle spend time beyond the fused decide?"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import dataclasses
import random

import torch

from mi355x_autoscaler.ops.engine_gpu import GpuEngine
from mi355x_autoscaler.sim.synth import make_pod_json
from mi355x_autoscaler.sim.synth_soa import synth_burst_snapshot
from mi355x_autoscaler.snapshot.delta import DeltaSnapshotCache


def ms(fn, n=30):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


def main():
    P = 1_000_000
    snap = synth_burst_snapshot(P, n_nodes_per_pool=(400, 300, 300),
                                gpu_fraction=0.35, seed=3)
    snap.pool_max[:] = 1 << 20
    snap.pod_uids = [f"p{i}" for i in range(snap.num_pods)]
    snap.pod_refs = [("default", f"p{i}") for i in range(snap.num_pods)]
    base = dataclasses.replace(
        snap, pod_req=snap.pod_req[:0], pod_flags=snap.pod_flags[:0],
        pod_sel_hash=snap.pod_sel_hash[:0], pod_sel_pairs=snap.pod_sel_pairs[:0],
        pod_sel_count=snap.pod_sel_count[:0], pod_node=snap.pod_node[:0],
        pod_pdb=snap.pod_pdb[:0], pod_uids=[], pod_refs=[])
    cache = DeltaSnapshotCache()
    cache.prime(snap, {}, [])
    eng = GpuEngine("cuda:0")
    rng = random.Random(0)
    d = eng.decide_resident(cache, base)
    torch.cuda.synchronize()

    def churn():
        for e in range(2000):
            cache.apply_event({"type": "ADDED", "object": make_pod_json(
                f"c-{churn.i}-{e}", cpu_milli=rng.choice([100, 500]),
                pending=True)})
        churn.i += 1
    churn.i = 0

    print(f"decide_resident (no churn): {ms(lambda: eng.decide_resident(cache, base)):7.3f} ms")
    print(f"resident_ds (no churn)    : {ms(lambda: eng.resident_ds(cache, base)):7.3f} ms")
    churn()
    rows = cache.drain_dirty()
    print(f"drain_dirty rows after churn: {rows.size}")
    cache.dirty.update(rows.tolist())

    def scatter_only():
        cache.dirty.update(rows.tolist())
        eng._scatter_rows(eng._resident["pod_views"], cache, cache.drain_dirty())
    print(f"scatter 2000 rows         : {ms(scatter_only):7.3f} ms")
    rest = eng._rest_packer
    print(f"base pack_views           : {ms(lambda: rest.pack_views(base, eng.device)):7.3f} ms")
    ds = eng.resident_ds(cache, base)
    print(f"fused from resident ds    : {ms(lambda: eng._decide_fused(ds, base.pool_names, stable_pods=True)):7.3f} ms")


if __name__ == "__main__":
    main()
