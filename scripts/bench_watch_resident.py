"""Resident-controller measurement at 1M pods: watch-event churn into
the HBM-resident snapshot (kernel E scatter) + per-cycle decisions."""
import random, statistics, sys, time

sys.path.insert(0, "/root/repo")
import torch

from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.models.kube import KubeNode
from mi355x_autoscaler.ops.engine_gpu import GpuEngine
from mi355x_autoscaler.sim.engine import CpuEngine
from mi355x_autoscaler.sim.synth import make_node_json, make_pod_json
from mi355x_autoscaler.sim.synth_soa import synth_burst_snapshot
from mi355x_autoscaler.snapshot.delta import DeltaSnapshotCache
from mi355x_autoscaler.snapshot.encode import encode_snapshot

P = 1_000_000
snap = synth_burst_snapshot(P, n_nodes_per_pool=(400, 300, 300), gpu_fraction=0.35, seed=3)
snap.pool_max[:] = 1 << 20
snap.pod_uids = [f"p{i}" for i in range(snap.num_pods)]
snap.pod_refs = [("default", f"p{i}") for i in range(snap.num_pods)]

nodes_json = snap  # node side re-encode base:
nodes = [KubeNode(make_node_json("agentpool1", i)) for i in range(0)]
# base = node/pool side only; reuse from the synthetic snapshot itself
import dataclasses
base = dataclasses.replace(snap, pod_req=snap.pod_req[:0], pod_flags=snap.pod_flags[:0],
                           pod_sel_hash=snap.pod_sel_hash[:0], pod_sel_pairs=snap.pod_sel_pairs[:0],
                           pod_sel_count=snap.pod_sel_count[:0], pod_node=snap.pod_node[:0],
                           pod_pdb=snap.pod_pdb[:0], pod_uids=[], pod_refs=[])

cache = DeltaSnapshotCache()
cache.prime(snap, {}, [])
gpu = GpuEngine("cuda:0")

rng = random.Random(0)
t0 = time.perf_counter()
d = gpu.decide_resident(cache, base)
torch.cuda.synchronize()
print(f"prime+first decide: {(time.perf_counter()-t0)*1000:.1f} ms; units={sum(d.units_needed.values())}")

apply_ts, batch_ts, decide_ts = [], [], []
EVENTS = 2000
import json as _json

import mi355x_autoscaler._hip_ops as _ops

CYCLES = int(__import__("os").environ.get("CYCLES", "15"))
for cycle in range(CYCLES):
    pods = [make_pod_json(f"churn-{cycle}-{e}", cpu_milli=rng.choice([100, 500]),
                          mem_mib=rng.choice([128, 1024]), pending=True)
            for e in range(EVENTS)]
    if cycle % 2 == 0:
        # per-event Python reference path
        t0 = time.perf_counter()
        for pod in pods:
            cache.apply_event({"type": "ADDED", "object": pod})
        apply_ts.append(time.perf_counter() - t0)
    else:
        # production path: raw JSONL -> threaded C++ decode -> C row
        # resolution -> vectorized column writes
        blob = ("\n".join(
            _json.dumps({"type": "ADDED", "object": p}) for p in pods
        ) + "\n").encode()
        t0 = time.perf_counter()
        decoded = _ops.decode_watch_events(blob, cache.node_index,
                                           list(cache.pdb_selectors))
        cache.apply_batch(decoded)
        batch_ts.append(time.perf_counter() - t0)
    t0 = time.perf_counter()
    d = gpu.decide_resident(cache, base)
    torch.cuda.synchronize()
    decide_ts.append(time.perf_counter() - t0)

print(f"event apply: {EVENTS/statistics.median(apply_ts):.0f} events/s per-event python; "
      f"{EVENTS/statistics.median(batch_ts):.0f} events/s batch (C++ decode + C resolve)")
print(f"resident decide @1M+churn: p50 {statistics.median(decide_ts)*1000:.3f} ms "
      f"(min {min(decide_ts)*1000:.3f}, max {max(decide_ts)*1000:.3f})")
d_cpu = CpuEngine().decide(cache.build_snapshot(base))
assert d.units_needed == d_cpu.units_needed, (d.units_needed, d_cpu.units_needed)
print("oracle cross-check ok")
