"""Crossover probe: single-workgroup vs cooperative mega at several P.
Env MI355X_AS_SINGLE_MAX_P decides the path (read once per process)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.ops.engine_gpu import GpuEngine
from mi355x_autoscaler.sim.synth import burst_trace, wrap
from mi355x_autoscaler.snapshot.encode import encode_snapshot


def main():
    P = int(sys.argv[1])
    shape = sys.argv[2] if len(sys.argv) > 2 else "a"
    eng = GpuEngine("cuda:0")
    if shape == "a":  # config #3 shape: 100 nodes, 3 pools
        pods_json, nodes_json = burst_trace(
            P, n_nodes_per_pool=(40, 30, 30),
            pools=(("agentpool1", "Standard_D2_v2"),
                   ("bigpool", "Standard_D4_v2"),
                   ("gpupool", "Standard_ND96_MI355X_v1")),
            gpu_fraction=0.35, seed=0, bound_pods_per_node=3)
    else:  # config #2 shape: 8 nodes, 1 GPU pool
        pods_json, nodes_json = burst_trace(
            P, n_nodes_per_pool=(8,),
            pools=(("gpupool", "Standard_ND96_MI355X_v1"),),
            gpu_fraction=1.0, seed=0, bound_pods_per_node=3)
    pods, nodes = wrap(pods_json, nodes_json)
    snap = encode_snapshot(pods, nodes, group_nodes_into_pools(nodes))
    for _ in range(10):
        eng.decide(snap)
    torch.cuda.synchronize()
    ts = []
    for _ in range(60):
        t0 = time.perf_counter()
        eng.decide(snap)
        ts.append(time.perf_counter() - t0)
    ts.sort()
    mode = os.environ.get("MI355X_AS_SINGLE_MAX_P", "default(1024)")
    print(f"P={P} shape={shape} single_max_p={mode} p50={ts[30]*1e3:.3f} ms")


main()
