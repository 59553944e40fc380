"""Maintain (drain-candidate scoring) at 5000 nodes — scale check for
the scale-down path beyond BASELINE config #4's 500 nodes."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.ops.engine_gpu import GpuEngine
from mi355x_autoscaler.sim.engine import CpuEngine
from mi355x_autoscaler.sim.synth import scaledown_trace, wrap
from mi355x_autoscaler.snapshot.encode import encode_snapshot

n = int(sys.argv[1]) if len(sys.argv) > 1 else 5000
pods_json, nodes_json, pdbs = scaledown_trace(n_nodes=n, seed=7)
pods, nodes = wrap(pods_json, nodes_json)
snap = encode_snapshot(pods, nodes, group_nodes_into_pools(nodes, max_size=100000), pdbs)
eng = GpuEngine("cuda:0")
for _ in range(5):
    eng.maintain(snap)
torch.cuda.synchronize()
ts = []
for _ in range(30):
    t0 = time.perf_counter()
    rep = eng.maintain(snap)
    ts.append(time.perf_counter() - t0)
ts.sort()
rep_cpu = CpuEngine().maintain(snap)
import numpy as np
np.testing.assert_array_equal(rep.node_util_permille, rep_cpu.node_util_permille)
np.testing.assert_array_equal(rep.node_pdb_ok, rep_cpu.node_pdb_ok)
assert [c.node_name for c in rep.candidates] == [c.node_name for c in rep_cpu.candidates]
print(f"maintain n={n}: p50={ts[15]*1e3:.3f} ms (min {ts[0]*1e3:.3f}), "
      f"{snap.num_pods} pods, candidates={len(rep_cpu.candidates)}, oracle-equal ok")
