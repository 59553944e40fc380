import os, sys
sys.path.insert(0, "/root/repo")
import numpy as np
import torch
from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.sim.engine import CpuEngine
from mi355x_autoscaler.sim.synth import burst_trace, wrap
from mi355x_autoscaler.snapshot.encode import encode_snapshot
from mi355x_autoscaler.ops.engine_gpu import GpuEngine, build_segments_device

pods_json, nodes_json = burst_trace(
    1000, n_nodes_per_pool=(4,), pools=(("gpupool", "Standard_ND96_MI355X_v1"),),
    gpu_fraction=1.0, seed=0, bound_pods_per_node=3)
pods, nodes = wrap(pods_json, nodes_json)
pools = group_nodes_into_pools(nodes)
snap = encode_snapshot(pods, nodes, pools)
ref = CpuEngine().decide(snap)
print("CPU:", ref.units_needed, ref.pods_placed_existing, ref.pods_unfulfillable)

os.environ["MI355X_AS_NO_MEGA"] = "1"
e = GpuEngine("cuda:0")
for label in ("classic", "py"):
    e.force_py_pipeline = label == "py"
    d = e.decide(snap)
    print(f"{label}: units={d.units_needed} placed={d.pods_placed_existing} unful={d.pods_unfulfillable}")

# manual pipeline: segments + routing vs oracle
ds = e.packer.pack(snap, e.device)
segs = e.build_segments_hash(ds)
G = segs.G
print("GPU G:", G)
g_dev = torch.full((1,), G, dtype=torch.int32, device=e.device)
seg_pool = torch.full((G,), -9, dtype=torch.int32, device=e.device)
e.ops.seg_pool_route(segs.seg_sel_pairs.contiguous(), segs.seg_sel_count.contiguous(),
                     segs.seg_req.contiguous(), ds.pool_labels, ds.pool_label_count,
                     ds.pool_unit, g_dev, seg_pool)
sp = seg_pool.cpu().numpy()
from mi355x_autoscaler.sim.oracle import build_segments, OracleSimulator
osegs = build_segments(snap)
print("oracle G:", len(osegs))
sim = OracleSimulator(snap)
bad = 0
for i, s_ in enumerate(osegs):
    cands = sim.pool_candidates(s_)
    ok = cands[0] if cands else None
    gk = sp[i] if i < G else None
    if (ok if ok is not None else -1) != gk:
        if bad < 5:
            print(f"  seg {i}: oracle pool={ok} gpu={gk} selc={s_.sel_count} pairs={s_.sel_pairs[:4].tolist()} req={s_.req.tolist()}")
        bad += 1
print("routing mismatches:", bad)
print("sp all:", sp.tolist())
print("gpu seg_req:", segs.seg_req[:13].cpu().tolist())
print("orc seg_req:", [x.req.tolist() for x in osegs])
print("gpu seg_count:", segs.seg_count[:13].cpu().tolist())
print("orc count:", [x.count for x in osegs])
print("gpu pairs all:", segs.seg_sel_pairs[:13, :2].cpu().tolist())
print("gpu selc all:", segs.seg_sel_count[:13].cpu().tolist())
print("gpu seg_sel_count[:10]:", segs.seg_sel_count[:10].cpu().tolist())
print("gpu pairs[1][:6]:", segs.seg_sel_pairs[1][:6].cpu().tolist())
print("orc pairs[1][:6]:", osegs[1].sel_pairs[:6].tolist() if len(osegs) > 1 else None)
print("pool_label_count:", ds.pool_label_count.cpu().tolist())
print("pool_labels[0][:8]:", ds.pool_labels[0][:8].cpu().tolist())
