import os, sys
sys.path.insert(0, "/root/repo")
import torch
from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.sim.engine import CpuEngine
from mi355x_autoscaler.sim.synth import make_node_json, make_pod_json, wrap
from mi355x_autoscaler.snapshot.encode import encode_snapshot
from mi355x_autoscaler.ops.engine_gpu import GpuEngine

sel = {f"k{i:02d}": f"v{i}" for i in range(12)}
full = make_node_json("agentpool1", 0)
full["metadata"]["labels"].update(sel)
partial = make_node_json("agentpool1", 1)
partial["metadata"]["labels"].update({k: sel[k] for k in list(sel)[:11]})
pods = [make_pod_json(f"w{i}", cpu_milli=100, pending=True, node_selector=sel)
        for i in range(5)]
pods += [make_pod_json("plain", cpu_milli=100, pending=True)]
p, n = wrap(pods, [full, partial])
pools = group_nodes_into_pools(n)
snap = encode_snapshot(p, n, pools)
ref = CpuEngine().decide(snap)
print("CPU:", ref.units_needed, ref.pods_placed_existing, ref.pods_unfulfillable)
print("node_label_count:", snap.node_label_count.tolist())
print("pod_sel_count:", snap.pod_sel_count.tolist())

eng = GpuEngine("cuda:0")
for label, (mega, py) in [("mega", (True, False)), ("classic", (False, False)),
                          ("py", (False, True))]:
    os.environ["MI355X_AS_NO_MEGA"] = "0" if mega else "1"
    # env read once in C++ static; emulate by fresh processes is costly —
    # instead classic/py via force flags; mega only in first pass
    eng.force_py_pipeline = py
    d = eng.decide(snap)
    print(f"{label}: units={d.units_needed} placed={d.pods_placed_existing} unful={d.pods_unfulfillable}")

# eligibility dump
ds = eng.packer.pack(snap, eng.device)
segs = eng.build_segments_hash(ds)
G = segs.G
g_dev = torch.full((1,), G, dtype=torch.int32, device=eng.device)
W = (ds.N + 63) // 64
elig = torch.zeros((max(G,1), max(W,1)), dtype=torch.int64, device=eng.device)
eng.ops.fit_eligibility(ds.node_labels, ds.node_label_count, ds.node_unsched,
                        segs.seg_sel_pairs.contiguous(), segs.seg_sel_count.contiguous(),
                        g_dev, elig)
print("G:", G, "selc:", segs.seg_sel_count[:G].cpu().tolist())
print("elig bits:", [bin(int(x)) for x in elig[:G, 0].cpu().tolist()])
print("gpu pairs:", segs.seg_sel_pairs[:G, :13].cpu().tolist())
print("node labels[0][:20]:", ds.node_labels[0][:20].cpu().tolist())
