import sys, time, statistics
sys.path.insert(0, "/root/repo")
import torch
from mi355x_autoscaler.ops.engine_gpu import GpuEngine
from mi355x_autoscaler.sim.synth_soa import synth_burst_snapshot

P = 10_000_000
t0=time.perf_counter()
snap = synth_burst_snapshot(P, n_nodes_per_pool=(4000, 3000, 3000), gpu_fraction=0.35, seed=1)
snap.pool_max[:] = 1 << 24
print(f"synth {P} pods: {time.perf_counter()-t0:.1f}s, SoA {snap.nbytes()/1e9:.2f} GB")
gpu = GpuEngine("cuda:0")
ds = gpu.packer.pack(snap, gpu.device)
torch.cuda.synchronize()
d0 = None
ts=[]
for i in range(8):
    t0=time.perf_counter()
    segs = gpu.build_segments(ds)
    d = gpu.decide_device(ds, segs, snap.pool_names)
    torch.cuda.synchronize()
    ts.append(time.perf_counter()-t0)
    if d0 is None:
        d0 = d
    else:
        assert d.units_needed == d0.units_needed  # repeat determinism
total = d.pods_placed_existing + d.pods_placed_new + d.pods_unfulfillable
assert total == P, (total, P)
print(f"10M-pod resident decision: p50 {statistics.median(ts)*1000:.2f} ms "
      f"(units={sum(d.units_needed.values())}, placed={d.pods_placed_existing})")
free, tot = torch.cuda.mem_get_info()
print(f"HBM in use: {(tot-free)/1e9:.1f} GB of {tot/1e9:.0f} GB")
