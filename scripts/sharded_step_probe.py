import sys, time, statistics
sys.path.insert(0, "/root/repo")
import torch
from bench import build_snapshot
from mi355x_autoscaler.ops.engine_gpu import GpuEngine
from mi355x_autoscaler.parallel.engine_sharded import ShardedEngine
snap = build_snapshot(0, 10000)
inner = GpuEngine("cuda:0")
eng = ShardedEngine(inner, check_agreement=False)  # world 1: collectives no-op
for _ in range(5): eng.decide(snap)
torch.cuda.synchronize()
ts=[]
for _ in range(40):
    t0=time.perf_counter(); eng.decide(snap); torch.cuda.synchronize()
    ts.append(time.perf_counter()-t0)
print(f"sharded world-1 step (per-rank cost minus collectives): p50 {statistics.median(ts)*1000:.3f} ms")
for _ in range(5): inner.decide(snap)
ts=[]
for _ in range(40):
    t0=time.perf_counter(); inner.decide(snap); torch.cuda.synchronize()
    ts.append(time.perf_counter()-t0)
print(f"plain engine step: p50 {statistics.median(ts)*1000:.3f} ms")
