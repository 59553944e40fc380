# long churn soak (cycles from argv, default 120) on hardware: watch-mode loop + GPU engine with
# pod add/remove churn and periodic scale-up/down, asserting stability.
import random, sys, time
CYCLES = int(sys.argv[1]) if len(sys.argv) > 1 else 120
sys.path.insert(0, "/root/repo")
from mi355x_autoscaler.cloud.fake import FakeCloudProvider
from mi355x_autoscaler.ctl.loop import Cluster, ClusterOptions
from mi355x_autoscaler.kube.client import KubeApiClient
from mi355x_autoscaler.kube.fake_api import FakeKubeServer
from mi355x_autoscaler.ops.engine_gpu import GpuEngine
from mi355x_autoscaler.sim.engine import CpuEngine
from mi355x_autoscaler.sim.synth import burst_trace, make_pod_json

rng = random.Random(0)
with FakeKubeServer() as srv:
    pods, nodes = burst_trace(500, n_nodes_per_pool=(6,), seed=0, bound_pods_per_node=2)
    srv.state.load(pods, nodes)
    kube = KubeApiClient(srv.base_url)
    provider = FakeCloudProvider(initial_pool_sizes={"agentpool1": 6}, kube_state=srv.state)
    opts = ClusterOptions(sleep_seconds=0, idle_threshold_seconds=0, spare_agents=1,
                          max_drains_per_cycle=2, watch=True)
    gpu = GpuEngine("cuda:0")
    cpu_check = CpuEngine()
    cluster = Cluster(kube, provider, gpu, options=opts)
    added = 0
    t0 = time.time()
    for cycle in range(CYCLES):
        # churn: add a burst, delete some pods
        for _ in range(rng.randrange(0, 30)):
            srv.state.add_pod(make_pod_json(f"churn-{added}", cpu_milli=rng.choice([100, 300, 700]),
                                            mem_mib=rng.choice([64, 256, 1024]), pending=True))
            added += 1
        # occasionally delete random churned pods (exercise tombstones)
        if cycle % 7 == 3:
            with srv.state.lock:
                keys = [k for k in srv.state.pods if "churn-" in k][:20]
            for k in keys:
                ns, name = k.split("/", 1)
                srv.state.remove_pod(ns, name)
        time.sleep(0.03)
        res = cluster.loop_logic()
        # every 20 cycles cross-check the decision against the CPU oracle
        if cycle % 25 == 0 and res.snapshot is not None:
            d_cpu = cpu_check.decide(res.snapshot)
            assert d_cpu.units_needed == res.decision.units_needed, (
                cycle, d_cpu.units_needed, res.decision.units_needed)
    dt = time.time() - t0
    print(f"soak ok: {CYCLES} cycles, {added} churned pods, {dt:.1f}s "
          f"({dt/CYCLES*1000:.1f} ms/cycle), pools={provider.pool_sizes}, "
          f"scale_ups={sum(1 for e in provider.events if e.kind=='scale_up')}, "
          f"deletes={sum(1 for e in provider.events if e.kind=='delete_node')}")
