"""Steady-state control-loop cycle floor: watch caches warm, no churn.
The reference's cycle floor is a full pod+node relist (seconds at 10k
pods); ours is queue drains + cached node/PDB sets + decide+maintain."""
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from mi355x_autoscaler.cloud.fake import FakeCloudProvider
from mi355x_autoscaler.ctl.loop import Cluster, ClusterOptions
from mi355x_autoscaler.kube.client import KubeApiClient
from mi355x_autoscaler.kube.fake_api import FakeKubeServer
from mi355x_autoscaler.sim.engine import CpuEngine
from mi355x_autoscaler.sim.synth import burst_trace
from mi355x_autoscaler.utils.notification import Notifier


def main():
    pods, nodes = burst_trace(
        10000, n_nodes_per_pool=(40, 30, 30),
        pools=(("agentpool1", "Standard_D2_v2"), ("bigpool", "Standard_D4_v2"),
               ("gpupool", "Standard_ND96_MI355X_v1")),
        gpu_fraction=0.35, seed=0, bound_pods_per_node=3)
    with FakeKubeServer() as srv:
        srv.state.load(pods, nodes)
        kube = KubeApiClient(srv.base_url)
        if torch.cuda.is_available():
            from mi355x_autoscaler.ops.engine_gpu import GpuEngine

            engine = GpuEngine("cuda:0")
        else:
            engine = CpuEngine()
        opts = ClusterOptions(sleep_seconds=0, no_scale=True,
                              idle_threshold_seconds=10**9, watch=True)
        cluster = Cluster(kube, FakeCloudProvider(kube_state=srv.state),
                          engine, options=opts, notifier=Notifier(dry_run=True))
        for _ in range(5):
            cluster.loop_logic()
        ts = []
        for _ in range(50):
            t0 = time.perf_counter()
            res = cluster.loop_logic()
            ts.append(time.perf_counter() - t0)
        p50 = statistics.median(ts) * 1e3
        p99 = sorted(ts)[int(0.98 * len(ts))] * 1e3
        print(f"steady-state cycle (10k pods resident, watch, "
              f"{engine.__class__.__name__}): p50 {p50:.3f} ms "
              f"p99 {p99:.3f} ms; phases "
              f"{ {k: round(v * 1e3, 3) for k, v in res.phase_times.items()} }")


if __name__ == "__main__":
    main()
