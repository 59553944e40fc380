"""End-to-end product path on hardware: the control loop against the
in-process fake k8s API + fake cloud with the GPU engine — scale-up
materializes nodes, the next cycle converges, scale-down drains."""

import pytest

pytestmark = pytest.mark.gpu

torch = pytest.importorskip("torch")

from mi355x_autoscaler.cloud.fake import FakeCloudProvider
from mi355x_autoscaler.ctl.loop import Cluster, ClusterOptions
from mi355x_autoscaler.kube.client import KubeApiClient
from mi355x_autoscaler.kube.fake_api import FakeKubeServer
from mi355x_autoscaler.sim.synth import burst_trace, make_node_json, make_pod_json


@pytest.fixture(scope="module")
def server():
    with FakeKubeServer() as srv:
        yield srv


@pytest.fixture(scope="module")
def gpu_engine():
    from mi355x_autoscaler.ops.engine_gpu import GpuEngine

    return GpuEngine("cuda:0")


def _cluster(server, engine, **opt_kw):
    kube = KubeApiClient(server.base_url)
    provider = FakeCloudProvider(
        initial_pool_sizes={"agentpool1": 3}, kube_state=server.state
    )
    opts = ClusterOptions(sleep_seconds=0, idle_threshold_seconds=0,
                          spare_agents=0, **opt_kw)
    return Cluster(kube, provider, engine, options=opts), provider


def test_gpu_loop_scales_up_and_converges(server, gpu_engine):
    pods, nodes = burst_trace(200, n_nodes_per_pool=(3,), seed=0)
    server.state.load(pods, nodes)
    cluster, provider = _cluster(server, gpu_engine, no_maintenance=True)

    res = cluster.loop_logic()
    assert res.pods_pending == 200
    assert res.decision.wants_scale_up()
    assert provider.pool_sizes["agentpool1"] > 3

    res2 = cluster.loop_logic()
    assert res2.decision.pods_placed_existing == 200
    assert not res2.decision.wants_scale_up()


def test_gpu_loop_scales_down(server, gpu_engine):
    nodes = [make_node_json("agentpool1", i) for i in range(4)]
    pods = [make_pod_json("w", cpu_milli=1500, node_name=nodes[0]["metadata"]["name"])]
    server.state.load(pods, nodes)
    cluster, provider = _cluster(server, gpu_engine, no_scale=True,
                                 max_drains_per_cycle=2)
    res = cluster.loop_logic()
    assert len(res.drained_nodes) == 2
    for name in res.drained_nodes:
        assert name in server.state.deleted_nodes


def test_gpu_loop_watch_mode(server, gpu_engine):
    import time

    pods, nodes = burst_trace(50, n_nodes_per_pool=(3,), seed=2)
    server.state.load(pods, nodes)
    cluster, _ = _cluster(server, gpu_engine, no_maintenance=True,
                          no_scale=True, watch=True, native_codec=True)
    res1 = cluster.loop_logic()
    assert res1.pods_pending == 50
    for i in range(9):
        server.state.add_pod(make_pod_json(f"w{i}", cpu_milli=50, pending=True))
    time.sleep(0.5)
    res2 = cluster.loop_logic()
    assert res2.pods_pending == 59
