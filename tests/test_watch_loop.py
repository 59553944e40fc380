"""End-to-end watch-mode control loop against the fake API server:
delta ingest, stream-drop recovery, node-churn re-prime."""

import time

import pytest

from mi355x_autoscaler.cloud.fake import FakeCloudProvider
from mi355x_autoscaler.ctl.loop import Cluster, ClusterOptions
from mi355x_autoscaler.kube.client import KubeApiClient
from mi355x_autoscaler.kube.fake_api import FakeKubeServer
from mi355x_autoscaler.sim.engine import CpuEngine
from mi355x_autoscaler.sim.synth import burst_trace, make_node_json, make_pod_json


@pytest.fixture()
def server():
    with FakeKubeServer() as srv:
        yield srv


def _cluster(server, **opt_kw):
    kube = KubeApiClient(server.base_url)
    provider = FakeCloudProvider(initial_pool_sizes={"agentpool1": 3}, kube_state=server.state)
    opts = ClusterOptions(sleep_seconds=0, idle_threshold_seconds=0, spare_agents=0,
                          watch=True, native_codec=False, **opt_kw)
    return Cluster(kube, provider, CpuEngine(), options=opts), provider


def _wait_events(cluster, min_applied=1, timeout=5.0):
    """Poll until the watch pump delivered something (thread timing)."""
    ws = cluster._watch_state
    deadline = time.time() + timeout
    while time.time() < deadline:
        if ws and not ws["q"].empty():
            return
        time.sleep(0.05)


def test_watch_mode_sees_added_pods(server):
    pods, nodes = burst_trace(5, n_nodes_per_pool=(3,), seed=0)
    server.state.load(pods, nodes)
    cluster, provider = _cluster(server, no_maintenance=True, no_scale=True)

    res1 = cluster.loop_logic()  # primes the cache + starts the pump
    assert res1.pods_total == 5
    assert cluster._watch_state is not None

    # pods arrive via watch only (no relist)
    for i in range(7):
        server.state.add_pod(make_pod_json(f"late{i}", cpu_milli=100, pending=True))
    _wait_events(cluster)
    time.sleep(0.2)
    res2 = cluster.loop_logic()
    assert res2.pods_pending == 12

    # deletion shrinks the pending set
    server.state.remove_pod("default", "late0")
    time.sleep(0.3)
    res3 = cluster.loop_logic()
    assert res3.pods_pending == 11


def test_watch_stream_drop_reprimes(server):
    pods, nodes = burst_trace(3, n_nodes_per_pool=(3,), seed=1)
    server.state.load(pods, nodes)
    server.state.drop_watch_after = 2  # cut the stream after 2 events
    cluster, _ = _cluster(server, no_maintenance=True, no_scale=True)
    cluster.loop_logic()
    for i in range(5):
        server.state.add_pod(make_pod_json(f"x{i}", cpu_milli=50, pending=True))
    time.sleep(0.5)  # stream delivers 2 then dies -> sentinel queued
    cluster.loop_logic()          # observes the break, marks broken
    server.state.drop_watch_after = -1
    res = cluster.loop_logic()    # re-primes from a full relist
    assert res.pods_pending == 8  # nothing lost despite the drop


def test_watch_node_churn_reprimes(server):
    pods, nodes = burst_trace(4, n_nodes_per_pool=(3,), seed=2)
    server.state.load(pods, nodes)
    cluster, _ = _cluster(server, no_maintenance=True, no_scale=True)
    cluster.loop_logic()
    first_state = cluster._watch_state
    # add a node out of band: indices shift -> cache must re-prime
    server.state.nodes["k8s-agentpool1-12345678-9"] = make_node_json("agentpool1", 9)
    res = cluster.loop_logic()
    assert cluster._watch_state is not first_state
    assert res.nodes_total == 4
    assert res.pods_pending == 4
