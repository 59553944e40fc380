from mi355x_autoscaler.models.kube import KubeNode, KubePod
from mi355x_autoscaler.sim.synth import make_node_json, make_pod_json


def test_pending_classification():
    p = KubePod(make_pod_json("a", pending=True))
    assert p.is_pending_unschedulable
    q = KubePod(make_pod_json("b", pending=False, node_name="k8s-agentpool1-12345678-0"))
    assert not q.is_pending_unschedulable


def test_pod_request_summing_multi_container():
    obj = make_pod_json("multi", cpu_milli=100, mem_mib=128)
    obj["spec"]["containers"].append(
        {"name": "side", "resources": {"requests": {"cpu": "50m", "memory": "64Mi"}}}
    )
    p = KubePod(obj)
    assert p.resources.cpu == 150
    assert p.resources.memory == (128 + 64) * 1024**2
    assert p.resources.pods == 1


def test_drainability_flags():
    rs = KubePod(make_pod_json("rs", owner_kind="ReplicaSet"))
    assert rs.is_replicated and rs.is_drainable and not rs.blocks_drain

    bare = KubePod(make_pod_json("bare", owner_kind=None))
    assert not bare.is_replicated and bare.blocks_drain

    ds = KubePod(make_pod_json("ds", owner_kind="DaemonSet"))
    assert ds.is_daemonset and not ds.blocks_drain

    mirror = KubePod(
        make_pod_json("m", owner_kind=None, annotations={"kubernetes.io/config.mirror": "x"})
    )
    assert mirror.is_mirrored and not mirror.blocks_drain

    pinned = KubePod(
        make_pod_json(
            "pin",
            owner_kind="ReplicaSet",
            annotations={"cluster-autoscaler.kubernetes.io/safe-to-evict": "false"},
        )
    )
    assert not pinned.is_drainable and pinned.blocks_drain


def test_node_pool_parse_and_provider_id():
    n = KubeNode(make_node_json("agentpool1", 3))
    assert n.pool_name == "agentpool1"
    assert n.pool_index == 3
    assert n.instance_id == "k8s-agentpool1-12345678-3"
    assert n.instance_type == "Standard_D2_v2"
    assert n.is_ready
    master = KubeNode({"metadata": {"name": "k8s-master-12345678-0x"}, "spec": {}, "status": {}})
    assert master.pool_name is None


def test_node_actuation_verbs_against_fake_api():
    """KubeNode.cordon/uncordon/drain/delete and KubePod.delete — the
    reference's verb surface — against the in-process API server."""
    from mi355x_autoscaler.kube.client import KubeApiClient
    from mi355x_autoscaler.kube.fake_api import FakeKubeServer
    from mi355x_autoscaler.sim.synth import make_node_json, make_pod_json

    with FakeKubeServer() as srv:
        nj = make_node_json("agentpool1", 0)
        name = nj["metadata"]["name"]
        pods = [
            make_pod_json("w1", cpu_milli=100, node_name=name),
            make_pod_json("ds", cpu_milli=50, node_name=name, owner_kind="DaemonSet"),
        ]
        srv.state.load(pods, [nj])
        kube = KubeApiClient(srv.base_url)

        node = KubeNode(nj)
        node.cordon(kube)
        assert srv.state.nodes[name]["spec"]["unschedulable"] is True
        assert node.unschedulable
        node.uncordon(kube)
        assert not srv.state.nodes[name]["spec"].get("unschedulable")

        wrapped = [KubePod(p) for p in pods]
        n = node.drain(kube, wrapped)
        assert n == 1  # daemonset pod skipped, like kubectl drain
        assert "default/w1" not in srv.state.pods

        node.delete(kube)
        assert name not in srv.state.nodes
