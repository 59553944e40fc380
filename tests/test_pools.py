from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.models.capacity import get_capacity
from mi355x_autoscaler.models.kube import KubeNode
from mi355x_autoscaler.sim.synth import make_node_json


def _nodes(pool, n, **kw):
    return [KubeNode(make_node_json(pool, i, **kw)) for i in range(n)]


def test_group_nodes_into_pools():
    nodes = _nodes("agentpool1", 3) + _nodes("gpupool", 2, instance_type="Standard_ND96_MI355X_v1", gpu=8)
    pools = group_nodes_into_pools(nodes)
    assert set(pools) == {"agentpool1", "gpupool"}
    assert pools["agentpool1"].actual_capacity == 3
    assert pools["gpupool"].instance_type == "Standard_ND96_MI355X_v1"


def test_ignore_pools():
    nodes = _nodes("agentpool1", 2) + _nodes("noscale", 1)
    pools = group_nodes_into_pools(nodes, ignore_pools=["noscale"])
    assert set(pools) == {"agentpool1"}


def test_unit_capacity_prefers_live_allocatable():
    nodes = _nodes("agentpool1", 1, cpu_milli=1900, mem_mib=6000)
    pools = group_nodes_into_pools(nodes)
    unit = pools["agentpool1"].unit_capacity
    assert unit.cpu == 1900  # live allocatable, not the SKU table


def test_capacity_table_mi355x_rows():
    cap = get_capacity("Standard_ND96_MI355X_v1")
    assert cap is not None and cap.gpu == 8 and cap.cpu == 96000
    assert get_capacity("Standard_D2_v2").cpu == 2000
    assert get_capacity("NoSuch_SKU") is None


def test_selector_matches_prototype_labels():
    nodes = _nodes("gpupool", 1, instance_type="Standard_ND96_MI355X_v1", gpu=8)
    pools = group_nodes_into_pools(nodes)
    assert pools["gpupool"].selector_matches({"accelerator": "amd-mi355x"})
    assert not pools["gpupool"].selector_matches({"accelerator": "tpu"})
