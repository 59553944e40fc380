"""Hypothesis property tests: the oracle's scan-step FFD
vs the naive per-pod reference on randomized traces, quantity-parsing
round trips, and pack-key order preservation."""

import numpy as np
from hypothesis import given, settings, strategies as st

from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.models.resources import parse_cpu_milli, parse_quantity
from mi355x_autoscaler.sim.oracle import OracleSimulator, _pack_req_key, pack_req_keys
from mi355x_autoscaler.sim.synth import make_node_json, make_pod_json, wrap
from mi355x_autoscaler.snapshot.encode import encode_snapshot

from naive_reference import naive_fulfill


@st.composite
def cluster_case(draw):
    n_nodes = draw(st.integers(1, 6))
    nodes = []
    for i in range(n_nodes):
        nodes.append(make_node_json(
            "agentpool1", i,
            cpu_milli=draw(st.sampled_from([1000, 2000, 4000])),
            mem_mib=draw(st.sampled_from([1024, 4096, 7168])),
            pods=draw(st.sampled_from([5, 20, 110])),
        ))
    n_shapes = draw(st.integers(1, 5))
    shapes = [
        (draw(st.integers(0, 2500)), draw(st.integers(0, 2048)))
        for _ in range(n_shapes)
    ]
    pods = []
    n_pods = draw(st.integers(0, 60))
    for i in range(n_pods):
        cpu, mem = shapes[draw(st.integers(0, n_shapes - 1))]
        pods.append(make_pod_json(f"p{i}", cpu_milli=cpu, mem_mib=mem, pending=True))
    max_size = draw(st.integers(n_nodes, n_nodes + 10))
    return pods, nodes, max_size


@settings(max_examples=60, deadline=None)
@given(cluster_case())
def test_oracle_equals_naive_first_fit(case):
    pods_json, nodes_json, max_size = case
    pods, nodes = wrap(pods_json, nodes_json)
    pools = group_nodes_into_pools(nodes, max_size=max_size)
    snap = encode_snapshot(pods, nodes, pools)
    d = OracleSimulator(snap).fulfill_pending()
    naive = naive_fulfill(snap)
    assert d.pods_placed_existing == naive["placed_existing"]
    assert d.pods_placed_new == naive["placed_new"]
    assert d.pods_unfulfillable == naive["unfulfillable"]
    assert d.units_needed == naive["units_needed"]


@settings(max_examples=200, deadline=None)
@given(st.integers(0, 10**7), st.sampled_from(["m", "", "k"]))
def test_cpu_quantity_roundtrip(n, suffix):
    s = f"{n}{suffix}"
    v = parse_quantity(s)
    assert v >= 0
    milli = parse_cpu_milli(s)
    assert milli == round(v * 1000)


@settings(max_examples=200, deadline=None)
@given(
    st.tuples(st.integers(0, 2**21 - 1), st.integers(0, 2**23 - 1),
              st.just(1), st.integers(0, 255)),
    st.tuples(st.integers(0, 2**21 - 1), st.integers(0, 2**23 - 1),
              st.just(1), st.integers(0, 255)),
)
def test_pack_key_order_preserving(a, b):
    """pack_req_key must order exactly like the (cpu, mem, gpu) tuple
    within its clamp range — the FFD 'decreasing' order depends on it."""
    ka = _pack_req_key(np.array(a))
    kb = _pack_req_key(np.array(b))
    ta = (a[0], a[1], a[3])
    tb = (b[0], b[1], b[3])
    assert (ka > kb) == (ta > tb)
    assert (ka == kb) == (ta == tb)
    # vectorized form agrees with the scalar form
    arr = np.array([a, b], np.int64)
    ks = pack_req_keys(arr)
    assert int(ks[0]) == ka and int(ks[1]) == kb


@st.composite
def heterogeneous_case(draw):
    """Wider shapes: two pools (CPU + MI355X), selectors that match one
    pool / neither, gpu requests, criticals, bound pods, zero requests."""
    n_cpu = draw(st.integers(1, 4))
    n_gpu = draw(st.integers(0, 3))
    nodes = [make_node_json("agentpool1", i,
                            cpu_milli=draw(st.sampled_from([2000, 4000])))
             for i in range(n_cpu)]
    nodes += [make_node_json("gpupool", i, instance_type="Standard_ND96_MI355X_v1",
                             cpu_milli=96000, mem_mib=65536, gpu=8)
              for i in range(n_gpu)]
    sel_menu = [
        None,
        {"accelerator": "amd-mi355x"},            # matches gpupool only
        {"agentpool": "agentpool1"},              # matches the CPU pool
        {"nosuch": "label"},                      # matches nothing
    ]
    pods = []
    for i in range(draw(st.integers(0, 50))):
        gpu = draw(st.sampled_from([0, 0, 0, 1, 2]))
        pods.append(make_pod_json(
            f"p{i}",
            cpu_milli=draw(st.sampled_from([0, 100, 1500, 3000])),
            mem_mib=draw(st.sampled_from([0, 256, 2048])),
            gpu=gpu,
            pending=True,
            node_selector=draw(st.sampled_from(sel_menu)),
            annotations=(
                {"scheduler.alpha.kubernetes.io/critical-pod": "true"}
                if draw(st.booleans()) else None),
        ))
    # a few bound pods consume node capacity
    for b in range(draw(st.integers(0, 4))):
        j = draw(st.integers(0, len(nodes) - 1))
        pods.append(make_pod_json(
            f"b{b}", cpu_milli=draw(st.sampled_from([200, 900])),
            node_name=nodes[j]["metadata"]["name"]))
    max_size = draw(st.integers(1, 8))
    return pods, nodes, max_size


@settings(max_examples=60, deadline=None)
@given(heterogeneous_case())
def test_oracle_equals_naive_heterogeneous(case):
    pods_json, nodes_json, max_size = case
    pods, nodes = wrap(pods_json, nodes_json)
    pools = group_nodes_into_pools(nodes, max_size=max_size)
    snap = encode_snapshot(pods, nodes, pools)
    d = OracleSimulator(snap).fulfill_pending()
    naive = naive_fulfill(snap)
    assert d.pods_placed_existing == naive["placed_existing"]
    assert d.pods_placed_new == naive["placed_new"]
    assert d.pods_unfulfillable == naive["unfulfillable"]
    assert d.units_needed == naive["units_needed"]
