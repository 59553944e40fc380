"""Differential tests: the C++ snapshot codec must reproduce the Python
encoder's pod columns byte-for-byte on every trace it can see."""

import json

import numpy as np
import pytest

from mi355x_autoscaler.kube.objects import PodDisruptionBudget
from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.snapshot.codec import codec_available, encode_snapshot_fast
from mi355x_autoscaler.snapshot.encode import encode_snapshot
from mi355x_autoscaler.sim.synth import burst_trace, make_node_json, make_pod_json, scaledown_trace, wrap

pytestmark = pytest.mark.skipif(not codec_available(), reason="native codec not built")

POD_COLS = ("pod_req", "pod_flags", "pod_sel_hash", "pod_sel_pairs",
            "pod_sel_count", "pod_node", "pod_pdb",
            "pod_label_pairs", "pod_label_count")


def _compare(pods_json, nodes_json, pdbs=None):
    pods, nodes = wrap(pods_json, nodes_json)
    pools = group_nodes_into_pools(nodes)
    ref = encode_snapshot(pods, nodes, pools, pdbs)
    body = json.dumps({"kind": "PodList", "items": pods_json, "metadata": {}}).encode()
    fast = encode_snapshot_fast(body, nodes, pools, pdbs)
    for col in POD_COLS:
        np.testing.assert_array_equal(
            getattr(fast, col), getattr(ref, col), err_msg=col
        )
    np.testing.assert_array_equal(fast.node_free, ref.node_free)
    assert fast.pod_uids == ref.pod_uids
    assert fast.pod_refs == ref.pod_refs
    return fast


def test_codec_burst_trace():
    pods_json, nodes_json = burst_trace(
        500, n_nodes_per_pool=(5, 3),
        pools=(("agentpool1", "Standard_D2_v2"), ("gpupool", "Standard_ND96_MI355X_v1")),
        gpu_fraction=0.4, seed=5, bound_pods_per_node=3,
    )
    _compare(pods_json, nodes_json)


def test_codec_scaledown_with_pdbs():
    pods_json, nodes_json, pdbs = scaledown_trace(n_nodes=30, seed=2)
    _compare(pods_json, nodes_json, pdbs)


def test_codec_edge_cases():
    node = make_node_json("agentpool1", 0)
    nn = node["metadata"]["name"]
    pods = [
        make_pod_json("zero", cpu_milli=0, mem_mib=0, pending=True),
        make_pod_json("frac", cpu_milli=0, mem_mib=0, pending=True),
        make_pod_json("mirror", node_name=nn, owner_kind=None,
                      annotations={"kubernetes.io/config.mirror": "x"}),
        make_pod_json("crit", node_name=nn,
                      annotations={"scheduler.alpha.kubernetes.io/critical-pod": "true"}),
        make_pod_json("noev", node_name=nn,
                      annotations={"cluster-autoscaler.kubernetes.io/safe-to-evict": "False"}),
        make_pod_json("ds", node_name=nn, owner_kind="DaemonSet"),
        make_pod_json("bare", node_name=nn, owner_kind=None),
        make_pod_json("ghost", node_name="k8s-nosuch-12345678-9"),
    ]
    # odd quantities: fractional cpu, raw bytes, decimal suffixes
    pods[1]["spec"]["containers"][0]["resources"]["requests"] = {
        "cpu": "1.5", "memory": "1000001", "amd.com/gpu": "2",
    }
    # multi-container + init container
    pods[0]["spec"]["containers"].append(
        {"name": "side", "resources": {"requests": {"cpu": "250m", "memory": "100M"}}}
    )
    pods[0]["spec"]["initContainers"] = [
        {"name": "init", "resources": {"requests": {"cpu": "3", "memory": "2Gi"}}}
    ]
    _compare(pods, [node])


def test_codec_unicode_and_escapes():
    node = make_node_json("agentpool1", 0)
    p = make_pod_json("uni", pending=True, node_selector={"zone": "eu-westé"},
                      labels={"app": 'with"quote\\and\tctrl'})
    _compare([p], [node])


def test_codec_selector_hash_matches_python():
    node = make_node_json("gpupool", 0, gpu=8, instance_type="Standard_ND96_MI355X_v1")
    p = make_pod_json("g", gpu=1, pending=True,
                      node_selector={"accelerator": "amd-mi355x", "zone": "a"})
    snap = _compare([p], [node])
    assert snap.pod_sel_count[0] == 2


def test_codec_empty_list():
    node = make_node_json("agentpool1", 0)
    _compare([], [node])


def test_codec_selector_widths_and_overflow_poison():
    """Widths around the 32-pair lattice edge: both encoders must agree
    column-exactly, including the >32 poison (matches-nothing) row."""
    from mi355x_autoscaler.snapshot.encode import (
        MAX_SELECTOR_PAIRS, SELECTOR_POISON, PodFlags)

    node = make_node_json("agentpool1", 0)
    pods = []
    for w in (31, 32, 33, 40):
        sel = {f"k{i:02d}": f"v{i}" for i in range(w)}
        pods.append(make_pod_json(f"w{w}", cpu_milli=50, pending=True,
                                  node_selector=sel))
    snap = _compare(pods, [node])
    assert list(snap.pod_sel_count) == [31, 32, 32, 32]
    for row, w in enumerate((31, 32, 33, 40)):
        if w > MAX_SELECTOR_PAIRS:
            assert snap.pod_flags[row] & PodFlags.SELECTOR_OVERFLOW
            assert snap.pod_sel_pairs[row, -1] == SELECTOR_POISON
        else:
            assert not (snap.pod_flags[row] & PodFlags.SELECTOR_OVERFLOW)


def test_codec_decision_parity():
    """End goal: identical decisions through either ingest path."""
    from mi355x_autoscaler.sim.engine import CpuEngine

    pods_json, nodes_json = burst_trace(800, n_nodes_per_pool=(4,), seed=9,
                                        bound_pods_per_node=2)
    pods, nodes = wrap(pods_json, nodes_json)
    pools = group_nodes_into_pools(nodes)
    ref = encode_snapshot(pods, nodes, pools)
    body = json.dumps({"items": pods_json}).encode()
    fast = encode_snapshot_fast(body, nodes, pools)
    d1 = CpuEngine().decide(ref)
    d2 = CpuEngine().decide(fast)
    assert d1.units_needed == d2.units_needed
    assert d1.pods_placed_existing == d2.pods_placed_existing


def test_decode_watch_events_batch_matches_per_event():
    """C++ batch decode + vectorized apply == per-event Python apply."""
    import mi355x_autoscaler._hip_ops as ops
    from mi355x_autoscaler.models.kube import KubeNode
    from mi355x_autoscaler.snapshot.delta import DeltaSnapshotCache

    nodes = [KubeNode(make_node_json("agentpool1", i)) for i in range(3)]
    pools = group_nodes_into_pools(nodes)
    base = encode_snapshot([], nodes, pools)
    node_index = {n.name: i for i, n in enumerate(nodes)}

    initial = [make_pod_json(f"p{i}", cpu_milli=100, pending=True) for i in range(10)]
    from mi355x_autoscaler.models.kube import KubePod

    snap0 = encode_snapshot([KubePod(p) for p in initial], nodes, pools)

    events = []
    for i in range(30):
        events.append({"type": "ADDED", "object": make_pod_json(
            f"n{i}", cpu_milli=200 + i, mem_mib=64, pending=True,
            labels={"app": f"w{i % 4}", "tier": "web"} if i % 3 else None)})
    events.append({"type": "MODIFIED", "object": make_pod_json(
        "p0", cpu_milli=999, node_name=nodes[0].name)})
    events.append({"type": "DELETED", "object": initial[1]})
    events.append({"type": "BOOKMARK", "object": initial[2]})  # ignored kind
    blob = ("\n".join(json.dumps(e) for e in events) + "\n").encode()

    # path A: per-event python
    ca = DeltaSnapshotCache()
    ca.prime(snap0, node_index)
    for e in events:
        if e["type"] in ("ADDED", "MODIFIED", "DELETED"):
            ca.apply_event(e)

    # path B: C++ batch decode + vectorized apply
    cb = DeltaSnapshotCache()
    cb.prime(snap0, node_index)
    decoded = ops.decode_watch_events(blob, node_index, [])
    cb.apply_batch(decoded)

    sa = ca.build_snapshot(base)
    sb = cb.build_snapshot(base)
    for col in POD_COLS:
        order_a = np.argsort(np.array(sa.pod_uids, dtype=object).astype(str))
        order_b = np.argsort(np.array(sb.pod_uids, dtype=object).astype(str))
        np.testing.assert_array_equal(
            getattr(sa, col)[order_a], getattr(sb, col)[order_b], err_msg=col)
    assert sorted(u for u in sa.pod_uids if u) == sorted(u for u in sb.pod_uids if u)


def test_resolve_batch_rows_matches_python_loop():
    """C++ row resolution == the Python reference loop on adversarial
    orderings: duplicate MODIFIED, ADD-then-DELETE, DELETE-then-re-ADD
    (row reuse within one batch)."""
    import mi355x_autoscaler._hip_ops as ops
    from mi355x_autoscaler.models.kube import KubeNode, KubePod
    from mi355x_autoscaler.snapshot.delta import DeltaSnapshotCache

    nodes = [KubeNode(make_node_json("agentpool1", i)) for i in range(2)]
    pools = group_nodes_into_pools(nodes)
    node_index = {n.name: i for i, n in enumerate(nodes)}
    initial = [make_pod_json(f"p{i}", cpu_milli=100, pending=True) for i in range(5)]
    snap0 = encode_snapshot([KubePod(p) for p in initial], nodes, pools)
    base = encode_snapshot([], nodes, pools)

    events = []
    events.append({"type": "MODIFIED", "object": make_pod_json("p0", cpu_milli=111, pending=True)})
    events.append({"type": "MODIFIED", "object": make_pod_json("p0", cpu_milli=222, pending=True)})
    events.append({"type": "ADDED", "object": make_pod_json("tmp", cpu_milli=300, pending=True)})
    events.append({"type": "DELETED", "object": make_pod_json("tmp")})
    events.append({"type": "DELETED", "object": initial[1]})
    events.append({"type": "ADDED", "object": make_pod_json("reuse", cpu_milli=400, pending=True)})
    events.append({"type": "DELETED", "object": make_pod_json("ghost-never-seen")})
    blob = ("\n".join(json.dumps(e) for e in events) + "\n").encode()
    decoded = ops.decode_watch_events(blob, node_index, [])

    caches = []
    for use_cpp in (True, False):
        c = DeltaSnapshotCache()
        c.prime(snap0, node_index)
        c.drain_dirty()
        if use_cpp:
            c.apply_batch(decoded)
        else:
            c._apply_batch_py(decoded)
        caches.append(c)
    ca, cb = caches
    assert sorted(u for u in ca.pod_uids[: ca.size] if u) == \
           sorted(u for u in cb.pod_uids[: cb.size] if u)
    assert ca.uid_row == cb.uid_row
    assert sorted(ca.free_rows) == sorted(cb.free_rows)
    sa, sb = ca.build_snapshot(base), cb.build_snapshot(base)
    for col in POD_COLS:
        np.testing.assert_array_equal(getattr(sa, col), getattr(sb, col), err_msg=col)
    # the duplicate-MODIFIED pod must hold the LAST value
    row = ca.uid_row[[u for u in ca.pod_uids if u and "p0" in u][0]]
    assert ca.arrays["pod_req"][row][0] == 222
