import numpy as np

from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.snapshot.encode import (
    MIB,
    PodFlags,
    encode_snapshot,
    fnv1a64,
    hash_label_pair,
    selector_subset_match,
)
from mi355x_autoscaler.sim.synth import burst_trace, make_node_json, make_pod_json, wrap


def _snap(pods_json, nodes_json, pdbs=None):
    pods, nodes = wrap(pods_json, nodes_json)
    pools = group_nodes_into_pools(nodes)
    return encode_snapshot(pods, nodes, pools, pdbs)


def test_fnv1a64_stability():
    # pinned value: the HIP-side hasher must reproduce these exactly
    assert fnv1a64(b"") == 0xCBF29CE484222325
    assert fnv1a64(b"a") == 0xAF63DC4C8601EC8C
    h1 = hash_label_pair("agentpool", "gpupool")
    h2 = hash_label_pair("agentpool", "gpupool")
    assert h1 == h2
    assert hash_label_pair("a", "b") != hash_label_pair("a", "c")
    # key/value boundary is unambiguous (a\x1fb vs ab)
    assert hash_label_pair("ab", "") != hash_label_pair("a", "b")


def test_quantization_rounding():
    # request memory rounds UP to MiB, allocatable rounds DOWN
    pod = make_pod_json("p", cpu_milli=100, mem_mib=0, pending=True)
    pod["spec"]["containers"][0]["resources"]["requests"]["memory"] = "1000001"  # bytes
    node = make_node_json("agentpool1", 0)
    node["status"]["allocatable"]["memory"] = "1048577"  # 1 MiB + 1 byte
    snap = _snap([pod], [node])
    assert snap.pod_req[0, 1] == 1  # ceil(1000001 B / MiB)
    assert snap.node_alloc[0, 1] == 1  # floor


def test_flags_and_binding():
    node = make_node_json("agentpool1", 0)
    bound = make_pod_json("b", node_name=node["metadata"]["name"])
    pend = make_pod_json("p", pending=True)
    snap = _snap([bound, pend], [node])
    assert snap.pod_flags[0] & PodFlags.BOUND
    assert snap.pod_node[0] == 0
    assert snap.pod_flags[1] & PodFlags.PENDING
    assert snap.pod_node[1] == -1


def test_node_free_subtracts_bound_pods():
    node = make_node_json("agentpool1", 0, cpu_milli=2000, mem_mib=4096)
    p1 = make_pod_json("a", cpu_milli=500, mem_mib=1024, node_name=node["metadata"]["name"])
    p2 = make_pod_json("b", cpu_milli=300, mem_mib=512, node_name=node["metadata"]["name"])
    snap = _snap([p1, p2], [node])
    assert snap.node_free[0, 0] == 2000 - 800
    assert snap.node_free[0, 1] == 4096 - 1536
    assert snap.node_free[0, 2] == snap.node_alloc[0, 2] - 2


def test_selector_subset_match_device_form():
    node = make_node_json("gpupool", 0, gpu=8, instance_type="Standard_ND96_MI355X_v1")
    pod = make_pod_json("g", gpu=1, node_selector={"accelerator": "amd-mi355x"}, pending=True)
    bad = make_pod_json("x", node_selector={"accelerator": "tpu"}, pending=True)
    snap = _snap([pod, bad], [node])
    ok = selector_subset_match(
        snap.pod_sel_pairs[0], int(snap.pod_sel_count[0]),
        snap.node_labels[0], int(snap.node_label_count[0]),
    )
    assert ok
    assert not selector_subset_match(
        snap.pod_sel_pairs[1], int(snap.pod_sel_count[1]),
        snap.node_labels[0], int(snap.node_label_count[0]),
    )
    # empty selector matches everything
    empty = np.zeros(8, np.int64)
    assert selector_subset_match(empty, 0, snap.node_labels[0], int(snap.node_label_count[0]))


def test_pdb_assignment():
    from mi355x_autoscaler.kube.objects import PodDisruptionBudget

    node = make_node_json("agentpool1", 0)
    p = make_pod_json("guarded", node_name=node["metadata"]["name"], labels={"app": "db"})
    q = make_pod_json("free", node_name=node["metadata"]["name"], labels={"app": "web"})
    pdb = PodDisruptionBudget(name="db-pdb", match_labels={"app": "db"}, disruptions_allowed=0)
    snap = _snap([p, q], [node], [pdb])
    assert snap.pod_pdb[0] == 0
    assert snap.pod_pdb[1] == -1
    assert snap.pdb_allowed[0] == 0


def test_snapshot_scales_and_is_compact():
    pods_json, nodes_json = burst_trace(2000, n_nodes_per_pool=(50,), seed=1)
    snap = _snap(pods_json, nodes_json)
    assert snap.num_pods == 2000
    assert snap.num_nodes == 50
    # SoA must stay lean: << 1 KiB per pod (288 GB HBM fits ~1e9 pods)
    assert snap.nbytes() / snap.num_pods < 1024


def test_synth_soa_matches_lattice_semantics():
    """The vectorized SoA generator must produce pods the oracle routes
    exactly like JSON-encoded equivalents (same hashes, same flags)."""
    from mi355x_autoscaler.sim.oracle import OracleSimulator
    from mi355x_autoscaler.sim.synth_soa import synth_burst_snapshot

    snap = synth_burst_snapshot(500, n_nodes_per_pool=(3, 2), pools=(
        ("agentpool1", "Standard_D2_v2"), ("gpupool", "Standard_ND96_MI355X_v1")),
        gpu_fraction=0.5, seed=4, bound_pods_per_node=2)
    assert snap.num_pods == 500 + 5 * 2
    d = OracleSimulator(snap).fulfill_pending()
    assert d.pods_placed_existing + d.pods_placed_new + d.pods_unfulfillable == 500
    # GPU-selector pods may only land on / route to the gpu pool
    if "gpupool" in d.units_needed:
        assert d.units_needed["gpupool"] > 0

    # hash equivalence with the JSON path for a gpu pod
    from mi355x_autoscaler.models.kube import KubePod
    from mi355x_autoscaler.sim.synth import make_pod_json
    from mi355x_autoscaler.snapshot.encode import hash_label_pair

    pod = KubePod(make_pod_json("g", gpu=1, node_selector={"accelerator": "amd-mi355x"}, pending=True))
    h = hash_label_pair("accelerator", "amd-mi355x")
    h_i64 = h - (1 << 64) if h >= (1 << 63) else h
    gpu_rows = snap.pod_sel_count[:500] == 1
    assert (snap.pod_sel_hash[:500][gpu_rows] == h_i64).all()


def test_synth_soa_million_scale_fast():
    """1M-pod generation must be vectorized (sub-ten-seconds, SoA only)."""
    import time

    from mi355x_autoscaler.sim.synth_soa import synth_burst_snapshot

    t0 = time.time()
    snap = synth_burst_snapshot(1_000_000, n_nodes_per_pool=(50, 25, 25), seed=1)
    dt = time.time() - t0
    assert snap.num_pods == 1_000_000
    assert dt < 10.0
    assert snap.nbytes() < 300 * 1024 * 1024


def test_wide_selector_sets_match_exactly():
    """12 selector pairs (> the old 8-pair width) must match exactly —
    no truncation over-match; >32 sets SELECTOR_OVERFLOW + poison."""
    from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
    from mi355x_autoscaler.models.kube import KubeNode, KubePod
    from mi355x_autoscaler.sim.engine import CpuEngine
    from mi355x_autoscaler.sim.synth import make_node_json, make_pod_json
    from mi355x_autoscaler.snapshot.encode import PodFlags, encode_snapshot

    sel = {f"k{i:02d}": f"v{i}" for i in range(12)}
    node_full = make_node_json("agentpool1", 0)
    node_full["metadata"]["labels"].update(sel)
    node_partial = make_node_json("agentpool1", 1)
    node_partial["metadata"]["labels"].update({k: sel[k] for k in list(sel)[:11]})
    nodes = [KubeNode(node_full), KubeNode(node_partial)]
    pod = make_pod_json("wide", cpu_milli=100, pending=True, node_selector=sel)
    snap = encode_snapshot([KubePod(pod)], nodes, group_nodes_into_pools(nodes))
    assert snap.pod_sel_count[0] == 12
    assert not (snap.pod_flags[0] & PodFlags.SELECTOR_OVERFLOW)
    d = CpuEngine().decide(snap)
    assert d.pods_placed_existing == 1  # fits node_full only

    # 17 pairs sat above the old 16-wide lattice; now matches exactly.
    sel17 = {f"k{i:02d}": f"v{i}" for i in range(17)}
    node17 = make_node_json("agentpool1", 2)
    node17["metadata"]["labels"].update(sel17)
    nodes17 = [KubeNode(node17), nodes[1]]
    pod17 = make_pod_json("wider", cpu_milli=100, pending=True, node_selector=sel17)
    snap17 = encode_snapshot(
        [KubePod(pod17)], nodes17, group_nodes_into_pools(nodes17))
    assert snap17.pod_sel_count[0] == 17
    assert not (snap17.pod_flags[0] & PodFlags.SELECTOR_OVERFLOW)
    d17 = CpuEngine().decide(snap17)
    assert d17.pods_placed_existing == 1  # fits node17 only

    # >32 pairs cannot subset-match any representable (<=32-label) node:
    # flagged SELECTOR_OVERFLOW and poisoned so it matches NOTHING, even
    # a node whose first 32 sorted labels cover the truncated pairs.
    sel33 = {f"k{i:02d}": f"v{i}" for i in range(33)}
    node33 = make_node_json("agentpool1", 3)
    node33["metadata"]["labels"].update(sel33)  # >32 labels: truncates too
    nodes33 = [KubeNode(node33)]
    pod33 = make_pod_json("widest", cpu_milli=100, pending=True, node_selector=sel33)
    snap33 = encode_snapshot(
        [KubePod(pod33)], nodes33, group_nodes_into_pools(nodes33))
    assert snap33.pod_flags[0] & PodFlags.SELECTOR_OVERFLOW
    from mi355x_autoscaler.snapshot.encode import SELECTOR_POISON
    assert snap33.pod_sel_pairs[0, -1] == SELECTOR_POISON
    d33 = CpuEngine().decide(snap33)
    assert d33.pods_placed_existing == 0


def test_subset_match_mixed_sign_hashes():
    """The pair hashes are sorted UNSIGNED; the subset walk must compare
    unsigned too (a 12-pair selector mixes positive- and negative-cast
    int64 hashes — the old signed walk failed on exactly that)."""
    from mi355x_autoscaler.snapshot.encode import _hash_pairs, selector_subset_match

    sel = {f"k{i:02d}": f"v{i}" for i in range(12)}
    labels = dict(sel)
    labels.update({f"extra{i}": "x" for i in range(5)})
    sp, sc, _ = _hash_pairs(sel, 16)
    lp, lc, _ = _hash_pairs(labels, 32)
    assert any(int(x) < 0 for x in sp[:sc]) and any(int(x) > 0 for x in sp[:sc])
    assert selector_subset_match(sp, sc, lp, min(lc, 32))
    # removing one selected label must break the match
    labels2 = dict(labels)
    del labels2["k07"]
    lp2, lc2, _ = _hash_pairs(labels2, 32)
    assert not selector_subset_match(sp, sc, lp2, min(lc2, 32))


def test_pool_routing_wide_selector():
    """Oracle pool routing (selector_subset_match walk) with a 12-pair
    selector: the pool whose prototype labels carry all pairs routes."""
    from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
    from mi355x_autoscaler.models.kube import KubeNode, KubePod
    from mi355x_autoscaler.sim.engine import CpuEngine
    from mi355x_autoscaler.sim.synth import make_node_json, make_pod_json
    from mi355x_autoscaler.snapshot.encode import encode_snapshot

    sel = {f"k{i:02d}": f"v{i}" for i in range(12)}
    node = make_node_json("agentpool1", 0, cpu_milli=2000, mem_mib=2048)
    node["metadata"]["labels"].update(sel)
    nodes = [KubeNode(node)]
    # more demand than one node holds -> must route to the pool
    pods = [make_pod_json(f"w{i}", cpu_milli=1500, pending=True, node_selector=sel)
            for i in range(6)]
    snap = encode_snapshot([KubePod(p) for p in pods], nodes,
                           group_nodes_into_pools(nodes))
    d = CpuEngine().decide(snap)
    assert d.pods_placed_existing == 1
    assert d.units_needed.get("agentpool1", 0) >= 1
    assert d.pods_unfulfillable == 0
