import numpy as np

from mi355x_autoscaler.kube.objects import PodDisruptionBudget
from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.sim.oracle import OracleSimulator
from mi355x_autoscaler.sim.synth import make_node_json, make_pod_json, scaledown_trace, wrap
from mi355x_autoscaler.snapshot.encode import encode_snapshot


def _snap(pods_json, nodes_json, pdbs=None):
    pods, nodes = wrap(pods_json, nodes_json)
    pools = group_nodes_into_pools(nodes)
    return encode_snapshot(pods, nodes, pools, pdbs)


def test_idle_node_is_candidate():
    busy = make_node_json("agentpool1", 0)
    idle = make_node_json("agentpool1", 1)
    pod = make_pod_json("w", cpu_milli=1500, node_name=busy["metadata"]["name"])
    snap = _snap([pod], [busy, idle])
    rep = OracleSimulator(snap).maintain_report()
    names = [c.node_name for c in rep.candidates]
    assert idle["metadata"]["name"] in names
    assert rep.node_idle[1] == 1
    assert rep.node_util_permille[0] == 750  # 1500/2000


def test_pinned_pod_blocks_drain():
    node = make_node_json("agentpool1", 0)
    bare = make_pod_json("bare", cpu_milli=100, node_name=node["metadata"]["name"], owner_kind=None)
    snap = _snap([bare], [node])
    rep = OracleSimulator(snap).maintain_report()
    assert rep.node_blocked[0] == 1
    assert rep.candidates == []


def test_daemonset_and_mirror_pods_keep_node_idle():
    node = make_node_json("agentpool1", 0)
    ds = make_pod_json("ds", cpu_milli=100, node_name=node["metadata"]["name"], owner_kind="DaemonSet")
    mirror = make_pod_json(
        "m", cpu_milli=50, node_name=node["metadata"]["name"], owner_kind=None,
        annotations={"kubernetes.io/config.mirror": "abc"},
    )
    snap = _snap([ds, mirror], [node])
    rep = OracleSimulator(snap).maintain_report()
    assert rep.node_idle[0] == 1
    assert rep.node_blocked[0] == 0


def test_pdb_zero_budget_blocks_drain():
    node = make_node_json("agentpool1", 0)
    pod = make_pod_json("db", cpu_milli=100, node_name=node["metadata"]["name"], labels={"app": "db"})
    pdb0 = PodDisruptionBudget("g", match_labels={"app": "db"}, disruptions_allowed=0)
    snap = _snap([pod], [node], [pdb0])
    rep = OracleSimulator(snap).maintain_report()
    assert rep.node_pdb_ok[0] == 0
    assert rep.candidates == []

    pdb1 = PodDisruptionBudget("g", match_labels={"app": "db"}, disruptions_allowed=1)
    snap2 = _snap([pod], [node], [pdb1])
    rep2 = OracleSimulator(snap2).maintain_report()
    assert rep2.node_pdb_ok[0] == 1
    assert len(rep2.candidates) == 1


def test_under_utilized_vs_busy_threshold():
    node = make_node_json("agentpool1", 0)
    pod = make_pod_json("w", cpu_milli=500, node_name=node["metadata"]["name"])  # 250 permille
    snap = _snap([pod], [node])
    rep = OracleSimulator(snap).maintain_report(util_threshold_permille=300)
    assert [c.node_name for c in rep.candidates] == [node["metadata"]["name"]]
    rep2 = OracleSimulator(snap).maintain_report(util_threshold_permille=200)
    assert rep2.candidates == []


def test_candidates_sorted_lowest_util_first():
    n0 = make_node_json("agentpool1", 0)
    n1 = make_node_json("agentpool1", 1)
    p = make_pod_json("w", cpu_milli=400, node_name=n1["metadata"]["name"])  # 200 permille
    snap = _snap([p], [n0, n1])
    rep = OracleSimulator(snap).maintain_report()
    assert rep.candidates[0].node_name == n0["metadata"]["name"]  # idle (0) first


def test_scaledown_trace_500_nodes():
    pods_json, nodes_json, pdbs = scaledown_trace(n_nodes=100, seed=3)
    snap = _snap(pods_json, nodes_json, pdbs)
    rep = OracleSimulator(snap).maintain_report()
    # every idle node (util level 0) must be a candidate
    idle_count = int(rep.node_idle.sum())
    assert idle_count >= 100 // 5  # util_levels has a 0 slot
    assert len(rep.candidates) > 0
    # candidates exclude every blocked / pdb-bad node
    for c in rep.candidates:
        assert rep.node_blocked[c.node_index] == 0
        assert rep.node_pdb_ok[c.node_index] == 1


def test_pdb_namespace_scoping():
    """A v1 PDB selects pods in ITS OWN namespace only: identical
    labels in another namespace must not be guarded — consistently
    across the Python encoder, the C++ codec, and the watch cache."""
    import json

    from mi355x_autoscaler.kube.objects import PodDisruptionBudget
    from mi355x_autoscaler.models.kube import KubeNode, KubePod
    from mi355x_autoscaler.snapshot.codec import codec_available, encode_snapshot_fast
    from mi355x_autoscaler.snapshot.delta import DeltaSnapshotCache

    node = make_node_json("agentpool1", 0)
    nodes = [KubeNode(node)]
    pools = group_nodes_into_pools(nodes)
    nn = node["metadata"]["name"]
    guarded = make_pod_json("g", cpu_milli=100, node_name=nn,
                            labels={"app": "web"})
    other = make_pod_json("o", cpu_milli=100, node_name=nn,
                          namespace="other", labels={"app": "web"})
    pdb = PodDisruptionBudget(name="guard", match_labels={"app": "web"},
                              disruptions_allowed=0, namespace="default")

    pods_json = [guarded, other]
    snap = encode_snapshot([KubePod(p) for p in pods_json], nodes, pools, [pdb])
    by_uid = {u: i for i, u in enumerate(snap.pod_uids)}
    assert snap.pod_pdb[by_uid["pod-default-g"]] == 0
    assert snap.pod_pdb[by_uid["pod-other-o"]] == -1  # different namespace

    if codec_available():
        body = json.dumps({"items": pods_json}).encode()
        fast = encode_snapshot_fast(body, nodes, pools, [pdb])
        fby = {u: i for i, u in enumerate(fast.pod_uids)}
        assert fast.pod_pdb[fby["pod-default-g"]] == 0
        assert fast.pod_pdb[fby["pod-other-o"]] == -1

    # watch cache: per-event encode + in-place reassign
    cache = DeltaSnapshotCache()
    cache.prime(encode_snapshot([], nodes, pools, [pdb]), {nn: 0},
                [(pdb.namespace, dict(pdb.match_labels))])
    cache.apply_event({"type": "ADDED", "object": guarded})
    cache.apply_event({"type": "ADDED", "object": other})
    rows = {u: i for i, u in enumerate(cache.pod_uids[: cache.size])}
    assert cache.arrays["pod_pdb"][rows["pod-default-g"]] == 0
    assert cache.arrays["pod_pdb"][rows["pod-other-o"]] == -1
    # reassign with the PDB moved to namespace "other"
    n = cache.reassign_pdbs([("other", {"app": "web"})])
    assert n == 2  # g unguarded, o guarded
    assert cache.arrays["pod_pdb"][rows["pod-default-g"]] == -1
    assert cache.arrays["pod_pdb"][rows["pod-other-o"]] == 0


def test_pdb_empty_selector_selects_whole_namespace():
    """policy/v1: a PDB with EMPTY matchLabels selects every pod in its
    namespace — consistent across the Python encoder, the C++ codec,
    and the watch cache's in-place reassign."""
    import json

    from mi355x_autoscaler.models.kube import KubeNode, KubePod
    from mi355x_autoscaler.snapshot.codec import codec_available, encode_snapshot_fast
    from mi355x_autoscaler.snapshot.delta import DeltaSnapshotCache

    node = make_node_json("agentpool1", 0)
    nodes = [KubeNode(node)]
    pools = group_nodes_into_pools(nodes)
    nn = node["metadata"]["name"]
    a = make_pod_json("a", cpu_milli=100, node_name=nn, labels={"app": "x"})
    b = make_pod_json("b", cpu_milli=100, node_name=nn)  # no labels at all
    other = make_pod_json("o", cpu_milli=100, node_name=nn, namespace="other")
    pdb = PodDisruptionBudget(name="all", match_labels={},
                              disruptions_allowed=0, namespace="default")

    pods_json = [a, b, other]
    snap = encode_snapshot([KubePod(p) for p in pods_json], nodes, pools, [pdb])
    by_uid = {u: i for i, u in enumerate(snap.pod_uids)}
    assert snap.pod_pdb[by_uid["pod-default-a"]] == 0
    assert snap.pod_pdb[by_uid["pod-default-b"]] == 0
    assert snap.pod_pdb[by_uid["pod-other-o"]] == -1  # other namespace

    if codec_available():
        body = json.dumps({"items": pods_json}).encode()
        fast = encode_snapshot_fast(body, nodes, pools, [pdb])
        fby = {u: i for i, u in enumerate(fast.pod_uids)}
        for u, want in (("pod-default-a", 0), ("pod-default-b", 0),
                        ("pod-other-o", -1)):
            assert fast.pod_pdb[fby[u]] == want, u

    # in-place reassign from cached label hashes must agree
    cache = DeltaSnapshotCache()
    cache.prime(snap, {nn: 0}, [("default", {})])
    changed = cache.reassign_pdbs([("default", {})])
    assert changed == 0  # already assigned identically by encode


def test_pdb_on_daemonset_pod_does_not_block_drain():
    """A zero-budget PDB guarding only a DAEMONSET pod must not mark
    the node pdb_bad: drains never evict daemonset/mirror pods, so no
    disruption is consumed (kernel D + oracle + host accounting)."""
    node = make_node_json("agentpool1", 0)
    ds = make_pod_json("ds", cpu_milli=100, node_name=node["metadata"]["name"],
                       owner_kind="DaemonSet", labels={"app": "infra"})
    pdb0 = PodDisruptionBudget("g", match_labels={"app": "infra"},
                               disruptions_allowed=0)
    snap = _snap([ds], [node], [pdb0])
    assert snap.pod_pdb[0] == 0  # the pod IS guarded...
    rep = OracleSimulator(snap).maintain_report()
    assert rep.node_pdb_ok[0] == 1  # ...but consumes no budget
    assert len(rep.candidates) == 1
