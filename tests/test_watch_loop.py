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


def test_watch_node_churn_remaps_in_place(server):
    """Node additions/removals must NOT re-prime (relist every pod):
    the cache remaps pod->node indices in place and decisions stay
    identical to a fresh full encode."""
    pods, nodes = burst_trace(4, n_nodes_per_pool=(3,), seed=2,
                              bound_pods_per_node=2)
    server.state.load(pods, nodes)
    cluster, _ = _cluster(server, no_maintenance=True, no_scale=True)
    cluster.loop_logic()
    first_state = cluster._watch_state
    first_cache = first_state["cache"]
    # add a node (watch-visible): the POD cache must survive un-primed
    server.state.add_node(make_node_json("agentpool1", 9))
    time.sleep(0.3)  # let the node watch deliver
    res = cluster.loop_logic()
    assert cluster._watch_state is first_state  # NOT re-primed
    assert cluster._watch_state["cache"] is first_cache
    assert res.nodes_total == 4
    assert res.pods_pending == 4

    # remove a node that has bound pods: indices of later nodes shift
    # and the orphaned bindings must clear, exactly as a relist would
    victim = sorted(server.state.nodes)[0]
    server.state.remove_node(victim)
    time.sleep(0.3)  # let the node watch deliver
    res2 = cluster.loop_logic()
    assert cluster._watch_state is first_state
    snap = res2.snapshot
    from mi355x_autoscaler.kube.client import KubeApiClient as _K
    from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
    from mi355x_autoscaler.models.kube import KubeNode, KubePod
    from mi355x_autoscaler.snapshot.encode import encode_snapshot

    kube = _K(server.base_url)
    fresh_nodes = [KubeNode(n) for n in kube.list_nodes()]
    fresh = encode_snapshot([KubePod(p) for p in kube.list_pods()],
                            fresh_nodes, group_nodes_into_pools(fresh_nodes))
    # same node table and identical per-node free capacity
    assert snap.node_names == fresh.node_names
    import numpy as np

    order_a = np.argsort(np.array([u or "" for u in snap.pod_uids], dtype=object).astype(str))
    order_b = np.argsort(np.array([u or "" for u in fresh.pod_uids], dtype=object).astype(str))
    live_a = np.array([bool(u) for u in snap.pod_uids])[order_a]
    np.testing.assert_array_equal(
        snap.pod_node[order_a][live_a],
        fresh.pod_node[order_b][np.array([bool(u) for u in fresh.pod_uids])[order_b]])
    np.testing.assert_array_equal(snap.node_free, fresh.node_free)
    d1 = CpuEngine().decide(snap)
    d2 = CpuEngine().decide(fresh)
    assert d1.units_needed == d2.units_needed
    assert d1.pods_placed_existing == d2.pods_placed_existing


def test_remap_nodes_unit():
    """DeltaSnapshotCache.remap_nodes: shifted indices rewrite, lost
    nodes unbind + clear BOUND, dirty rows are tracked."""
    import numpy as np

    from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
    from mi355x_autoscaler.models.kube import KubeNode, KubePod
    from mi355x_autoscaler.snapshot.delta import DeltaSnapshotCache
    from mi355x_autoscaler.snapshot.encode import PodFlags, encode_snapshot

    nodes = [KubeNode(make_node_json("agentpool1", i)) for i in range(3)]
    names = [n.name for n in nodes]
    pods = [make_pod_json(f"b{i}", cpu_milli=100, node_name=names[i]) for i in range(3)]
    pods.append(make_pod_json("p", cpu_milli=50, pending=True))
    snap = encode_snapshot([KubePod(p) for p in pods], nodes,
                           group_nodes_into_pools(nodes))
    cache = DeltaSnapshotCache()
    cache.prime(snap, {n: i for i, n in enumerate(names)})
    cache.drain_dirty()

    # drop node 0; node 1 -> 0, node 2 -> 1; new node appended at 2
    new_index = {names[1]: 0, names[2]: 1, "k8s-agentpool1-12345678-7": 2}
    changed = cache.remap_nodes(new_index)
    assert changed == 3  # all three bound rows changed
    pn = cache.arrays["pod_node"][: cache.size]
    fl = cache.arrays["pod_flags"][: cache.size]
    # rows 0..2 were the bound pods, in pod order
    assert pn[0] == -1 and not (fl[0] & PodFlags.BOUND)
    assert pn[1] == 0 and (fl[1] & PodFlags.BOUND)
    assert pn[2] == 1 and (fl[2] & PodFlags.BOUND)
    rows = cache.drain_dirty()
    assert set(rows.tolist()) >= {0, 1, 2}


def test_pdb_set_change_reassigns_in_place(server):
    """A changed PDB set re-derives the cached pod_pdb column from the
    cached pod LABEL pair hashes — NO re-prime (pod relist) — and
    drain scoring honors the new budget."""
    nodes = [make_node_json("agentpool1", i) for i in range(3)]
    busy = make_pod_json("busy", cpu_milli=1500,
                         node_name=nodes[0]["metadata"]["name"])
    guarded = make_pod_json("g", cpu_milli=100,
                            node_name=nodes[1]["metadata"]["name"],
                            labels={"app": "guarded"})
    server.state.load([busy, guarded], nodes)
    cluster, provider = _cluster(server, no_scale=True, max_drains_per_cycle=10)

    # prime WITH a zero-budget PDB guarding the pod: no drain
    server.state.add_pdb({
        "metadata": {"name": "guard"},
        "spec": {"selector": {"matchLabels": {"app": "guarded"}}},
        "status": {"disruptionsAllowed": 0}})
    res1 = cluster.loop_logic()
    cache1 = cluster._watch_state["cache"]
    assert nodes[1]["metadata"]["name"] not in res1.drained_nodes

    # PDB set changes (removed, watch-visible): the pod un-guards IN
    # PLACE — same cache object, no pod relist — and the node drains
    server.state.remove_pdb("guard")
    time.sleep(0.3)  # let the PDB watch deliver
    res2 = cluster.loop_logic()
    assert cluster._watch_state["cache"] is cache1  # reassigned in place
    assert nodes[1]["metadata"]["name"] in res2.drained_nodes


def test_watch_mode_native_codec_pump(server):
    """The C++ ingest pump end-to-end through the loop: raw JSONL
    watch lines -> batched decode_watch_events -> vectorized apply ->
    decision — including a PDB-set change reassigned in place."""
    from mi355x_autoscaler.snapshot.codec import codec_available

    if not codec_available():
        pytest.skip("native codec not built")
    pods, nodes = burst_trace(6, n_nodes_per_pool=(3,), seed=4)
    server.state.load(pods, nodes)
    kube = KubeApiClient(server.base_url)
    provider = FakeCloudProvider(initial_pool_sizes={"agentpool1": 3},
                                 kube_state=server.state)
    opts = ClusterOptions(sleep_seconds=0, idle_threshold_seconds=0,
                          spare_agents=0, watch=True, native_codec=True,
                          no_scale=True, no_maintenance=True)
    cluster = Cluster(kube, provider, CpuEngine(), options=opts)

    res1 = cluster.loop_logic()
    assert res1.pods_pending == 6
    cache1 = cluster._watch_state["cache"]

    for i in range(5):
        server.state.add_pod(make_pod_json(
            f"late{i}", cpu_milli=100, pending=True,
            labels={"app": "guarded"}))
    _wait_events(cluster)
    time.sleep(0.3)
    res2 = cluster.loop_logic()
    assert res2.pods_pending == 11
    assert cluster._watch_state["cache"] is cache1

    # PDB appears (watch-visible): the C++-decoded rows carry label
    # pairs, so the reassign happens in place even for watch-added pods
    server.state.add_pdb({
        "metadata": {"name": "guard"},
        "spec": {"selector": {"matchLabels": {"app": "guarded"}}},
        "status": {"disruptionsAllowed": 0}})
    time.sleep(0.3)  # let the PDB watch deliver
    res3 = cluster.loop_logic()
    assert cluster._watch_state["cache"] is cache1  # no re-prime
    snap = res3.snapshot
    import numpy as np

    assert int((snap.pod_pdb >= 0).sum()) == 5  # the guarded late pods


def test_node_watch_caches_and_tracks(server):
    """World-1 watch mode keeps the node set under a watch too: the
    wrapped node list is reused while nothing changes (no per-cycle
    relist/re-wrap) and watch events update it within a cycle."""
    pods, nodes = burst_trace(5, n_nodes_per_pool=(3,), seed=4)
    server.state.load(pods, nodes)
    cluster, _ = _cluster(server, no_maintenance=True, no_scale=True)
    cluster.loop_logic()
    nw = cluster._node_watch
    assert nw is not None and not nw["broken"]
    wrapped1 = nw["wrapped"]
    assert len(wrapped1) == 3
    cluster.loop_logic()
    assert cluster._node_watch["wrapped"] is wrapped1  # reused, no rebuild

    server.state.add_node(make_node_json("agentpool1", 7))
    time.sleep(0.3)
    res = cluster.loop_logic()
    assert res.nodes_total == 4
    assert cluster._node_watch["wrapped"] is not wrapped1

    server.state.remove_node("k8s-agentpool1-12345678-7")
    time.sleep(0.3)
    assert cluster.loop_logic().nodes_total == 3


def test_bulk_injection_reaches_watchers(server):
    """bulk_add_pods (pre-serialized POST bodies) + the batched watch
    stream + bulk_remove_pods — the e2e reaction bench's path — must
    deliver every event to a watch-mode loop."""
    import json as _json

    pods, nodes = burst_trace(3, n_nodes_per_pool=(3,), seed=8)
    server.state.load(pods, nodes)
    cluster, _ = _cluster(server, no_maintenance=True, no_scale=True)
    cluster.loop_logic()  # prime

    wave = [make_pod_json(f"bulk{i}", cpu_milli=50, pending=True)
            for i in range(40)]
    lines = [_json.dumps({"type": "ADDED", "object": p}) + "\n" for p in wave]
    server.state.bulk_add_pods(wave, copy_objs=False, lines=lines)
    deadline = time.time() + 10
    while time.time() < deadline:
        res = cluster.loop_logic()
        if res.pods_pending >= 43:
            break
        time.sleep(0.05)
    assert res.pods_pending == 43

    server.state.bulk_remove_pods([f"default/bulk{i}" for i in range(40)])
    deadline = time.time() + 10
    while time.time() < deadline:
        res = cluster.loop_logic()
        if res.pods_pending <= 3:
            break
        time.sleep(0.05)
    assert res.pods_pending == 3


def test_pdb_watch_caches_and_tracks(server):
    """PDBs are watched too at world 1: the wrapped list is reused
    while nothing changes (no per-cycle PDB list round trip) and a
    MODIFIED budget reaches the drain decision via the watch."""
    nodes = [make_node_json("agentpool1", i) for i in range(3)]
    busy = make_pod_json("busy", cpu_milli=1500,
                         node_name=nodes[0]["metadata"]["name"])
    g = make_pod_json("g", cpu_milli=100,
                      node_name=nodes[1]["metadata"]["name"],
                      labels={"app": "db"})
    pdb0 = {"metadata": {"name": "dbguard"},
            "spec": {"selector": {"matchLabels": {"app": "db"}}},
            "status": {"disruptionsAllowed": 0}}
    server.state.load([busy, g], nodes, pdbs=[pdb0])
    cluster, _ = _cluster(server, no_scale=True, max_drains_per_cycle=10)
    res = cluster.loop_logic()
    # the guarded node cannot drain while the zero-budget PDB exists
    assert nodes[1]["metadata"]["name"] not in res.drained_nodes
    assert cluster._pdb_watch is not None
    wrapped1 = cluster._pdb_watch["wrapped"]
    res = cluster.loop_logic()
    assert cluster._pdb_watch["wrapped"] is wrapped1  # reused, no relist
    assert nodes[1]["metadata"]["name"] not in res.drained_nodes

    server.state.modify_pdb({
        "metadata": {"name": "dbguard"},
        "spec": {"selector": {"matchLabels": {"app": "db"}}},
        "status": {"disruptionsAllowed": 3}})
    time.sleep(0.3)
    res2 = cluster.loop_logic()
    assert cluster._pdb_watch["wrapped"] is not wrapped1  # event applied
    assert nodes[1]["metadata"]["name"] in res2.drained_nodes


def test_decoder_pipeline_stale_batch_redecodes(server):
    """A pre-decoded batch whose node-index/PDB identities no longer
    match the cache (a remap/reassign landed in between) must be
    re-decoded from the carried raw lines, not applied stale."""
    import json as _json

    from mi355x_autoscaler.snapshot.codec import codec_available

    if not codec_available():
        pytest.skip("native codec not built")
    import mi355x_autoscaler._hip_ops as _hip_ops

    pods, nodes = burst_trace(4, n_nodes_per_pool=(3,), seed=6)
    server.state.load(pods, nodes)
    kube = KubeApiClient(server.base_url)
    provider = FakeCloudProvider(initial_pool_sizes={"agentpool1": 3},
                                 kube_state=server.state)
    opts = ClusterOptions(sleep_seconds=0, idle_threshold_seconds=10**9,
                          spare_agents=0, watch=True, native_codec=True,
                          no_scale=True, no_maintenance=True)
    cluster = Cluster(kube, provider, CpuEngine(), options=opts)
    cluster.loop_logic()  # prime
    ws = cluster._watch_state
    cache = ws["cache"]

    # a pod BOUND to node index 2 under the CURRENT index
    target = sorted(server.state.nodes)[2]
    pod = make_pod_json("bound-late", cpu_milli=100, node_name=target)
    raw = [_json.dumps({"type": "ADDED", "object": pod}).encode()]
    stale_index = dict(cache.node_index)  # decoded against THIS map
    decoded = _hip_ops.decode_watch_events(
        b"\n".join(raw) + b"\n", stale_index, list(cache.pdb_selectors))
    # remap AFTER the decode: drop node 0, shifting the target 2 -> 1
    new_index = {n: i for i, n in
                 enumerate(nm for nm in sorted(cache.node_index) if
                           nm != sorted(cache.node_index)[0])}
    cache.remap_nodes(new_index)
    ws["node_names"] = tuple(sorted(new_index, key=new_index.get))
    # feed the now-STALE tuple as the decoder pump would have
    ws["q"].put(("decoded", id(stale_index), id(cache.pdb_selectors),
                 decoded, raw))
    server.state.remove_node(sorted(server.state.nodes)[0])
    import time as _t

    _t.sleep(0.3)  # node watch delivers the matching node set
    res = cluster.loop_logic()
    snap = res.snapshot
    rows = {u: i for i, u in enumerate(snap.pod_uids)}
    r = rows["pod-default-bound-late"]
    # correct binding under the NEW index (re-decoded), not the stale 2
    assert snap.node_names[int(snap.pod_node[r])] == target
