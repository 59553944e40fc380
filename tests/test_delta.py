"""DeltaSnapshotCache: watch-event delta ingest must converge to the
same snapshot a full re-encode would produce, through adds, mods,
deletes, slot reuse, compaction, and checkpoint/resume."""

import numpy as np
import pytest

from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.models.kube import KubeNode, KubePod
from mi355x_autoscaler.sim.engine import CpuEngine
from mi355x_autoscaler.sim.synth import make_node_json, make_pod_json
from mi355x_autoscaler.snapshot.delta import DeltaSnapshotCache
from mi355x_autoscaler.snapshot.encode import encode_snapshot


def _base(nodes):
    pools = group_nodes_into_pools(nodes)
    return encode_snapshot([], nodes, pools), pools


def _full(pods_json, nodes):
    pools = group_nodes_into_pools(nodes)
    return encode_snapshot([KubePod(p) for p in pods_json], nodes, pools)


def _assert_equivalent_decisions(snap_a, snap_b):
    da = CpuEngine().decide(snap_a)
    db = CpuEngine().decide(snap_b)
    assert da.units_needed == db.units_needed
    assert da.pods_placed_existing == db.pods_placed_existing
    assert da.pods_unfulfillable == db.pods_unfulfillable
    ra = CpuEngine().maintain(snap_a)
    rb = CpuEngine().maintain(snap_b)
    np.testing.assert_array_equal(ra.node_util_permille, rb.node_util_permille)


def test_delta_add_modify_delete_converges():
    nodes = [KubeNode(make_node_json("agentpool1", i)) for i in range(3)]
    base, pools = _base(nodes)
    initial = [make_pod_json(f"p{i}", cpu_milli=100, pending=True) for i in range(5)]
    cache = DeltaSnapshotCache()
    cache.prime(_full(initial, nodes), {n.name: i for i, n in enumerate(nodes)})

    # adds, a modification, a bind, deletes
    added = [make_pod_json(f"n{i}", cpu_milli=300, mem_mib=256, pending=True) for i in range(4)]
    for p in added:
        cache.apply_event({"type": "ADDED", "object": p})
    mod = dict(initial[0])
    mod_bound = make_pod_json("p0", cpu_milli=100, node_name=nodes[0].name)
    cache.apply_event({"type": "MODIFIED", "object": mod_bound})
    cache.apply_event({"type": "DELETED", "object": initial[1]})
    cache.apply_event({"type": "DELETED", "object": initial[2]})
    # slot reuse
    readd = make_pod_json("fresh", cpu_milli=50, pending=True)
    cache.apply_event({"type": "ADDED", "object": readd})

    final_json = [mod_bound, initial[3], initial[4]] + added + [readd]
    snap_delta = cache.build_snapshot(base)
    snap_full = _full(final_json, nodes)
    _assert_equivalent_decisions(snap_delta, snap_full)
    assert len(cache.drain_dirty()) > 0
    assert len(cache.drain_dirty()) == 0  # drained


def test_delta_overflow_selector_rows_match_full_encode():
    """Watch-added pods with >32-pair selectors take the poison path in
    the cache's row encoder exactly as in encode_snapshot: same pairs
    column (incl. the poison slot), same flags, same decision."""
    from mi355x_autoscaler.snapshot.encode import SELECTOR_POISON, PodFlags

    nodes = [KubeNode(make_node_json("agentpool1", i)) for i in range(2)]
    base, pools = _base(nodes)
    cache = DeltaSnapshotCache()
    cache.prime(_full([], nodes), {n.name: i for i, n in enumerate(nodes)})

    sel33 = {f"k{i:02d}": f"v{i}" for i in range(33)}
    ovf = make_pod_json("ovf", cpu_milli=100, pending=True, node_selector=sel33)
    plain = make_pod_json("plain", cpu_milli=100, pending=True)
    cache.apply_event({"type": "ADDED", "object": ovf})
    cache.apply_event({"type": "ADDED", "object": plain})

    snap_delta = cache.build_snapshot(base)
    snap_full = _full([ovf, plain], nodes)
    rows = {u: i for i, u in enumerate(snap_delta.pod_uids)}
    full_rows = {u: i for i, u in enumerate(snap_full.pod_uids)}
    for uid in ("pod-default-ovf", "pod-default-plain"):
        a = snap_delta.pod_sel_pairs[rows[uid]]
        b = snap_full.pod_sel_pairs[full_rows[uid]]
        np.testing.assert_array_equal(a, b, err_msg=uid)
        assert snap_delta.pod_flags[rows[uid]] == snap_full.pod_flags[full_rows[uid]]
    ovf_row = rows["pod-default-ovf"]
    assert snap_delta.pod_sel_pairs[ovf_row][-1] == SELECTOR_POISON
    assert snap_delta.pod_flags[ovf_row] & PodFlags.SELECTOR_OVERFLOW
    _assert_equivalent_decisions(snap_delta, snap_full)


def test_delta_growth_and_compaction():
    nodes = [KubeNode(make_node_json("agentpool1", 0))]
    base, pools = _base(nodes)
    cache = DeltaSnapshotCache()
    cache.prime(_full([], nodes), {nodes[0].name: 0})
    pods = [make_pod_json(f"g{i}", cpu_milli=10, pending=True) for i in range(300)]
    for p in pods:
        cache.apply_event({"type": "ADDED", "object": p})
    assert cache.size == 300
    for p in pods[:250]:
        cache.apply_event({"type": "DELETED", "object": p})
    assert cache.dead_fraction > 0.5
    assert cache.maybe_compact()
    assert cache.size == 50
    snap = cache.build_snapshot(base)
    snap_full = _full(pods[250:], nodes)
    _assert_equivalent_decisions(snap, snap_full)


def test_delta_checkpoint_resume(tmp_path):
    nodes = [KubeNode(make_node_json("agentpool1", 0))]
    base, pools = _base(nodes)
    cache = DeltaSnapshotCache()
    cache.prime(_full([], nodes), {nodes[0].name: 0})
    pods = [make_pod_json(f"c{i}", cpu_milli=100, pending=True) for i in range(20)]
    for p in pods:
        cache.apply_event({"type": "ADDED", "object": p})
    path = str(tmp_path / "spill.npz")
    cache.save(path)

    fresh = DeltaSnapshotCache()
    fresh.load(path, {nodes[0].name: 0})
    # resume mid-burst: apply more events on the restored cache
    more = [make_pod_json(f"m{i}", cpu_milli=200, pending=True) for i in range(5)]
    for p in more:
        fresh.apply_event({"type": "ADDED", "object": p})
    snap = fresh.build_snapshot(base)
    snap_full = _full(pods + more, nodes)
    _assert_equivalent_decisions(snap, snap_full)


def test_reassign_pdbs_matches_fresh_encode():
    """PDB-set change: the in-place rematch from cached label pairs
    must equal a fresh encode with the new PDBs, only changed rows go
    dirty, and overflow label sets refuse (caller re-primes)."""
    from mi355x_autoscaler.kube.objects import PodDisruptionBudget
    from mi355x_autoscaler.snapshot.encode import MAX_POD_LABEL_PAIRS

    nodes = [KubeNode(make_node_json("agentpool1", 0))]
    pods_json = [
        make_pod_json("a", cpu_milli=100, pending=True, labels={"app": "x"}),
        make_pod_json("b", cpu_milli=100, pending=True,
                      labels={"app": "y", "tier": "db"}),
        make_pod_json("c", cpu_milli=100, pending=True),  # no labels
    ]
    base, pools = _base(nodes)
    cache = DeltaSnapshotCache()
    pdbs0 = [PodDisruptionBudget(name="p0", match_labels={"app": "x"},
                                 disruptions_allowed=1)]
    snap0 = encode_snapshot([KubePod(p) for p in pods_json], nodes, pools, pdbs0)
    cache.prime(snap0, {nodes[0].name: 0}, [dict(p.match_labels) for p in pdbs0])
    cache.drain_dirty()

    # new PDB set: p0 now guards app=y+tier=db; p1 guards app=x
    pdbs1 = [
        PodDisruptionBudget(name="n0", match_labels={"app": "y", "tier": "db"},
                            disruptions_allowed=2),
        PodDisruptionBudget(name="n1", match_labels={"app": "x"},
                            disruptions_allowed=0),
    ]
    n = cache.reassign_pdbs([dict(p.match_labels) for p in pdbs1])
    assert n == 2  # rows a (0 -> 1) and b (-1 -> 0) changed
    fresh = encode_snapshot([KubePod(p) for p in pods_json], nodes, pools, pdbs1)
    rows = {u: i for i, u in enumerate(cache.pod_uids[: cache.size])}
    for i, u in enumerate(fresh.pod_uids):
        assert cache.arrays["pod_pdb"][rows[u]] == fresh.pod_pdb[i], u
    dirty = set(cache.drain_dirty().tolist())
    assert len(dirty) == 2

    # label overflow: >MAX_POD_LABEL_PAIRS labels on a live pod
    wide = make_pod_json("wide", cpu_milli=50, pending=True,
                         labels={f"l{i:02d}": "v"
                                 for i in range(MAX_POD_LABEL_PAIRS + 1)})
    cache.apply_event({"type": "ADDED", "object": wide})
    assert cache.reassign_pdbs([{"l00": "v"}]) is None


def test_reassign_pdbs_refuses_without_label_data():
    from mi355x_autoscaler.snapshot.delta import DeltaSnapshotCache
    from mi355x_autoscaler.sim.synth_soa import synth_burst_snapshot

    snap = synth_burst_snapshot(100, n_nodes_per_pool=(2,), seed=0)
    cache = DeltaSnapshotCache()
    cache.prime(snap, {})
    assert not cache.has_label_pairs
    assert cache.reassign_pdbs([{"app": "x"}]) is None


def test_randomized_delta_fuzz():
    """Randomized watch-event/PDB-change/compaction sequences converge
    to the full re-encode, column-exact (scripts/fuzz_delta_cpu.py
    runs the longer form; 1000 iterations green at commit time)."""
    import importlib.util
    import os

    spec = importlib.util.spec_from_file_location(
        "fuzz_delta_cpu", os.path.join(os.path.dirname(__file__), "..",
                                       "scripts", "fuzz_delta_cpu.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    mod.main(iters=8)


def test_apply_batch_without_label_columns_disables_reassign():
    """A decoded batch lacking the label columns (mismatched compiled
    codec) must clear has_label_pairs so reassign_pdbs falls back to
    the re-prime instead of trusting stale label hashes."""
    import numpy as np

    from mi355x_autoscaler.snapshot.encode import (
        MAX_POD_LABEL_PAIRS, MAX_SELECTOR_PAIRS, NUM_RESOURCES)

    nodes = [KubeNode(make_node_json("agentpool1", 0))]
    base, pools = _base(nodes)
    pods_json = [make_pod_json("a", cpu_milli=100, pending=True,
                               labels={"app": "x"})]
    snap = encode_snapshot([KubePod(p) for p in pods_json], nodes, pools)
    cache = DeltaSnapshotCache()
    cache.prime(snap, {nodes[0].name: 0})
    assert cache.has_label_pairs

    n = 1
    decoded = {
        "types": np.zeros(n, np.int32),
        "pod_uids": ["uid-batch-a"],
        "pod_refs": [("default", "batch-a")],
        "pod_req": np.zeros((n, NUM_RESOURCES), np.int32),
        "pod_flags": np.zeros(n, np.int32),
        "pod_sel_hash": np.zeros(n, np.int64),
        "pod_sel_pairs": np.zeros((n, MAX_SELECTOR_PAIRS), np.int64),
        "pod_sel_count": np.zeros(n, np.int32),
        "pod_node": np.full(n, -1, np.int32),
        "pod_pdb": np.full(n, -1, np.int32),
        # NOTE: no pod_label_pairs / pod_label_count columns
    }
    cache.apply_batch(decoded)
    assert not cache.has_label_pairs
    assert cache.reassign_pdbs([{"app": "x"}]) is None
