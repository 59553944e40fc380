"""Randomized differential fuzz of the watch delta cache (CPU-only):
random event sequences — adds, modifies, deletes, re-adds, node-set
remaps, PDB-set changes (in-place reassign), compactions — applied to
DeltaSnapshotCache must converge to the same decisions AND the same
pod columns as a fresh full encode of the surviving pod set.

Run directly (ITERS from argv) or via tests/test_delta.py's wrapper.
"""
import random
import sys

import numpy as np

def _argv_iters():
    try:
        return int(sys.argv[1])
    except (IndexError, ValueError):  # imported under pytest etc.
        return 30


ITERS = _argv_iters()

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from mi355x_autoscaler.kube.objects import PodDisruptionBudget
from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.models.kube import KubeNode, KubePod
from mi355x_autoscaler.sim.engine import CpuEngine
from mi355x_autoscaler.sim.synth import make_node_json, make_pod_json
from mi355x_autoscaler.snapshot.delta import DeltaSnapshotCache
from mi355x_autoscaler.snapshot.encode import encode_snapshot

CMP_COLS = ("pod_req", "pod_flags", "pod_sel_hash", "pod_sel_pairs",
            "pod_sel_count", "pod_node", "pod_pdb",
            "pod_label_pairs", "pod_label_count")


def rand_pod(rng, i, node_names):
    sel = None
    r = rng.random()
    if r < 0.15:
        sel = {"accelerator": "amd-mi355x"}
    elif r < 0.2:
        sel = {f"k{j}": "v" for j in range(rng.randint(1, 34))}
    labels = None
    if rng.random() < 0.5:
        labels = {"app": rng.choice(["a", "b", "guarded"]),
                  "tier": rng.choice(["web", "db"])}
        if rng.random() < 0.05:
            labels.update({f"l{j:02d}": "v" for j in range(18)})  # overflow
    node = rng.choice(node_names + [None, None])
    return make_pod_json(
        f"f{i}", namespace=rng.choice(["default", "default", "team-a"]),
        cpu_milli=rng.choice([0, 100, 900]),
        mem_mib=rng.choice([0, 128, 2048]),
        gpu=rng.choice([0, 0, 1]),
        pending=node is None, node_name=node,
        node_selector=sel, labels=labels,
        owner_kind=rng.choice(["ReplicaSet", "ReplicaSet", None, "DaemonSet"]),
    )


def pdb_menu(rng):
    out = []
    for name, sel in (("g", {"app": "guarded"}), ("w", {"tier": "web"}),
                      ("ab", {"app": "a", "tier": "db"})):
        if rng.random() < 0.6:
            out.append(PodDisruptionBudget(
                name=name, match_labels=sel,
                namespace=rng.choice(["default", "team-a"]),
                disruptions_allowed=rng.randint(0, 2)))
    return out


def check(cache, base, live_json, nodes, pools, pdbs, ctx):
    snap = cache.build_snapshot(base)
    fresh = encode_snapshot([KubePod(p) for p in live_json], nodes, pools, pdbs)
    rows = {u: i for i, u in enumerate(snap.pod_uids) if u}
    frows = {u: i for i, u in enumerate(fresh.pod_uids)}
    assert set(rows) == set(frows), ctx
    for uid, i in rows.items():
        j = frows[uid]
        for col in CMP_COLS:
            a, b = getattr(snap, col)[i], getattr(fresh, col)[j]
            np.testing.assert_array_equal(a, b, err_msg=f"{ctx} {uid} {col}")
    da, db = CpuEngine().decide(snap), CpuEngine().decide(fresh)
    assert da.units_needed == db.units_needed, ctx
    assert da.pods_unfulfillable == db.pods_unfulfillable, ctx
    ra, rb = CpuEngine().maintain(snap), CpuEngine().maintain(fresh)
    np.testing.assert_array_equal(ra.node_util_permille, rb.node_util_permille,
                                  err_msg=ctx)
    np.testing.assert_array_equal(ra.node_pdb_ok, rb.node_pdb_ok, err_msg=ctx)


def main(iters=None):
    rng = random.Random(777)
    n_iters = ITERS if iters is None else iters
    for it in range(n_iters):
        nodes_json = [make_node_json("agentpool1", i) for i in range(rng.randint(1, 5))]
        nodes_json += [make_node_json("gpupool", i, gpu=8,
                                      instance_type="Standard_ND96_MI355X_v1")
                       for i in range(rng.randint(0, 2))]
        nodes = [KubeNode(n) for n in nodes_json]
        node_names = [n.name for n in nodes]
        pools = group_nodes_into_pools(nodes)
        pdbs = pdb_menu(rng)

        live = {}
        initial = [rand_pod(rng, i, node_names) for i in range(rng.randint(0, 40))]
        for p in initial:
            live[p["metadata"]["uid"]] = p
        snap0 = encode_snapshot([KubePod(p) for p in initial], nodes, pools, pdbs)
        cache = DeltaSnapshotCache()
        cache.prime(snap0, {n: i for i, n in enumerate(node_names)},
                    [(p.namespace, dict(p.match_labels)) for p in pdbs])

        next_id = len(initial)
        for step in range(rng.randint(1, 6)):
            ctx = f"iter {it} step {step}"
            op = rng.random()
            if op < 0.55:  # event batch
                for _ in range(rng.randint(1, 25)):
                    e = rng.random()
                    if e < 0.5 or not live:
                        p = rand_pod(rng, next_id, node_names)
                        next_id += 1
                        live[p["metadata"]["uid"]] = p
                        cache.apply_event({"type": "ADDED", "object": p})
                    elif e < 0.75:
                        uid = rng.choice(list(live))
                        name = live[uid]["metadata"]["name"]
                        p = rand_pod(rng, 0, node_names)
                        p["metadata"]["name"] = name
                        p["metadata"]["uid"] = uid
                        live[uid] = p
                        cache.apply_event({"type": "MODIFIED", "object": p})
                    else:
                        uid = rng.choice(list(live))
                        cache.apply_event({"type": "DELETED", "object": live.pop(uid)})
            elif op < 0.75:  # PDB-set change
                pdbs = pdb_menu(rng)
                n = cache.reassign_pdbs([(p.namespace, dict(p.match_labels)) for p in pdbs])
                if n is None:  # overflow rows: re-prime like the loop does
                    snapn = encode_snapshot([KubePod(p) for p in live.values()],
                                            nodes, pools, pdbs)
                    cache = DeltaSnapshotCache()
                    cache.prime(snapn, {n2: i for i, n2 in enumerate(node_names)},
                                [(p.namespace, dict(p.match_labels)) for p in pdbs])
            else:
                cache.maybe_compact()
            base = encode_snapshot([], nodes, pools, pdbs)
            check(cache, base, list(live.values()), nodes, pools, pdbs, ctx)
    print(f"delta fuzz ok: {n_iters} iterations")


if __name__ == "__main__":
    main()
