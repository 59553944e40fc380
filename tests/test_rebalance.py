"""Shard rebalancing (row H): skewed shards even out over all_to_all
and the global decision is unchanged (ownership-invariance)."""

import multiprocessing as mp
import os
import pickle

import numpy as np
import pytest

from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.sim.engine import CpuEngine
from mi355x_autoscaler.sim.synth import burst_trace, wrap
from mi355x_autoscaler.snapshot.encode import encode_snapshot

WORLD = 2


def _snap_for_rank(rank, skewed=True, world=WORLD):
    pods_json, nodes_json = burst_trace(400, n_nodes_per_pool=(4,), seed=8,
                                        bound_pods_per_node=2)
    if skewed:
        # rank 0 gets ~everything; the tail splits over the others
        cut = int(len(pods_json) * 0.9)
        if rank == 0:
            pods_json = pods_json[:cut]
        else:
            pods_json = pods_json[cut:][rank - 1 :: world - 1]
    else:
        pods_json = pods_json[rank::world]
    pods, nodes = wrap(pods_json, nodes_json)
    return encode_snapshot(pods, nodes, group_nodes_into_pools(nodes))


def _worker(rank, q, world=WORLD, port="29631"):
    os.environ.update(WORLD_SIZE=str(world), RANK=str(rank), LOCAL_RANK=str(rank),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=port)
    import torch.distributed as dist

    from mi355x_autoscaler.parallel.engine_sharded import ShardedEngine
    from mi355x_autoscaler.parallel.rebalance import rebalance_shard

    try:
        engine = ShardedEngine(CpuEngine())
        snap = _snap_for_rank(rank, world=world)
        before = snap.num_pods
        d_before = engine.decide(snap)
        snap2, did = rebalance_shard(snap, skew_threshold=1.2)
        d_after = engine.decide(snap2)
        q.put((rank, pickle.dumps(dict(
            before=before, after=snap2.num_pods, did=did,
            units_b=d_before.units_needed, units_a=d_after.units_needed,
            pe_b=d_before.pods_placed_existing, pe_a=d_after.pods_placed_existing,
        ))))
    finally:
        dist.destroy_process_group()


def _run_world(world, port):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, q, world, port))
             for r in range(world)]
    for p in procs:
        p.start()
    res = {}
    for _ in range(world):
        rank, blob = q.get(timeout=180)
        res[rank] = pickle.loads(blob)
    for p in procs:
        p.join(timeout=30)
        assert p.exitcode == 0
    return res


def test_rebalance_evens_shards_and_preserves_decision():
    res = _run_world(WORLD, "29631")
    assert res[0]["did"] and res[1]["did"]
    total = res[0]["before"] + res[1]["before"]
    assert res[0]["after"] + res[1]["after"] == total
    assert abs(res[0]["after"] - res[1]["after"]) <= 1  # balanced
    # ownership-invariance of the global decision
    assert res[0]["units_b"] == res[0]["units_a"] == res[1]["units_a"]
    assert res[0]["pe_b"] == res[0]["pe_a"]


def test_rebalance_world4():
    """Four-rank all_to_all redistribution: balanced shards, conserved
    pod count, decision unchanged on every rank."""
    res = _run_world(4, "29673")
    total = sum(res[r]["before"] for r in range(4))
    after = [res[r]["after"] for r in range(4)]
    assert sum(after) == total
    assert max(after) - min(after) <= 2
    for r in range(4):
        assert res[r]["units_a"] == res[0]["units_b"]


def test_rebalance_noop_when_balanced():
    from mi355x_autoscaler.parallel.rebalance import rebalance_shard

    snap = _snap_for_rank(0, skewed=False)
    out, did = rebalance_shard(snap)  # world == 1 here
    assert not did and out is snap


def test_pack_rows_roundtrip_wide_selectors():
    """Regression: pack_rows once hardcoded 8-pair-era column offsets,
    so sel_count/node/pdb stomped selector pairs 8-10 on the wire.
    Round-trip a pod with 12 selector pairs and assert every column."""
    from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
    from mi355x_autoscaler.models.kube import KubeNode, KubePod
    from mi355x_autoscaler.parallel.rebalance import pack_rows, unpack_rows
    from mi355x_autoscaler.sim.synth import make_node_json, make_pod_json
    from mi355x_autoscaler.snapshot.encode import encode_snapshot

    sel = {f"k{i:02d}": f"v{i}" for i in range(12)}
    node = make_node_json("agentpool1", 0)
    node["metadata"]["labels"].update(sel)
    nodes = [KubeNode(node)]
    pods = [KubePod(make_pod_json("w", cpu_milli=100, pending=True,
                                  node_selector=sel))]
    snap = encode_snapshot(pods, nodes, group_nodes_into_pools(nodes))
    cols = unpack_rows(pack_rows(snap, np.arange(1)))
    np.testing.assert_array_equal(cols["pod_sel_pairs"], snap.pod_sel_pairs)
    assert cols["pod_sel_count"][0] == 12
    np.testing.assert_array_equal(cols["pod_req"], snap.pod_req)
    assert cols["pod_node"][0] == snap.pod_node[0]
    assert cols["pod_pdb"][0] == snap.pod_pdb[0]
