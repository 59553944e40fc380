"""Sharded CONTROL LOOP e2e (gloo world 2): every rank lists the full
cluster from the shared fake API, self-shards by uid-hash, agrees on
the decision, and only rank 0 actuates."""

import multiprocessing as mp
import os
import pickle

import pytest

from mi355x_autoscaler.sim.synth import burst_trace, make_node_json, make_pod_json

WORLD = 2


def _server_proc(q_url, q_done):
    from mi355x_autoscaler.kube.fake_api import FakeKubeServer

    pods, nodes = burst_trace(120, n_nodes_per_pool=(3,), seed=0)
    # one idle node beyond the busy ones for the drain path
    with FakeKubeServer() as srv:
        srv.state.load(pods, nodes)
        q_url.put(srv.base_url)
        q_done.get(timeout=120)  # hold the server until ranks finish


def _rank_proc(rank, url, q_out):
    os.environ.update(WORLD_SIZE=str(WORLD), RANK=str(rank), LOCAL_RANK=str(rank),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT="29661")
    import torch.distributed as dist

    from mi355x_autoscaler.cloud.fake import FakeCloudProvider
    from mi355x_autoscaler.ctl.loop import Cluster, ClusterOptions
    from mi355x_autoscaler.kube.client import KubeApiClient
    from mi355x_autoscaler.parallel.engine_sharded import ShardedEngine
    from mi355x_autoscaler.sim.engine import CpuEngine

    kube = KubeApiClient(url)
    provider = FakeCloudProvider(initial_pool_sizes={"agentpool1": 3})
    opts = ClusterOptions(sleep_seconds=0, idle_threshold_seconds=0, spare_agents=0)
    cluster = Cluster(kube, provider, ShardedEngine(CpuEngine()), options=opts)
    try:
        res = cluster.loop_logic()
        q_out.put((rank, pickle.dumps(dict(
            units=res.decision.units_needed,
            placed=res.decision.pods_placed_existing,
            pods_total=res.pods_total,
            events=[e.kind for e in provider.events],
            pool_sizes=provider.pool_sizes,
        ))))
    finally:
        dist.destroy_process_group()


def test_sharded_loop_rank0_actuates():
    ctx = mp.get_context("spawn")
    q_url, q_done, q_out = ctx.Queue(), ctx.Queue(), ctx.Queue()
    server = ctx.Process(target=_server_proc, args=(q_url, q_done))
    server.start()
    url = q_url.get(timeout=60)
    ranks = [ctx.Process(target=_rank_proc, args=(r, url, q_out)) for r in range(WORLD)]
    for p in ranks:
        p.start()
    results = {}
    try:
        for _ in range(WORLD):
            rank, blob = q_out.get(timeout=180)
            results[rank] = pickle.loads(blob)
    finally:
        q_done.put(True)
        for p in ranks:
            p.join(timeout=30)
        server.join(timeout=30)
    assert all(p.exitcode == 0 for p in ranks)

    # identical decision over identical FULL ingest (self-sharded)
    assert results[0]["pods_total"] == results[1]["pods_total"] == 120
    assert results[0]["units"] == results[1]["units"]
    assert results[0]["placed"] == results[1]["placed"]

    # single-rank actuation
    assert results[0]["events"], "rank 0 must actuate"
    assert results[1]["events"] == [], "rank 1 must NOT actuate"
    assert results[0]["pool_sizes"]["agentpool1"] > 3
    assert results[1]["pool_sizes"]["agentpool1"] == 3

    # the sharded decision equals the single-process one
    from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
    from mi355x_autoscaler.sim.engine import CpuEngine
    from mi355x_autoscaler.sim.synth import burst_trace as _bt
    from mi355x_autoscaler.sim.synth import wrap
    from mi355x_autoscaler.snapshot.encode import encode_snapshot

    pods, nodes = _bt(120, n_nodes_per_pool=(3,), seed=0)
    kp, kn = wrap(pods, nodes)
    snap = encode_snapshot(kp, kn, group_nodes_into_pools(kn))
    d = CpuEngine().decide(snap)
    assert results[0]["units"] == d.units_needed
    assert results[0]["placed"] == d.pods_placed_existing
