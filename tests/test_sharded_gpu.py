"""Sharded GPU engine on HARDWARE (world 2, gloo, both ranks on
cuda:0): the ds-packed and resident sharded decide paths run real HIP
kernels per rank with cross-rank row merge + agreement — the closest
1-GPU-box approximation of the driver's 8-GPU run (the nccl device
merge swaps in by backend switch only; its row protocol is covered by
test_decide_rows_device_* in test_gpu_engine.py)."""

import multiprocessing as mp
import os
import pickle

import pytest

from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.sim.engine import CpuEngine
from mi355x_autoscaler.sim.synth import burst_trace, make_pod_json, wrap
from mi355x_autoscaler.snapshot.encode import encode_snapshot

pytestmark = pytest.mark.gpu

WORLD = 2


def _trace():
    return burst_trace(
        4000, n_nodes_per_pool=(20, 10),
        pools=(("agentpool1", "Standard_D2_v2"),
               ("gpupool", "Standard_ND96_MI355X_v1")),
        gpu_fraction=0.4, seed=77, bound_pods_per_node=3)


def _full_snap():
    pods_json, nodes_json = _trace()
    pods, nodes = wrap(pods_json, nodes_json)
    return encode_snapshot(pods, nodes, group_nodes_into_pools(nodes))


def _rank(rank, q, mode):
    os.environ.update(WORLD_SIZE=str(WORLD), RANK=str(rank),
                      LOCAL_RANK="0",  # both ranks share the one GPU
                      MASTER_ADDR="127.0.0.1", MASTER_PORT="29715")
    import torch.distributed as dist

    from mi355x_autoscaler.ops.engine_gpu import GpuEngine
    from mi355x_autoscaler.parallel.elastic import shard_snapshot
    from mi355x_autoscaler.parallel.engine_sharded import ShardedEngine
    from mi355x_autoscaler.snapshot.delta import DeltaSnapshotCache

    # explicit gloo: both ranks share ONE GPU, which RCCL refuses
    # ("Duplicate GPU detected") — init before ShardedEngine so
    # init_from_env sees the group and keeps it
    dist.init_process_group(backend="gloo")
    try:
        snap = _full_snap()
        shard = shard_snapshot(snap, rank, WORLD)
        engine = ShardedEngine(GpuEngine("cuda:0"), check_agreement=True)
        if mode == "decide":
            d = engine.decide(shard)
        else:  # resident: per-rank shard cache + churn
            pods_json, nodes_json = _trace()
            pods, nodes = wrap(pods_json, nodes_json)
            nodes_idx = {n.name: i for i, n in enumerate(nodes)}
            pools = group_nodes_into_pools(nodes)
            base = encode_snapshot([], nodes, pools)
            cache = DeltaSnapshotCache()
            cache.set_shard(rank, WORLD)
            cache.prime(shard, nodes_idx)
            for i in range(60):  # same stream on every rank; self-filtered
                cache.apply_event({"type": "ADDED",
                                   "object": make_pod_json(
                                       f"late{i}", cpu_milli=300,
                                       pending=True)})
            d = engine.decide_resident(cache, base)
        rep = engine.maintain(shard)
        q.put((rank, pickle.dumps(dict(
            units=d.units_needed, pe=d.pods_placed_existing,
            pn=d.pods_placed_new, un=d.pods_unfulfillable,
            cands=[c.node_name for c in rep.candidates]))))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def _run(mode):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_rank, args=(r, q, mode)) for r in range(WORLD)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(WORLD):
        rank, blob = q.get(timeout=240)
        results[rank] = pickle.loads(blob)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    return results


def test_sharded_gpu_decide_matches_single_process():
    results = _run("decide")
    assert results[0] == results[1]
    d = CpuEngine().decide(_full_snap())
    assert results[0]["units"] == d.units_needed
    assert results[0]["pe"] == d.pods_placed_existing
    assert results[0]["un"] == d.pods_unfulfillable
    rep = CpuEngine().maintain(_full_snap())
    assert results[0]["cands"] == [c.node_name for c in rep.candidates]


def test_sharded_gpu_resident_decide_matches_full():
    results = _run("resident")
    assert results[0] == results[1]
    # reference: full churned pod set, single process
    pods_json, nodes_json = _trace()
    extra = [make_pod_json(f"late{i}", cpu_milli=300, pending=True)
             for i in range(60)]
    pods, nodes = wrap(pods_json + extra, nodes_json)
    ref = CpuEngine().decide(
        encode_snapshot(pods, nodes, group_nodes_into_pools(nodes)))
    assert results[0]["units"] == ref.units_needed
    assert results[0]["pe"] == ref.pods_placed_existing
    assert results[0]["un"] == ref.pods_unfulfillable


def _rccl1_worker(q):
    """Child process: world-1 RCCL communicator + the full sharded
    decide surface (device merge, free hook, agreement check) — the
    exact collective API calls the 8-GPU driver run makes, executed
    cold on real RCCL."""
    os.environ.update(WORLD_SIZE="1", RANK="0", LOCAL_RANK="0",
                      MASTER_ADDR="127.0.0.1", MASTER_PORT="29725")
    import torch
    import torch.distributed as dist

    from mi355x_autoscaler.ops.engine_gpu import GpuEngine
    from mi355x_autoscaler.parallel import dist as pdist
    from mi355x_autoscaler.parallel.engine_sharded import ShardedEngine
    from mi355x_autoscaler.snapshot.delta import DeltaSnapshotCache

    try:
        dist.init_process_group(backend="nccl", world_size=1, rank=0)
        torch.cuda.set_device(0)
        # raw RCCL smoke: the world-1 fast paths in parallel.dist skip
        # collectives, so touch the library directly — communicator
        # init + the exact op/dtype shapes the sharded decide uses
        rows = torch.ones((2048, 40), dtype=torch.int64, device="cuda:0")
        dist.all_reduce(rows[0])
        outs = [torch.zeros_like(rows)]
        dist.all_gather(outs, rows)
        assert bool((outs[0] == rows).all())
        vec = torch.arange(8, dtype=torch.int64, device="cuda:0")
        dist.all_reduce(vec, op=dist.ReduceOp.MIN)
        dist.all_reduce(vec, op=dist.ReduceOp.MAX)
        dist.barrier()
        snap = _full_snap()
        engine = ShardedEngine(GpuEngine("cuda:0"), check_agreement=True)
        assert engine.active and pdist.backend() == "nccl"
        d1 = engine.decide(snap)          # device merge branch (nccl)
        rep = engine.maintain(snap)       # device all-reduce partials
        ds = engine.inner.packer.pack(snap, engine.inner.device)
        d2 = engine.decide_packed(ds, snap.pool_names)
        pods_json, nodes_json = _trace()
        pods, nodes = wrap(pods_json, nodes_json)
        pools = group_nodes_into_pools(nodes)
        base = encode_snapshot([], nodes, pools)
        cache = DeltaSnapshotCache()
        cache.prime(snap, {n.name: i for i, n in enumerate(nodes)})
        d3 = engine.decide_resident(cache, base)
        q.put(pickle.dumps(dict(
            u1=d1.units_needed, u2=d2.units_needed, u3=d3.units_needed,
            pe=d1.pods_placed_existing, un=d1.pods_unfulfillable,
            cands=[c.node_name for c in rep.candidates])))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_rccl_world1_collective_paths():
    """World-1 RCCL: communicator init + fixed-size device row gather
    + usage all-reduce + MIN/MAX agreement all execute on the real
    RCCL library (single-rank no-ops wire-wise, full API coverage),
    and every entry point agrees with the CPU oracle."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    p = ctx.Process(target=_rccl1_worker, args=(q,))
    p.start()
    blob = q.get(timeout=240)
    p.join(timeout=60)
    assert p.exitcode == 0
    res = pickle.loads(blob)
    ref = CpuEngine().decide(_full_snap())
    assert res["u1"] == ref.units_needed
    assert res["u2"] == ref.units_needed
    assert res["u3"] == ref.units_needed
    assert res["pe"] == ref.pods_placed_existing
    assert res["un"] == ref.pods_unfulfillable


def _loop_server(q_url, q_go, q_done):
    from mi355x_autoscaler.kube.fake_api import FakeKubeServer

    pods_json, nodes_json = _trace()
    with FakeKubeServer() as srv:
        srv.state.load(pods_json, nodes_json)
        q_url.put(srv.base_url)
        q_go.get(timeout=180)  # both ranks primed
        for i in range(40):
            srv.state.add_pod(make_pod_json(f"wave{i}", cpu_milli=400,
                                            pending=True))
        q_done.get(timeout=300)


def _loop_rank(rank, url, q_out, q_prime):
    os.environ.update(WORLD_SIZE=str(WORLD), RANK=str(rank),
                      LOCAL_RANK="0", MASTER_ADDR="127.0.0.1",
                      MASTER_PORT="29735")
    import time

    import torch.distributed as dist

    from mi355x_autoscaler.cloud.fake import FakeCloudProvider
    from mi355x_autoscaler.ctl.loop import Cluster, ClusterOptions
    from mi355x_autoscaler.kube.client import KubeApiClient
    from mi355x_autoscaler.ops.engine_gpu import GpuEngine
    from mi355x_autoscaler.parallel.engine_sharded import ShardedEngine

    dist.init_process_group(backend="gloo")  # one shared GPU: no RCCL
    try:
        kube = KubeApiClient(url)
        provider = FakeCloudProvider(initial_pool_sizes={"agentpool1": 20})
        opts = ClusterOptions(sleep_seconds=0, idle_threshold_seconds=10**9,
                              spare_agents=0, watch=True)
        cluster = Cluster(kube, provider,
                          ShardedEngine(GpuEngine("cuda:0")), options=opts)
        res1 = cluster.loop_logic()  # primes watch + shard caches
        assert cluster._watch_state.get("shard_cache") is not None
        q_prime.put(rank)
        dist.barrier()
        time.sleep(1.0)  # churn wave arrives via watch
        res2 = cluster.loop_logic()  # RESIDENT sharded decide on GPU
        q_out.put((rank, pickle.dumps(dict(
            u1=res1.decision.units_needed, u2=res2.decision.units_needed,
            p2=res2.decision.pods_placed_existing,
            total2=res2.pods_total,
            events=[e.kind for e in provider.events]))))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_sharded_gpu_watch_loop_world2():
    """Capstone: the FULL watch-mode control loop at world 2 with real
    GPU engines on hardware — prime, churn via watch, per-rank
    resident shard decide, identical decisions, rank-0-only
    actuation. gloo carries the collectives (one shared GPU); the
    8-GPU driver run swaps in RCCL by backend only."""
    ctx = mp.get_context("spawn")
    q_url, q_go, q_done, q_out, q_prime = (ctx.Queue() for _ in range(5))
    server = ctx.Process(target=_loop_server, args=(q_url, q_go, q_done))
    server.start()
    url = q_url.get(timeout=120)
    ranks = [ctx.Process(target=_loop_rank, args=(r, url, q_out, q_prime))
             for r in range(WORLD)]
    for p in ranks:
        p.start()
    results = {}
    try:
        for _ in range(WORLD):
            q_prime.get(timeout=240)
        q_go.put(True)
        for _ in range(WORLD):
            rank, blob = q_out.get(timeout=240)
            results[rank] = pickle.loads(blob)
    finally:
        q_done.put(True)
        for p in ranks:
            p.join(timeout=60)
        server.join(timeout=60)
    assert all(p.exitcode == 0 for p in ranks), [p.exitcode for p in ranks]
    assert results[0]["u1"] == results[1]["u1"]
    assert results[0]["u2"] == results[1]["u2"]
    assert results[0]["p2"] == results[1]["p2"]
    # 4000 pending + 30 nodes x 3 bound + 40-wave churn
    assert results[0]["total2"] == 4000 + 90 + 40
    assert results[0]["events"], "rank 0 must actuate"
    assert results[1]["events"] == []
