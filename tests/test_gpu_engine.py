"""GPU kernel tests: every HIP kernel and the whole GpuEngine pipeline
against the CPU oracle, bit-for-bit (SURVEY §4 plan items 1, 4)."""

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

torch = pytest.importorskip("torch")

from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.sim.engine import CpuEngine
from mi355x_autoscaler.sim.oracle import OracleSimulator, build_segments
from mi355x_autoscaler.sim.synth import burst_trace, make_node_json, make_pod_json, scaledown_trace, wrap
from mi355x_autoscaler.snapshot.encode import encode_snapshot


@pytest.fixture(scope="module")
def engine():
    from mi355x_autoscaler.ops.engine_gpu import GpuEngine

    return GpuEngine("cuda:0")


def _snap(pods_json, nodes_json, pdbs=None, max_size=100):
    pods, nodes = wrap(pods_json, nodes_json)
    pools = group_nodes_into_pools(nodes, max_size=max_size)
    return encode_snapshot(pods, nodes, pools, pdbs)


def _assert_decisions_equal(d_gpu, d_cpu):
    assert d_gpu.units_needed == d_cpu.units_needed
    assert d_gpu.new_pool_sizes == d_cpu.new_pool_sizes
    assert d_gpu.pods_placed_existing == d_cpu.pods_placed_existing
    assert d_gpu.pods_placed_new == d_cpu.pods_placed_new
    assert d_gpu.pods_unfulfillable == d_cpu.pods_unfulfillable
    np.testing.assert_array_equal(
        d_gpu.pool_residual_totals, d_cpu.pool_residual_totals
    )


def test_native_extension_is_intree(engine):
    import mi355x_autoscaler._hip_ops as ext

    assert "mi355x_autoscaler" in ext.__file__
    assert "site-packages" not in ext.__file__


@pytest.mark.parametrize("seed", [0, 1, 2])
@pytest.mark.parametrize(
    "n_pods,pools,sizes,gpufrac",
    [
        (64, (("agentpool1", "Standard_D2_v2"),), (3,), 0.0),
        (1000, (("gpupool", "Standard_ND96_MI355X_v1"),), (4,), 1.0),
        (
            5000,
            (("agentpool1", "Standard_D2_v2"), ("bigpool", "Standard_D4_v2"),
             ("gpupool", "Standard_ND96_MI355X_v1")),
            (40, 30, 30),
            0.35,
        ),
    ],
)
def test_decide_matches_oracle(engine, seed, n_pods, pools, sizes, gpufrac):
    pods_json, nodes_json = burst_trace(
        n_pods, n_nodes_per_pool=sizes, pools=pools, gpu_fraction=gpufrac,
        seed=seed, bound_pods_per_node=3,
    )
    snap = _snap(pods_json, nodes_json)
    _assert_decisions_equal(engine.decide(snap), CpuEngine().decide(snap))


def test_decide_boundary_cases(engine):
    # zero-request pods, exactly-full nodes, selector mismatch, pool at max
    node = make_node_json("agentpool1", 0, cpu_milli=2000, mem_mib=4096, pods=5)
    filler = make_pod_json("full", cpu_milli=2000, mem_mib=4096,
                           node_name=node["metadata"]["name"])
    pods = [filler]
    pods += [make_pod_json(f"z{i}", cpu_milli=0, mem_mib=0, pending=True) for i in range(7)]
    pods += [make_pod_json("sel", node_selector={"disk": "ssd"}, pending=True)]
    pods += [make_pod_json(f"c{i}", cpu_milli=1000, mem_mib=512, pending=True) for i in range(9)]
    snap = _snap(pods, [node], max_size=3)
    _assert_decisions_equal(engine.decide(snap), CpuEngine().decide(snap))


def test_decide_empty_pending(engine):
    pods_json, nodes_json = burst_trace(0, n_nodes_per_pool=(3,), seed=0, bound_pods_per_node=2)
    snap = _snap(pods_json, nodes_json)
    d = engine.decide(snap)
    assert not d.wants_scale_up()
    assert d.pods_placed_existing == 0


def test_over_provision_matches(engine):
    pods_json, nodes_json = burst_trace(300, n_nodes_per_pool=(2,), seed=4)
    snap = _snap(pods_json, nodes_json)
    d_gpu = engine.decide(snap, over_provision=2)
    d_cpu = CpuEngine().decide(snap, over_provision=2)
    _assert_decisions_equal(d_gpu, d_cpu)


@pytest.mark.parametrize("n_nodes", [10, 100, 500])
def test_maintain_matches_oracle(engine, n_nodes):
    pods_json, nodes_json, pdbs = scaledown_trace(n_nodes=n_nodes, seed=2)
    snap = _snap(pods_json, nodes_json, pdbs)
    rep_gpu = engine.maintain(snap)
    rep_cpu = CpuEngine().maintain(snap)
    np.testing.assert_array_equal(rep_gpu.node_util_permille, rep_cpu.node_util_permille)
    np.testing.assert_array_equal(rep_gpu.node_blocked, rep_cpu.node_blocked)
    np.testing.assert_array_equal(rep_gpu.node_pdb_ok, rep_cpu.node_pdb_ok)
    np.testing.assert_array_equal(rep_gpu.node_idle, rep_cpu.node_idle)
    np.testing.assert_array_equal(rep_gpu.node_drainable_pods, rep_cpu.node_drainable_pods)
    assert [c.node_name for c in rep_gpu.candidates] == [c.node_name for c in rep_cpu.candidates]


def test_segsum_mfma_exactness():
    """MFMA segmented sums vs numpy over adversarial shapes: max-value
    elements (limb-exactness bound), ragged segments, empty segments."""
    from mi355x_autoscaler.ops import load_ops
    from mi355x_autoscaler.ops.engine_gpu import segsum

    ops = load_ops()
    rng = np.random.default_rng(0)
    for M, S in [(1, 1), (64, 3), (1000, 17), (100000, 999), (300000, 4)]:
        seg = np.sort(rng.integers(0, S, M)).astype(np.int64)
        vals = rng.integers(0, 2**31 - 1, size=(M, 4), dtype=np.int64).astype(np.int32)
        # adversarial: a run of max-value elements in one segment
        vals[: min(M, 300)] = 2**31 - 1
        seg[: min(M, 300)] = 0
        seg = np.sort(seg)
        ref = np.zeros((S, 4), np.int64)
        np.add.at(ref, seg, vals.astype(np.int64))
        v = torch.from_numpy(vals).cuda()
        s = torch.from_numpy(seg).cuda()
        out = segsum(v, s, S, ops)
        np.testing.assert_array_equal(out.cpu().numpy(), ref)


def test_fit_eligibility_kernel_direct():
    """Kernel A vs the host predicate on a label-heavy node set."""
    from mi355x_autoscaler.ops import load_ops
    from mi355x_autoscaler.ops.engine_gpu import build_segments_device, upload_snapshot
    from mi355x_autoscaler.snapshot.encode import selector_subset_match

    ops = load_ops()
    pods_json, nodes_json = burst_trace(
        500, n_nodes_per_pool=(77, 33),
        pools=(("agentpool1", "Standard_D2_v2"), ("gpupool", "Standard_ND96_MI355X_v1")),
        gpu_fraction=0.5, seed=9,
    )
    # mark some nodes unschedulable
    for nd in nodes_json[::7]:
        nd["spec"]["unschedulable"] = True
    snap = _snap(pods_json, nodes_json)
    ds = upload_snapshot(snap, torch.device("cuda:0"))
    segs = build_segments_device(ds, ops)
    G, N = segs.G, snap.num_nodes
    W = (N + 63) // 64
    elig = torch.zeros((G, W), dtype=torch.int64, device="cuda:0")
    g_dev = torch.full((1,), G, dtype=torch.int32, device="cuda:0")
    ops.fit_eligibility(ds.node_labels, ds.node_label_count, ds.node_unsched,
                        segs.seg_sel_pairs.contiguous(), segs.seg_sel_count.contiguous(),
                        g_dev, elig)
    got = elig.cpu().numpy().astype(np.uint64)
    sel_pairs = segs.seg_sel_pairs.cpu().numpy()
    sel_count = segs.seg_sel_count.cpu().numpy()
    for g in range(G):
        for j in range(N):
            expect = (not snap.node_unsched[j]) and selector_subset_match(
                sel_pairs[g], int(sel_count[g]), snap.node_labels[j],
                int(snap.node_label_count[j]),
            )
            bit = (int(got[g, j // 64]) >> (j % 64)) & 1
            assert bit == int(expect), (g, j)


def test_gpu_pipeline_scales_10k(engine):
    """Config #3 shape: 10k mixed pods, 3 heterogeneous pools."""
    pods_json, nodes_json = burst_trace(
        10000,
        n_nodes_per_pool=(40, 30, 30),
        pools=(("agentpool1", "Standard_D2_v2"), ("bigpool", "Standard_D4_v2"),
               ("gpupool", "Standard_ND96_MI355X_v1")),
        gpu_fraction=0.35, seed=7, bound_pods_per_node=2,
    )
    snap = _snap(pods_json, nodes_json)
    from mi355x_autoscaler.ops.engine_gpu import FALLBACK_COUNTS

    before = dict(FALLBACK_COUNTS)
    _assert_decisions_equal(engine.decide(snap), CpuEngine().decide(snap))
    # the headline shape must run the HIP fast path end to end
    assert dict(FALLBACK_COUNTS) == before


def test_segment_hash_path_equals_torch_path(engine):
    """Differential test: hash-aggregation segment build vs the
    torch-sort fallback produce identical segments and decisions."""
    import os

    from mi355x_autoscaler.ops.engine_gpu import GpuEngine

    pods_json, nodes_json = burst_trace(
        4000, n_nodes_per_pool=(20, 10),
        pools=(("agentpool1", "Standard_D2_v2"), ("gpupool", "Standard_ND96_MI355X_v1")),
        gpu_fraction=0.4, seed=13, bound_pods_per_node=2,
    )
    snap = _snap(pods_json, nodes_json)

    ds = engine.packer.pack(snap, engine.device)
    segs_hash = engine.build_segments_hash(ds)
    assert segs_hash is not None
    from mi355x_autoscaler.ops.engine_gpu import build_segments_device

    segs_torch = build_segments_device(ds, engine.ops)
    assert segs_hash.G == segs_torch.G
    np.testing.assert_array_equal(
        segs_hash.seg_sel_hash.cpu().numpy(), segs_torch.seg_sel_hash.cpu().numpy()
    )
    np.testing.assert_array_equal(
        segs_hash.seg_req.cpu().numpy(), segs_torch.seg_req.cpu().numpy()
    )
    np.testing.assert_array_equal(
        segs_hash.seg_count.cpu().numpy(), segs_torch.seg_count.cpu().numpy()
    )

    d_hash = engine.decide(snap)
    try:
        engine.force_torch_segbuild = True
        d_torch = engine.decide(snap)
    finally:
        engine.force_torch_segbuild = False
    _assert_decisions_equal(d_hash, d_torch)
    _assert_decisions_equal(d_hash, CpuEngine().decide(snap))


def test_segment_hash_many_segments(engine):
    """Hundreds of distinct (selector, request) shapes still fit the
    hash table and match the oracle."""
    import random

    rng = random.Random(0)
    node = make_node_json("agentpool1", 0)
    pods = []
    for i in range(900):
        pods.append(make_pod_json(
            f"p{i}", cpu_milli=rng.randrange(1, 400), mem_mib=rng.randrange(1, 300),
            pending=True,
        ))
    snap = _snap(pods, [node])
    _assert_decisions_equal(engine.decide(snap), CpuEngine().decide(snap))


def test_burst1m_decide_matches_oracle(engine):
    """BASELINE config #5 scale: 1M pending pods, 4k nodes, decisions
    still bit-identical to the oracle."""
    from mi355x_autoscaler.sim.synth_soa import synth_burst_snapshot

    snap = synth_burst_snapshot(
        1_000_000, n_nodes_per_pool=(2000, 1000, 1000), gpu_fraction=0.35, seed=3,
    )
    snap.pool_max[:] = 1 << 20
    d_gpu = engine.decide(snap)
    d_cpu = CpuEngine().decide(snap)
    _assert_decisions_equal(d_gpu, d_cpu)
    assert d_gpu.pods_placed_existing + d_gpu.pods_placed_new + d_gpu.pods_unfulfillable == 1_000_000


def test_snapshot_scatter_delta_matches_full_upload(engine):
    """Kernel E: decisions on a delta-patched resident snapshot equal
    decisions on a fresh full upload."""
    from mi355x_autoscaler.models.kube import KubeNode, KubePod
    from mi355x_autoscaler.snapshot.delta import DeltaSnapshotCache
    from mi355x_autoscaler.snapshot.encode import encode_snapshot
    from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools

    nodes_json = [make_node_json("agentpool1", i) for i in range(4)]
    nodes = [KubeNode(n) for n in nodes_json]
    pools = group_nodes_into_pools(nodes)
    initial = [make_pod_json(f"p{i}", cpu_milli=100, pending=True) for i in range(50)]
    snap0 = encode_snapshot([KubePod(p) for p in initial], nodes, pools)

    cache = DeltaSnapshotCache()
    cache.prime(snap0, {n.name: i for i, n in enumerate(nodes)})
    cache.drain_dirty()
    ds = engine.packer.pack(snap0, engine.device)

    # mutate: adds within capacity, a modify, a delete
    for i in range(10):
        cache.apply_event({"type": "ADDED", "object": make_pod_json(
            f"d{i}", cpu_milli=400, mem_mib=256, pending=True)})
    cache.apply_event({"type": "DELETED", "object": initial[0]})
    cache.apply_event({"type": "MODIFIED", "object": make_pod_json(
        "p1", cpu_milli=900, mem_mib=512, pending=True)})

    base = encode_snapshot([], nodes, pools)
    snap1 = cache.build_snapshot(base)

    if cache.size <= ds.P:
        ds2 = engine.apply_pod_delta(ds, cache)
    else:
        ds2 = None
    if ds2 is None:
        ds2 = engine.packer.pack(snap1, engine.device)
    else:
        # P may exceed live rows in snap1? sizes must match for compare
        assert ds2.P == cache.size

    segs = engine.build_segments(ds2)
    d_delta = engine.decide_device(ds2, segs, snap1.pool_names)
    d_full = CpuEngine().decide(snap1)
    assert d_delta.units_needed == d_full.units_needed
    assert d_delta.pods_placed_existing == d_full.pods_placed_existing
    assert d_delta.pods_unfulfillable == d_full.pods_unfulfillable


def test_sharded_engine_world1_gpu_path(engine):
    """ShardedEngine(GpuEngine) at world 1 exercises the same
    row-serialization + merge + decide_device path the driver's 8-GPU
    run takes; decisions must equal the plain engine's."""
    from mi355x_autoscaler.parallel.engine_sharded import ShardedEngine

    pods_json, nodes_json = burst_trace(3000, n_nodes_per_pool=(10, 5),
        pools=(("agentpool1", "Standard_D2_v2"), ("gpupool", "Standard_ND96_MI355X_v1")),
        gpu_fraction=0.4, seed=21, bound_pods_per_node=2)
    snap = _snap(pods_json, nodes_json)
    sharded = ShardedEngine(engine, check_agreement=False)
    d1 = sharded.decide(snap)
    d2 = engine.decide(snap)
    _assert_decisions_equal(d1, d2)
    rep1 = sharded.maintain(snap)
    rep2 = engine.maintain(snap)
    np.testing.assert_array_equal(rep1.node_util_permille, rep2.node_util_permille)

    # the packed-resident entry (bench world>1 path) and the cache-
    # resident entry must produce the same decision as plain decide
    ds = engine.packer.pack(snap, engine.device)
    _assert_decisions_equal(sharded.decide_packed(ds, snap.pool_names), d2)

    from mi355x_autoscaler.snapshot.delta import DeltaSnapshotCache
    from mi355x_autoscaler.snapshot.encode import encode_snapshot as _enc

    pods, nodes = wrap(pods_json, nodes_json)
    pools = group_nodes_into_pools(nodes)
    base = _enc([], nodes, pools)
    cache = DeltaSnapshotCache()
    cache.prime(snap, {n.name: i for i, n in enumerate(nodes)})
    _assert_decisions_equal(sharded.decide_resident(cache, base), d2)


@pytest.mark.parametrize("n_pods,n_nodes", [
    (1, 1), (64, 3), (1000, 500), (10000, 10000), (100000, 10000),
])
def test_shape_class_sweep(engine, n_pods, n_nodes):
    """SURVEY §4 kernel shape classes: pods x nodes grid including the
    10k-node eligibility/binpack capacity paths."""
    from mi355x_autoscaler.sim.synth_soa import synth_burst_snapshot

    per_pool = max(n_nodes // 3, 1)
    snap = synth_burst_snapshot(
        n_pods,
        n_nodes_per_pool=(per_pool, per_pool, n_nodes - 2 * per_pool)
        if n_nodes >= 3 else (n_nodes,),
        pools=(("agentpool1", "Standard_D2_v2"), ("bigpool", "Standard_D4_v2"),
               ("gpupool", "Standard_ND96_MI355X_v1"))[: (3 if n_nodes >= 3 else 1)],
        gpu_fraction=0.3 if n_nodes >= 3 else 0.0,
        seed=n_pods + n_nodes,
        bound_pods_per_node=1 if n_nodes <= 500 else 0,
    )
    snap.pool_max[:] = 1 << 20
    _assert_decisions_equal(engine.decide(snap), CpuEngine().decide(snap))


def test_decide_is_deterministic_across_repeats(engine):
    """Race screen: the lock-free hash build + atomics must produce the
    SAME decision on every repeat (integer ops commute exactly; slot
    assignment varies but the sorted segment list must not)."""
    pods_json, nodes_json = burst_trace(
        20000, n_nodes_per_pool=(40, 30, 30),
        pools=(("agentpool1", "Standard_D2_v2"), ("bigpool", "Standard_D4_v2"),
               ("gpupool", "Standard_ND96_MI355X_v1")),
        gpu_fraction=0.35, seed=33, bound_pods_per_node=2,
    )
    snap = _snap(pods_json, nodes_json)
    first = engine.decide(snap)
    for _ in range(15):
        d = engine.decide(snap)
        _assert_decisions_equal(d, first)


def test_native_orchestrator_equals_python_pipeline(engine):
    """The one-call C++ orchestrator and the op-by-op Python sequencing
    of the fused pipeline must produce identical decisions."""
    pods_json, nodes_json = burst_trace(
        8000, n_nodes_per_pool=(40, 30, 30),
        pools=(("agentpool1", "Standard_D2_v2"), ("bigpool", "Standard_D4_v2"),
               ("gpupool", "Standard_ND96_MI355X_v1")),
        gpu_fraction=0.35, seed=44, bound_pods_per_node=2,
    )
    snap = _snap(pods_json, nodes_json)
    d_native = engine.decide(snap)
    try:
        engine.force_py_pipeline = True
        d_py = engine.decide(snap)
    finally:
        engine.force_py_pipeline = False
    _assert_decisions_equal(d_native, d_py)
    _assert_decisions_equal(d_native, CpuEngine().decide(snap))


def test_decide_resident_matches_oracle_through_churn(engine):
    """Resident-controller mode (kernel E end-to-end): prime, churn via
    watch events (adds/mods/deletes, capacity growth), and every
    decision over the HBM-resident snapshot equals the oracle."""
    import random

    from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
    from mi355x_autoscaler.models.kube import KubeNode, KubePod
    from mi355x_autoscaler.snapshot.delta import DeltaSnapshotCache
    from mi355x_autoscaler.snapshot.encode import encode_snapshot

    rng = random.Random(5)
    nodes_json = [make_node_json("agentpool1", i) for i in range(5)]
    nodes = [KubeNode(n) for n in nodes_json]
    pools = group_nodes_into_pools(nodes)
    initial = [make_pod_json(f"p{i}", cpu_milli=100, pending=True) for i in range(200)]
    snap0 = encode_snapshot([KubePod(p) for p in initial], nodes, pools)

    cache = DeltaSnapshotCache()
    cache.prime(snap0, {n.name: i for i, n in enumerate(nodes)})
    engine._resident = None
    live = {p["metadata"]["name"]: p for p in initial}

    for cycle in range(8):
        for _ in range(rng.randrange(5, 120)):  # growth forces cap changes
            p = make_pod_json(f"c{cycle}-{rng.randrange(10**6)}",
                              cpu_milli=rng.choice([100, 400, 900]),
                              mem_mib=rng.choice([64, 512]), pending=True)
            cache.apply_event({"type": "ADDED", "object": p})
            live[p["metadata"]["name"]] = p
        for name in rng.sample(sorted(live), min(len(live), rng.randrange(0, 40))):
            cache.apply_event({"type": "DELETED", "object": live.pop(name)})
        cache.maybe_compact()

        base = encode_snapshot([], nodes, pools)
        d_gpu = engine.decide_resident(cache, base)
        d_cpu = CpuEngine().decide(cache.build_snapshot(base))
        assert d_gpu.units_needed == d_cpu.units_needed, cycle
        assert d_gpu.pods_placed_existing == d_cpu.pods_placed_existing, cycle
        assert d_gpu.pods_unfulfillable == d_cpu.pods_unfulfillable, cycle


def test_no_pools_cluster(engine):
    """Edge: every node is a master (no agent pools) — pending pods
    that fit nowhere must count unfulfillable, not silently vanish."""
    master = {"metadata": {"name": "k8s-master-12345678-0", "uid": "m0",
                           "labels": {}},
              "spec": {}, "status": {"capacity": {"cpu": "2", "memory": "4Gi",
                                                  "pods": "10"}}}
    pods = [make_pod_json(f"p{i}", cpu_milli=500, pending=True) for i in range(5)]
    snap = _snap(pods, [master])
    assert snap.num_pools == 0
    d_gpu = engine.decide(snap)
    d_cpu = CpuEngine().decide(snap)
    # the master is schedulable capacity in the fit simulation: 4 x 500m
    # fit its 2000m; the 5th has no pool to grow -> unfulfillable
    assert d_gpu.pods_placed_existing == d_cpu.pods_placed_existing == 4
    assert d_gpu.pods_unfulfillable == d_cpu.pods_unfulfillable == 1
    assert not d_gpu.wants_scale_up()


def test_hipgraph_optin_replay_matches_eager_and_oracle(engine, monkeypatch):
    """hipGraph capture is opt-in (measured slower in situ on ROCm 7.2 —
    see engine_gpu.__init__ and profiles/hipgraph_probe.txt); when
    enabled, warm-up -> capture -> replay must still produce
    bit-identical decisions, and the graph cache must actually be
    exercised."""
    from mi355x_autoscaler.ops.engine_gpu import GpuEngine

    assert not engine._graph_ok  # default is eager
    monkeypatch.setenv("MI355X_AS_HIPGRAPH", "1")
    graphed = GpuEngine("cuda:0")
    assert graphed._graph_ok
    pods_json, nodes_json = burst_trace(
        2000, n_nodes_per_pool=(5,), seed=11, bound_pods_per_node=2)
    snap = _snap(pods_json, nodes_json)
    ref = CpuEngine().decide(snap)
    for _ in range(4):
        _assert_decisions_equal(graphed.decide(snap), ref)
        _assert_decisions_equal(engine.decide(snap), ref)
    if graphed._graph_ok:  # capture may legitimately disable itself
        assert any(not isinstance(v, str) for v in graphed._graph_cache.values()), \
            "expected at least one captured hipGraph after repeated decides"


def test_wide_selector_decide_matches_oracle(engine):
    """12-pair selector sets (old width was 8) through the full GPU
    pipeline: bit-equal to the oracle, correct node discrimination."""
    sel = {f"k{i:02d}": f"v{i}" for i in range(12)}
    full = make_node_json("agentpool1", 0)
    full["metadata"]["labels"].update(sel)
    partial = make_node_json("agentpool1", 1)
    partial["metadata"]["labels"].update({k: sel[k] for k in list(sel)[:11]})
    pods = [make_pod_json(f"w{i}", cpu_milli=100, pending=True, node_selector=sel)
            for i in range(5)]
    pods += [make_pod_json("plain", cpu_milli=100, pending=True)]
    snap = _snap(pods, [full, partial])
    _assert_decisions_equal(engine.decide(snap), CpuEngine().decide(snap))


def test_more_than_64_pools_falls_back_to_oracle(engine):
    """>64 agent pools exceeds the units kernel's fixed LDS budget: the
    engine must decide on the CPU oracle (bit-identical), not corrupt
    LDS or crash (the orchestrator TORCH_CHECKs as a second fence)."""
    nodes_json = [make_node_json(f"pool{i:02d}", 0) for i in range(66)]
    pods = [make_pod_json(f"p{i}", cpu_milli=500, pending=True)
            for i in range(40)]
    snap = _snap(pods, nodes_json)
    assert snap.num_pools == 66
    from mi355x_autoscaler.ops.engine_gpu import FALLBACK_COUNTS

    before = FALLBACK_COUNTS["many_pools_oracle"]
    _assert_decisions_equal(engine.decide(snap), CpuEngine().decide(snap))
    # the fallback must be LOUD: counter bumped (exported as a gauge)
    assert FALLBACK_COUNTS["many_pools_oracle"] == before + 1


def test_overflow_selector_matches_nothing_on_gpu(engine):
    """>32-pair selectors are poisoned at encode (exact matches-nothing
    semantics): the GPU pipeline must agree with the oracle and place
    zero such pods, even on a node whose first 32 sorted labels cover
    the truncated pair array."""
    sel33 = {f"k{i:02d}": f"v{i}" for i in range(33)}
    covered = make_node_json("agentpool1", 0)
    covered["metadata"]["labels"].update(sel33)
    pods = [make_pod_json(f"o{i}", cpu_milli=100, pending=True,
                          node_selector=sel33) for i in range(3)]
    pods += [make_pod_json("plain", cpu_milli=100, pending=True)]
    snap = _snap(pods, [covered])
    d_gpu, d_cpu = engine.decide(snap), CpuEngine().decide(snap)
    _assert_decisions_equal(d_gpu, d_cpu)
    assert d_gpu.pods_placed_existing == 1  # the plain pod only


def test_decide_rows_device_matches_host_merge(engine):
    """Sharded DEVICE merge (rows_hash + compact on device) must equal
    the host-side numpy merge path on identical gathered rows —
    including padded (count 0) rows from the size-aligned all-gather
    and cross-rank duplicates."""
    import torch

    from mi355x_autoscaler.parallel.engine_sharded import merge_rows

    pods_json, nodes_json = burst_trace(
        3000, n_nodes_per_pool=(5, 4),
        pools=(("agentpool1", "Standard_D2_v2"), ("gpupool", "Standard_ND96_MI355X_v1")),
        gpu_fraction=0.4, seed=7, bound_pods_per_node=2)
    snap = _snap(pods_json, nodes_json)
    ds = engine.packer.pack(snap, engine.device)
    segs = engine.build_segments(ds)
    rows_dev = engine.rows_device(segs)
    rows_h = rows_dev.cpu().numpy()

    # simulate a 3-rank gather with per-rank padding
    pad = torch.zeros((2, rows_dev.shape[1]), dtype=torch.int64,
                      device=engine.device)
    gathered = torch.cat([rows_dev, pad, rows_dev, pad, rows_dev, pad], dim=0)
    d_dev = engine.decide_rows_device(ds, gathered, snap.pool_names)

    merged = merge_rows(np.concatenate([rows_h, rows_h, rows_h], axis=0))
    d_host = engine.decide_merged_rows(ds, merged, snap.pool_names)
    _assert_decisions_equal(d_dev, d_host)
    assert d_dev.pods_placed_existing + d_dev.pods_placed_new + \
        d_dev.pods_unfulfillable == 3 * 3000


def test_decide_rows_device_poison_falls_back(engine):
    """A poison row (negative count = some rank's local hash overflow)
    must make decide_rows_device return None — the sharded engine then
    falls back on every rank together."""
    import torch

    from mi355x_autoscaler.snapshot.encode import MAX_SELECTOR_PAIRS as MS

    pods_json, nodes_json = burst_trace(500, n_nodes_per_pool=(3,), seed=3,
                                        bound_pods_per_node=2)
    snap = _snap(pods_json, nodes_json)
    ds = engine.packer.pack(snap, engine.device)
    segs = engine.build_segments(ds)
    rows = engine.rows_device(segs).clone()
    poison = torch.zeros((1, rows.shape[1]), dtype=torch.int64,
                         device=engine.device)
    poison[0, MS + 5] = -1
    gathered = torch.cat([rows, poison], dim=0)
    assert engine.decide_rows_device(ds, gathered, snap.pool_names) is None
    # without the poison the same rows decide normally
    d = engine.decide_rows_device(ds, rows, snap.pool_names)
    assert d is not None
    _assert_decisions_equal(d, CpuEngine().decide(snap))


def test_randomized_differential_fuzz(engine):
    """Randomized shapes (pools, sizes, selector widths, criticals,
    zero-requests) + simulated multi-rank sharded merges, all asserted
    bit-equal to the CPU oracle (scripts/fuzz_gpu.py runs the longer
    form)."""
    import importlib.util
    import os

    spec = importlib.util.spec_from_file_location(
        "fuzz_gpu", os.path.join(os.path.dirname(__file__), "..",
                                 "scripts", "fuzz_gpu.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    mod.ITERS = 12
    mod.main()


def test_units_long_run_lists_and_splits(engine):
    """Stress the round-2 array-backed units run store: one pool, ~700
    distinct request shapes over tiny nodes forces hundreds of open
    runs (multi-chunk wave scans), repeated mid-list boundary splits
    and deep ord shifts — bit-equal to the oracle."""
    import random

    rng = random.Random(7)
    nodes = [make_node_json("agentpool1", i) for i in range(2)]
    pods = []
    for i in range(1400):
        # alternate wide/narrow shapes so later segments partially
        # consume earlier runs (split + insert), not just append
        cpu = rng.choice([3, 7, 11, 37, 101, 397, 998])
        mem = rng.randrange(1, 900)
        for j in range(rng.randrange(1, 4)):
            pods.append(make_pod_json(f"u{i}-{j}", cpu_milli=cpu, mem_mib=mem,
                                      pending=True))
    snap = _snap(pods, nodes, max_size=4096)
    assert snap.pool_max[0] >= 4096
    d_gpu = engine.decide(snap)
    d_cpu = CpuEngine().decide(snap)
    _assert_decisions_equal(d_gpu, d_cpu)
    assert sum(d_cpu.units_needed.values()) > 64  # many runs really opened


def test_units_interleaved_pools_long_lists(engine):
    """Same stress across 3 pools (waves own different pools) with
    selector-pinned segments so each pool's ord list grows and splits
    independently."""
    import random

    rng = random.Random(11)
    pools = ["agentpool1", "bigpool", "gpupool"]
    nodes_json = []
    for p in pools:
        nodes_json.append(make_node_json(p, 0))
    pods = []
    for i in range(900):
        p = pools[i % 3]
        pods.append(make_pod_json(
            f"q{i}", cpu_milli=rng.choice([5, 13, 57, 333, 777]),
            mem_mib=rng.randrange(1, 600), pending=True,
            node_selector={"agentpool": p}))
    snap = _snap(pods, nodes_json, max_size=4096)
    _assert_decisions_equal(engine.decide(snap), CpuEngine().decide(snap))


def test_pool_demand_matches_numpy_both_paths(engine):
    """pool_demand() — the MFMA demand segsum — must equal the numpy
    per-pool pending request totals on BOTH execution paths: the
    in-mega idle-block segsum (small P) and the classic side-stream
    fork (P above the mega threshold)."""
    for n_pods in (8000, 150_000):  # mega path / classic split path
        pods_json, nodes_json = burst_trace(
            n_pods, n_nodes_per_pool=(20, 10),
            pools=(("agentpool1", "Standard_D2_v2"),
                   ("gpupool", "Standard_ND96_MI355X_v1")),
            gpu_fraction=0.4, seed=3, bound_pods_per_node=2)
        snap = _snap(pods_json, nodes_json)
        engine.decide(snap)
        demand = engine.pool_demand()
        assert demand is not None
        got = demand.cpu().numpy()
        assert got.shape == (snap.num_pools, 4)

        # CPU reference: each segment routes to its first candidate
        # pool (seg_pool_route semantics); sum its pods' requests there.
        sim = OracleSimulator(snap)
        ref = np.zeros((snap.num_pools, 4), np.int64)
        for seg in build_segments(snap):
            cands = sim.pool_candidates(seg)
            if cands:
                ref[cands[0]] += snap.pod_req[seg.pod_indices].astype(
                    np.int64).sum(axis=0)
        np.testing.assert_array_equal(got, ref)
