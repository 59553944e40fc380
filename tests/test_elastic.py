"""Elastic re-sharding: decisions are invariant to the world size the
pod axis is sharded over (the property that makes GPU-loss recovery a
pure relaunch)."""

import numpy as np

from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.parallel.elastic import shard_for, shard_indices, shard_snapshot
from mi355x_autoscaler.sim.engine import CpuEngine
from mi355x_autoscaler.sim.oracle import OracleSimulator, build_segments, merge_segment_counts
from mi355x_autoscaler.sim.synth import burst_trace, wrap
from mi355x_autoscaler.snapshot.encode import encode_snapshot


def _snap(seed=0, n=600):
    pods_json, nodes_json = burst_trace(n, n_nodes_per_pool=(5,), seed=seed,
                                        bound_pods_per_node=2)
    pods, nodes = wrap(pods_json, nodes_json)
    return encode_snapshot(pods, nodes, group_nodes_into_pools(nodes))


def test_shard_assignment_deterministic_and_complete():
    snap = _snap()
    for world in (1, 2, 3, 7, 8):
        shards = [shard_indices(snap.pod_uids, r, world) for r in range(world)]
        union = np.sort(np.concatenate(shards))
        np.testing.assert_array_equal(union, np.arange(snap.num_pods))
        # stability: same uid -> same rank on recompute
        for r in range(world):
            np.testing.assert_array_equal(shards[r], shard_indices(snap.pod_uids, r, world))


def test_decisions_invariant_under_world_change():
    """8-wide sharding, a 'rank loss', and 5-wide re-sharding all agree
    with the single-process decision (the recovery correctness core)."""
    snap = _snap(seed=3)
    d_ref = CpuEngine().decide(snap)
    for world in (8, 5, 2):
        segs = [build_segments(shard_snapshot(snap, r, world)) for r in range(world)]
        merged = merge_segment_counts(segs)
        d = OracleSimulator(snap).fulfill_pending(merged)
        assert d.units_needed == d_ref.units_needed
        assert d.pods_placed_existing == d_ref.pods_placed_existing
        assert d.pods_unfulfillable == d_ref.pods_unfulfillable


def test_rank_loss_error_translation():
    from mi355x_autoscaler.parallel.elastic import RankLossError, run_elastic_cycle

    class Dying:
        def decide(self, snap, over_provision=0):
            raise RuntimeError("NCCL communicator was aborted")

    class Other:
        def decide(self, snap, over_provision=0):
            raise RuntimeError("some unrelated bug")

    snap = _snap(seed=4, n=10)
    try:
        run_elastic_cycle(Dying(), snap)
        assert False
    except RankLossError:
        pass
    try:
        run_elastic_cycle(Other(), snap)
        assert False
    except RuntimeError as e:
        assert "unrelated" in str(e)


def test_loop_driver_translates_rank_loss(monkeypatch):
    """The sleep-loop retries ordinary cycle failures but must NOT
    spin against a dead communicator: at world>1 a collective failure
    surfaces as RankLossError for the supervisor to relaunch."""
    from mi355x_autoscaler.ctl.loop import Cluster, ClusterOptions
    from mi355x_autoscaler.parallel import dist as pdist
    from mi355x_autoscaler.parallel.elastic import RankLossError

    cluster = Cluster.__new__(Cluster)
    cluster.opts = ClusterOptions(sleep_seconds=0)
    cluster._cycle = 0

    class _N:
        def notify_error(self, msg):
            pass

    cluster.notifier = _N()

    def dead_cycle(now=None):
        raise RuntimeError("Connection closed by peer")

    cluster.loop_logic = dead_cycle

    # world 1: an ordinary failure — retried, loop terminates by count
    monkeypatch.setattr(pdist, "world_size", lambda: 1)
    cluster.loop(max_cycles=2)
    assert cluster._cycle == 2

    # world 4: the same error is a lost peer — loop must raise
    monkeypatch.setattr(pdist, "world_size", lambda: 4)
    cluster._cycle = 0
    try:
        cluster.loop(max_cycles=2)
        assert False, "expected RankLossError"
    except RankLossError:
        pass

    # a non-collective bug at world 4 still retries
    def buggy_cycle(now=None):
        raise RuntimeError("some unrelated bug")

    cluster.loop_logic = buggy_cycle
    cluster._cycle = 0
    cluster.loop(max_cycles=2)
    assert cluster._cycle == 2
