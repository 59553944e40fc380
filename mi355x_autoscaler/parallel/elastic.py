"""Elastic recovery for the sharded controller (SURVEY §5 failure
detection / elastic recovery).

The controller is stateless by design (the reference's crash-loop
resilience, inherited [R: cluster.py — reconstructed]): every cycle
re-derives the world from the API server, so losing a GPU/rank costs
at most one cycle. The recovery contract:

  1. pod->rank assignment is a PURE FUNCTION of (pod uid, world size)
     (`shard_for`, uid-hash mod world) — when the world shrinks from 8
     to 7, every rank recomputes its shard with zero coordination and
     the union is exactly the pod set;
  2. any collective failure (a peer died mid-all-reduce) raises out of
     ShardedEngine.decide; `run_elastic_cycle` catches it, destroys the
     process group and re-raises `RankLossError` so the supervisor
     (torchrun --max-restarts, or a k8s restartPolicy) relaunches with
     the surviving world — the next cycle is already correct;
  3. decision invariance across world sizes is the CI-tested property
     (tests/test_sharded_cpu.py, tests/test_elastic.py): decisions are
     a function of the merged segment counts, not of the sharding.
"""

from __future__ import annotations

from typing import Sequence

import numpy as np

from ..snapshot.encode import ClusterSnapshot, fnv1a64


class RankLossError(RuntimeError):
    """A peer rank died mid-collective; re-launch with the new world."""


def shard_for(uid: str, world: int) -> int:
    """Deterministic pod->rank assignment (uid-hash mod world)."""
    return int(fnv1a64(uid.encode()) % world) if world > 1 else 0


def shard_indices(pod_uids: Sequence[str], rank: int, world: int) -> np.ndarray:
    """Indices of the pods this rank owns under the current world."""
    if world <= 1:
        return np.arange(len(pod_uids))
    return np.array(
        [i for i, u in enumerate(pod_uids) if shard_for(u, world) == rank],
        dtype=np.int64,
    )


def shard_snapshot(snap: ClusterSnapshot, rank: int, world: int) -> ClusterSnapshot:
    """Restrict the pod axis to this rank's shard (nodes/pools stay
    replicated). Used by controllers that ingest the full snapshot and
    shard locally; re-sharding after rank loss is just calling this
    with the new world."""
    import dataclasses

    if world <= 1:
        return snap
    idx = shard_indices(snap.pod_uids, rank, world)
    has_lbl = snap.pod_label_pairs.shape[0] == snap.num_pods
    sub = dataclasses.replace(
        snap,
        pod_req=snap.pod_req[idx],
        pod_flags=snap.pod_flags[idx],
        pod_sel_hash=snap.pod_sel_hash[idx],
        pod_sel_pairs=snap.pod_sel_pairs[idx],
        pod_sel_count=snap.pod_sel_count[idx],
        pod_node=snap.pod_node[idx],
        pod_pdb=snap.pod_pdb[idx],
        pod_label_pairs=(snap.pod_label_pairs[idx] if has_lbl
                         else snap.pod_label_pairs),
        pod_label_count=(snap.pod_label_count[idx] if has_lbl
                         else snap.pod_label_count),
        pod_uids=[snap.pod_uids[i] for i in idx],
        pod_refs=[snap.pod_refs[i] for i in idx],
    )
    from ..snapshot.encode import compute_node_free

    compute_node_free(sub)
    return sub


_COLLECTIVE_FAILURE_KEYWORDS = (
    "nccl", "rccl", "gloo", "connection", "timed out", "timeout",
    "closed", "peer", "aborted", "econnreset", "broken pipe")


def is_collective_failure(exc: BaseException) -> bool:
    """Does this exception look like a lost peer / dead communicator
    (as opposed to an ordinary bug)?"""
    msg = str(exc).lower()
    return any(k in msg for k in _COLLECTIVE_FAILURE_KEYWORDS)


def run_elastic_cycle(engine, snap: ClusterSnapshot, over_provision: int = 0):
    """One decide() with rank-loss translation: collective failures
    (RCCL aborts, gloo timeouts) surface as RankLossError after tearing
    the group down, so the supervisor can relaunch cleanly."""
    import torch.distributed as dist

    try:
        return engine.decide(snap, over_provision=over_provision)
    except RuntimeError as e:
        if is_collective_failure(e):
            if dist.is_available() and dist.is_initialized():
                try:
                    dist.destroy_process_group()
                except Exception:
                    pass
            raise RankLossError(f"peer rank lost during collective: {e}") from e
        raise
