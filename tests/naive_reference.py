"""Naive per-pod first-fit — a direct re-creation of the reference's
interpreted triple loop [R: autoscaler/cluster.py · fulfill_pending —
reconstructed]. Used ONLY in tests to prove the oracle's uniform-step
scan decomposition computes exactly first-fit-decreasing, and in
BASELINE.md as the 'reference-equivalent' timing stand-in."""

from __future__ import annotations

from typing import Dict, List, Tuple

import numpy as np

from mi355x_autoscaler.models.resources import NUM_RESOURCES
from mi355x_autoscaler.sim.oracle import build_segments
from mi355x_autoscaler.snapshot.encode import ClusterSnapshot, selector_subset_match


def naive_fulfill(snap: ClusterSnapshot) -> Dict:
    """Place pending pods one at a time, first-fit over nodes in index
    order, in the oracle's global segment order; residuals FFD into
    virtual new nodes of the first eligible pool. Pure triple loop."""
    free = snap.node_free.astype(np.int64).copy()
    segments = build_segments(snap)
    placed_existing = 0
    residuals: List[Tuple] = []

    for seg in segments:
        elig = []
        for j in range(snap.num_nodes):
            if snap.node_unsched[j]:
                elig.append(False)
                continue
            elig.append(
                selector_subset_match(
                    seg.sel_pairs, seg.sel_count, snap.node_labels[j], int(snap.node_label_count[j])
                )
            )
        remaining = seg.count
        for _ in range(seg.count):
            placed = False
            for j in range(snap.num_nodes):
                if not elig[j]:
                    continue
                if all(int(seg.req[r]) <= int(free[j, r]) for r in range(NUM_RESOURCES)):
                    free[j] -= seg.req.astype(np.int64)
                    placed = True
                    break
            if placed:
                placed_existing += 1
                remaining -= 1
            else:
                break  # identical pods: if one doesn't fit anywhere, none do
        if remaining:
            residuals.append((seg, remaining))

    units: Dict[int, List[np.ndarray]] = {k: [] for k in range(snap.num_pools)}
    unfulfillable = 0
    placed_new = 0
    for seg, remaining in residuals:
        cand = None
        for k in range(snap.num_pools):
            if not selector_subset_match(
                seg.sel_pairs, seg.sel_count, snap.pool_labels[k], int(snap.pool_label_count[k])
            ):
                continue
            if all(int(seg.req[r]) <= int(snap.pool_unit[k, r]) for r in range(NUM_RESOURCES)):
                cand = k
                break
        if cand is None:
            unfulfillable += remaining
            continue
        max_new = max(0, int(snap.pool_max[cand]) - int(snap.pool_cur[cand]))
        req = seg.req.astype(np.int64)
        for _ in range(remaining):
            placed = False
            for u in units[cand]:
                if all(int(req[r]) <= int(u[r]) for r in range(NUM_RESOURCES)):
                    u -= req
                    placed = True
                    break
            if not placed and len(units[cand]) < max_new:
                u = snap.pool_unit[cand].astype(np.int64).copy()
                if all(int(req[r]) <= int(u[r]) for r in range(NUM_RESOURCES)):
                    u -= req
                    units[cand].append(u)
                    placed = True
            if placed:
                placed_new += 1
            else:
                unfulfillable += 1

    units_needed = {
        snap.pool_names[k]: len(v) for k, v in units.items() if len(v) > 0
    }
    return {
        "placed_existing": placed_existing,
        "placed_new": placed_new,
        "unfulfillable": unfulfillable,
        "units_needed": units_needed,
        "free": free,
    }
