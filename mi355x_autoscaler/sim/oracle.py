"""CPU oracle: the executable specification of the scheduling simulation.

Semantics re-created from the reference's hot path
[R: autoscaler/cluster.py · fulfill_pending / maintain — reconstructed,
unverified], then *re-stated in a GPU-decomposable form*:

  First-fit-decreasing placement over pending pods is expressed as a
  sequence of **uniform placement steps**, one per segment of
  interchangeable pods (same selector set + same quantized request),
  processed in (selector-hash asc, request desc) order. Each step is:

    1. per-node eligibility  (selector subset-match AND schedulable)
    2. per-node capacity     m_j = min_r floor(free[j,r] / req[r])
    3. exclusive prefix-scan of m_j in node-index order
    4. place min(count, total) pods; node j absorbs
       clamp(count - scan_j, 0, m_j)
    5. free[j] -= placed_j * req

  This is EXACTLY classic first-fit of the segment's pods one at a time
  (equal pods are interchangeable, and first-fit fills node j to its
  capacity before touching node j+1 within a segment), but steps 1-5 are
  embarrassingly parallel over nodes — which is what the HIP kernels
  A/B implement and what makes 1-GPU and 8-GPU decisions identical:
  ranks shard pods, all-reduce the per-segment counts, and every rank
  runs the same node-table math.

  Residual (unplaced) pods then bin-pack into *virtual new nodes* of the
  first eligible pool's unit capacity, in the same global segment order,
  yielding units_needed per pool (capped at pool max; the rest is
  unfulfillable) [R: autoscaler/cluster.py · units-needed accumulation].

Every HIP kernel is property-tested against this module (tests/).
All arithmetic is int32/int64 on the snapshot lattice — no floats in
any decision.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import numpy as np

from ..models.resources import NUM_RESOURCES
from ..snapshot.encode import ClusterSnapshot, PodFlags, selector_subset_match
from .decision import DrainCandidate, MaintainReport, ScaleDecision, Segment

_I64 = np.int64


def _pack_req_key(req: np.ndarray, critical: bool = False) -> int:
    """Pack a request vector into one uint64 sort key such that
    descending key order == descending (critical, cpu, memory, gpu)
    tuple order — critical pods [R: reference critical-pod annotation]
    outrank every non-critical shape within their selector group.
    cpu milli <= 2^21 (2096 cores), mem MiB <= 2^23 (8 TiB), gpu <= 2^8.
    Values clamp into their fields; the pods axis is the constant 1 for
    a single pod and is excluded. Must match ops/hip/common.h."""
    cpu = min(int(req[0]), (1 << 21) - 1)
    mem = min(int(req[1]), (1 << 23) - 1)
    gpu = min(int(req[3]), (1 << 8) - 1)
    return ((1 << 55) if critical else 0) | (cpu << 31) | (mem << 8) | gpu


def pack_req_keys(req: np.ndarray, critical: Optional[np.ndarray] = None) -> np.ndarray:
    """Vectorized _pack_req_key over a [M, 4] int array -> uint64[M]."""
    r = req.astype(np.int64)
    cpu = np.clip(r[:, 0], 0, (1 << 21) - 1).astype(np.uint64)
    mem = np.clip(r[:, 1], 0, (1 << 23) - 1).astype(np.uint64)
    gpu = np.clip(r[:, 3], 0, (1 << 8) - 1).astype(np.uint64)
    key = (cpu << np.uint64(31)) | (mem << np.uint64(8)) | gpu
    if critical is not None:
        key |= critical.astype(np.uint64) << np.uint64(55)
    return key


def build_segments(
    snap: ClusterSnapshot, pod_subset: Optional[np.ndarray] = None
) -> List[Segment]:
    """Group pending pods into placement segments, globally ordered by
    (selector-set hash as uint64 asc, request key desc).

    ``pod_subset``: optional index array (a shard); segment ORDER is a
    pure function of (sel_hash, req), so per-shard segment lists merge
    by key across ranks.
    """
    if pod_subset is None:
        pod_subset = np.arange(snap.num_pods)
    flags = snap.pod_flags[pod_subset]
    pending = pod_subset[(flags & PodFlags.PENDING) != 0]
    if pending.size == 0:
        return []

    sel = snap.pod_sel_hash[pending].astype(np.uint64)
    crit = (snap.pod_flags[pending] & PodFlags.CRITICAL) != 0
    keys = pack_req_keys(snap.pod_req[pending], crit)
    # order: sel asc, key desc, pod index asc (determinism of pod lists)
    order = np.lexsort((pending, np.iinfo(np.uint64).max - keys, sel))
    pending = pending[order]
    sel = sel[order]
    keys = keys[order]

    segments: List[Segment] = []
    start = 0
    for i in range(1, pending.size + 1):
        if i == pending.size or sel[i] != sel[start] or keys[i] != keys[start]:
            p0 = pending[start]
            segments.append(
                Segment(
                    sel_hash=int(snap.pod_sel_hash[p0]),
                    sel_pairs=snap.pod_sel_pairs[p0],
                    sel_count=int(snap.pod_sel_count[p0]),
                    req=snap.pod_req[p0].copy(),
                    count=int(i - start),
                    req_key=int(keys[start]),
                    pod_indices=pending[start:i].copy(),
                )
            )
            start = i
    return segments


def merge_segment_counts(per_rank: List[List[Segment]]) -> List[Segment]:
    """Merge per-shard segment lists into the global list (sum counts by
    (sel_hash, req) key, keep global order). CPU reference of the
    all-reduce in parallel/decision.py."""
    table: Dict[Tuple[int, int], Segment] = {}
    for segs in per_rank:
        for s in segs:
            k = ((int(s.sel_hash) & ((1 << 64) - 1)), int(s.req_key))
            if k in table:
                table[k].count += s.count
            else:
                table[k] = Segment(
                    sel_hash=s.sel_hash,
                    sel_pairs=s.sel_pairs,
                    sel_count=s.sel_count,
                    req=s.req.copy(),
                    count=s.count,
                    req_key=s.req_key,
                    pod_indices=None,
                )
    order = sorted(table.keys(), key=lambda k: (k[0], -k[1]))
    return [table[k] for k in order]


class OracleSimulator:
    """Slow, obviously-correct scheduling simulation on the snapshot
    lattice. The GPU engine must reproduce its decisions bit-for-bit."""

    def __init__(self, snap: ClusterSnapshot, over_provision: int = 0):
        self.snap = snap
        self.over_provision = over_provision

    # ------------------------------------------------------------------
    # scale-up
    # ------------------------------------------------------------------
    def node_eligibility(self, seg: Segment) -> np.ndarray:
        """[N] bool: schedulable node whose labels contain the selector.
        (Device form: kernel A fit_predicates.hip.) Vectorized subset
        test: selector pair s is present on node j iff any of the
        node's padded label hashes equals s — semantically identical to
        selector_subset_match (tested directly in test_snapshot.py),
        but numpy-parallel over nodes."""
        snap = self.snap
        sched = snap.node_unsched == 0
        if seg.sel_count == 0:
            return sched.copy()
        sel = seg.sel_pairs[: seg.sel_count]
        # [N, L, S]: node label l equals selector pair s; 0-padded label
        # slots never match a real pair hash (0 is not a valid FNV pair)
        present = (snap.node_labels[:, :, None] == sel[None, None, :]).any(axis=1)
        return sched & present.all(axis=1)

    @staticmethod
    def _per_node_capacity(free: np.ndarray, req: np.ndarray, elig: np.ndarray) -> np.ndarray:
        """m_j = min over resources of floor(free_j / req_r), 0 where
        ineligible or any free axis negative. int64 throughout."""
        N = free.shape[0]
        m = np.full(N, np.iinfo(_I64).max, dtype=_I64)
        for r in range(NUM_RESOURCES):
            rr = int(req[r])
            if rr <= 0:
                continue
            m = np.minimum(m, free[:, r] // rr)
        neg = (free < 0).any(axis=1)
        m[neg] = 0
        m[~elig] = 0
        m[m < 0] = 0
        # a node can never take more pods than its pod-slot axis anyway;
        # cap to a sane bound to keep the scan in int64 territory
        return np.minimum(m, 1 << 31)

    def place_segment(self, free: np.ndarray, seg: Segment, elig: np.ndarray) -> Tuple[int, np.ndarray]:
        """One uniform placement step. Returns (pods placed, per-node
        placement counts). Mutates ``free`` in place.
        (Device form: kernel B binpack.hip, one segment step)"""
        m = self._per_node_capacity(free, seg.req, elig)
        scan = np.concatenate(([0], np.cumsum(m)[:-1]))  # exclusive
        placed_j = np.clip(seg.count - scan, 0, m)
        total = int(placed_j.sum())
        if total > seg.count:  # clip can momentarily over-count at the boundary node
            raise AssertionError("scan placement overflow")
        if total:
            free -= placed_j[:, None] * seg.req.astype(_I64)[None, :]
        return total, placed_j

    def pool_candidates(self, seg: Segment) -> List[int]:
        """Pools (ascending index = sorted name) whose prototype labels
        match the selector and whose unit capacity fits one pod."""
        snap = self.snap
        out = []
        for k in range(snap.num_pools):
            if not selector_subset_match(
                seg.sel_pairs, seg.sel_count, snap.pool_labels[k], int(snap.pool_label_count[k])
            ):
                continue
            unit = snap.pool_unit[k].astype(_I64)
            if all(int(seg.req[r]) <= int(unit[r]) for r in range(NUM_RESOURCES)):
                out.append(k)
        return out

    def fulfill_pending(
        self,
        segments: Optional[List[Segment]] = None,
        node_free: Optional[np.ndarray] = None,
    ) -> ScaleDecision:
        """The whole scale-up decision [R: cluster.py · fulfill_pending].

        Phase 1: place segments onto existing nodes (uniform steps).
        Phase 2: residual pods bin-pack into virtual new nodes of the
                 first candidate pool, FFD in global segment order.

        ``node_free``: override for the free-capacity table (the
        sharded engine passes the all-reduced global table when pods
        are sharded across ranks).
        """
        snap = self.snap
        if segments is None:
            segments = build_segments(snap)
        base_free = snap.node_free if node_free is None else node_free
        free = base_free.astype(_I64).copy()
        decision = ScaleDecision()
        residuals: List[Tuple[Segment, int]] = []

        for seg in segments:
            elig = self.node_eligibility(seg)
            placed, _ = self.place_segment(free, seg, elig)
            decision.pods_placed_existing += placed
            if placed < seg.count:
                residuals.append((seg, seg.count - placed))

        # phase 2: virtual new nodes per pool
        units: Dict[int, List[np.ndarray]] = {k: [] for k in range(snap.num_pools)}
        max_new = {
            k: max(0, int(snap.pool_max[k]) - int(snap.pool_cur[k]))
            for k in range(snap.num_pools)
        }
        pool_residual = np.zeros((snap.num_pools, NUM_RESOURCES), dtype=_I64)

        for seg, remaining in residuals:
            cands = self.pool_candidates(seg)
            if not cands:
                decision.pods_unfulfillable += remaining
                decision.unfulfillable_detail.append(((int(seg.sel_hash) & ((1 << 64) - 1)), remaining))
                continue
            k = cands[0]
            pool_residual[k] += seg.req.astype(_I64) * remaining
            req = seg.req.astype(_I64)
            unit_cap = snap.pool_unit[k].astype(_I64)
            placed_new = 0
            # fill already-open virtual units first (uniform math per unit)
            for u in units[k]:
                if remaining == 0:
                    break
                m = self._unit_count(u, req)
                take = min(m, remaining)
                u -= req * take
                remaining -= take
                placed_new += take
            # open new units while allowed
            per_unit = self._unit_count(unit_cap, req)
            while remaining > 0 and per_unit > 0 and len(units[k]) < max_new[k]:
                u = unit_cap.copy()
                take = min(per_unit, remaining)
                u -= req * take
                units[k].append(u)
                remaining -= take
                placed_new += take
            decision.pods_placed_new += placed_new
            if remaining > 0:
                decision.pods_unfulfillable += remaining
                decision.unfulfillable_detail.append(((int(seg.sel_hash) & ((1 << 64) - 1)), remaining))

        for k in range(snap.num_pools):
            n_units = len(units[k])
            if n_units > 0 and self.over_provision > 0:
                n_units = min(n_units + self.over_provision, max_new[k])
            if n_units > 0:
                name = snap.pool_names[k]
                decision.units_needed[name] = n_units
                decision.new_pool_sizes[name] = int(snap.pool_cur[k]) + n_units
        decision.pool_residual_totals = pool_residual
        return decision

    @staticmethod
    def _unit_count(free: np.ndarray, req: np.ndarray) -> int:
        m = 1 << 31
        for r in range(NUM_RESOURCES):
            rr = int(req[r])
            if rr > 0:
                m = min(m, int(free[r]) // rr)
        return max(0, m)

    # ------------------------------------------------------------------
    # scale-down scoring
    # ------------------------------------------------------------------
    def maintain_partials(self) -> Dict[str, np.ndarray]:
        """Per-node accumulators over THIS snapshot's pods (a shard's
        partial when pods are sharded): raw integer sums/flags that
        all-reduce exactly (SUM for util/counts, MAX for flags).
        (Device form: drain_flags_kernel + segsum_mfma)"""
        snap = self.snap
        N = snap.num_nodes
        util = np.zeros((N, NUM_RESOURCES), dtype=_I64)
        blocked = np.zeros(N, np.int32)
        pdb_bad = np.zeros(N, np.int32)
        nontrivial = np.zeros(N, np.int32)  # non-mirror/ds resident pods
        drainable_pods = np.zeros(N, np.int32)

        for i in range(snap.num_pods):
            j = int(snap.pod_node[i])
            if j < 0:
                continue
            f = int(snap.pod_flags[i])
            util[j] += snap.pod_req[i].astype(_I64)
            if f & PodFlags.BLOCKS_DRAIN:
                blocked[j] = 1
            if not (f & (PodFlags.MIRROR | PodFlags.DAEMONSET)):
                nontrivial[j] += 1
                if f & PodFlags.DRAINABLE:
                    drainable_pods[j] += 1
                # drains never evict mirror/daemonset pods: only
                # evictable pods consume PDB budget (kernel D parity)
                d = int(snap.pod_pdb[i])
                if d >= 0 and int(snap.pdb_allowed[d]) < 1:
                    pdb_bad[j] = 1
        return dict(util=util, blocked=blocked, pdb_bad=pdb_bad,
                    nontrivial=nontrivial, drainable=drainable_pods)

    def finalize_maintain(
        self, partials: Dict[str, np.ndarray], util_threshold_permille: int = 300
    ) -> MaintainReport:
        """Permille utilization + candidate list from (possibly
        all-reduced) partials. (Device form: drain_finalize_kernel)"""
        snap = self.snap
        N = snap.num_nodes
        util = partials["util"]
        blocked = partials["blocked"]
        pdb_bad = partials["pdb_bad"]
        nontrivial = partials["nontrivial"]
        drainable_pods = partials["drainable"]

        # worst-axis permille, vectorized (pods axis is occupancy, not
        # utilization); integer floor division matches the device kernel
        alloc = snap.node_alloc.astype(_I64)
        u = util.astype(_I64)
        worst = np.zeros(N, dtype=_I64)
        for r in (0, 1, 3):
            a = alloc[:, r]
            with np.errstate(divide="ignore", invalid="ignore"):
                perm = np.where(a > 0, u[:, r] * 1000 // np.maximum(a, 1), 0)
            worst = np.maximum(worst, perm)
        util_permille = worst.astype(np.int32)

        idle = (nontrivial == 0).astype(np.uint8)
        report = MaintainReport(
            node_util_permille=util_permille,
            node_blocked=(blocked != 0).astype(np.uint8),
            node_pdb_ok=(pdb_bad == 0).astype(np.uint8),
            node_idle=idle,
            node_drainable_pods=drainable_pods.astype(np.int32),
        )
        # vectorized candidate predicate; python touches candidates only
        ok = (
            (report.node_blocked == 0)
            & (report.node_pdb_ok == 1)
            & (snap.node_pool >= 0)
            & ((idle == 1) | (util_permille < util_threshold_permille))
        )
        order = np.argsort(util_permille, kind="stable")
        for j in order[ok[order]]:
            j = int(j)
            report.candidates.append(
                DrainCandidate(
                    node_index=j,
                    node_name=snap.node_names[j] if snap.node_names else str(j),
                    util_permille=int(util_permille[j]),
                    drainable_pods=int(drainable_pods[j]),
                    idle=bool(idle[j]),
                )
            )
        return report

    def maintain_report(self, util_threshold_permille: int = 300) -> MaintainReport:
        """Single-process scale-down scoring [R: cluster.py · maintain]."""
        return self.finalize_maintain(self.maintain_partials(), util_threshold_permille)
