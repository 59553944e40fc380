"""Sharded controller engine: pods sharded across <=8 GPUs, one global
scale decision agreed over RCCL.

Protocol per decide():

  1. each rank classifies ITS pod shard and builds local placement
     segments (kernel C + device radix sort — the P-proportional work,
     which is what scales with ranks);
  2. `allgather_rows` of the per-rank segment rows (sel pairs + request
     + count: MAX_SELECTOR_PAIRS+8 i64 per segment, KB-sized, latency-bound);
  3. every rank merges counts by key into the IDENTICAL global segment
     list (same order as sim/oracle.py::merge_segment_counts);
  4. every rank runs the node-table placement (kernels A/B) on the
     replicated node table with the global counts -> decisions are
     bit-identical by construction (node tables are replicated;
     placement math is deterministic integer arithmetic);
  5. a final all-reduce(MIN)/all-reduce(MAX) over the per-pool units
     vector asserts the agreement (a cheap cross-rank determinism
     check, the test SURVEY §4.5 demands).

maintain(): per-rank partial per-node accumulators are all-reduced
(SUM for util/counts, MAX for the blocked/pdb flags) before the
finalize kernel — integer ops, exact under any shard split.

Works identically over gloo (CPU oracle engine, CI-tested here with
world_size 2) and nccl/RCCL (GPU engine on MI355X).
"""

from __future__ import annotations

import time

from typing import List

import numpy as np
import torch

from ..sim.decision import MaintainReport, ScaleDecision, Segment
from ..sim.engine import CpuEngine, SimulationEngine
from ..sim.oracle import OracleSimulator, build_segments
from ..snapshot.encode import MAX_SELECTOR_PAIRS, ClusterSnapshot
from . import dist as pdist

_U64 = (1 << 64) - 1
_ROW_COLS = MAX_SELECTOR_PAIRS + 8  # pairs + count + req(4) + seg_count + sel_hash + req_key


def _segments_to_rows(segs: List[Segment]) -> np.ndarray:
    rows = np.zeros((len(segs), _ROW_COLS), np.int64)
    for i, s in enumerate(segs):
        rows[i, :MAX_SELECTOR_PAIRS] = s.sel_pairs
        rows[i, MAX_SELECTOR_PAIRS] = s.sel_count
        rows[i, MAX_SELECTOR_PAIRS + 1 : MAX_SELECTOR_PAIRS + 5] = s.req
        rows[i, MAX_SELECTOR_PAIRS + 5] = s.count
        rows[i, MAX_SELECTOR_PAIRS + 6] = s.sel_hash
        rows[i, MAX_SELECTOR_PAIRS + 7] = s.req_key
    return rows


def _rows_to_segments(rows: np.ndarray) -> List[Segment]:
    out = []
    for r in rows:
        out.append(
            Segment(
                sel_hash=int(r[MAX_SELECTOR_PAIRS + 6]),
                sel_pairs=r[:MAX_SELECTOR_PAIRS].copy(),
                sel_count=int(r[MAX_SELECTOR_PAIRS]),
                req=r[MAX_SELECTOR_PAIRS + 1 : MAX_SELECTOR_PAIRS + 5].astype(np.int32),
                count=int(r[MAX_SELECTOR_PAIRS + 5]),
                req_key=int(r[MAX_SELECTOR_PAIRS + 7]),
            )
        )
    return out


def merge_rows(rows: np.ndarray) -> np.ndarray:
    """Sum counts of identical (sel_hash, req-key) rows; order by
    (sel_hash as uint64 asc, req desc) — vectorized equivalent of
    sim/oracle.py::merge_segment_counts."""
    if rows.shape[0] == 0:
        return rows
    sel_u = rows[:, MAX_SELECTOR_PAIRS + 6].astype(np.uint64)
    keys = rows[:, MAX_SELECTOR_PAIRS + 7].astype(np.uint64)
    order = np.lexsort((np.iinfo(np.uint64).max - keys, sel_u))
    rows = rows[order]
    sel_u = sel_u[order]
    keys = keys[order]
    boundary = np.ones(rows.shape[0], bool)
    boundary[1:] = (sel_u[1:] != sel_u[:-1]) | (keys[1:] != keys[:-1])
    group_id = np.cumsum(boundary) - 1
    n_groups = int(group_id[-1]) + 1
    counts = np.zeros(n_groups, np.int64)
    np.add.at(counts, group_id, rows[:, MAX_SELECTOR_PAIRS + 5])
    out = rows[boundary].copy()
    out[:, MAX_SELECTOR_PAIRS + 5] = counts
    return out


class ShardedEngine(SimulationEngine):
    """Wraps a CpuEngine or GpuEngine with the RCCL decision agreement.

    The snapshot passed to decide()/maintain() is the LOCAL shard:
    pods local to this rank; nodes/pools REPLICATED and identical
    across ranks (the caller shards pods, e.g. uid-hash mod world).
    """

    name = "sharded"

    def __init__(self, inner: SimulationEngine, check_agreement: bool = True):
        self.inner = inner
        self.check_agreement = check_agreement
        self.active = pdist.init_from_env()
        # SURVEY §5 observability: cumulative collective wall time
        self.comm_stats = {"allgather_s": 0.0, "allreduce_s": 0.0, "calls": 0}

    # ------------------------------------------------------------------
    def decide(self, snap: ClusterSnapshot, over_provision: int = 0) -> ScaleDecision:
        device = getattr(self.inner, "device", None)
        if isinstance(self.inner, CpuEngine) or device is None:
            local = build_segments(snap)
            rows = _segments_to_rows(local)
            t0 = time.perf_counter()
            rows = pdist.allgather_rows(rows)
            self.comm_stats["allgather_s"] += time.perf_counter() - t0
            merged = merge_rows(rows)
            segments = _rows_to_segments(merged)
            node_free = None
            if self.active:
                # globalize free capacity: usage partials sum exactly
                usage = torch.from_numpy(
                    (snap.node_alloc.astype(np.int64) - snap.node_free.astype(np.int64))
                )
                pdist.allreduce_tensor(usage, "sum")
                node_free = snap.node_alloc.astype(np.int64) - usage.numpy()
            sim = OracleSimulator(snap, over_provision=over_provision)
            decision = sim.fulfill_pending(segments, node_free=node_free)
        else:
            ds = self.inner.packer.pack(snap, device)
            decision = self._decide_ds(ds, snap.pool_names, over_provision)
        if self.active and self.check_agreement:
            t0 = time.perf_counter()
            self._assert_agreement(decision, snap.num_pools, snap.pool_names)
            self.comm_stats["allreduce_s"] += time.perf_counter() - t0
        self.comm_stats["calls"] += 1
        return decision

    # ------------------------------------------------------------------
    def _free_hook(self):
        """Globalize node free capacity inside the device placement:
        usage partials (alloc - local free) all-reduce exactly."""
        if not self.active:
            return None

        def free_hook(free, dsnap):
            usage = (dsnap.node_alloc - free).to(torch.int64)
            pdist.allreduce_tensor(usage, "sum")
            return (dsnap.node_alloc.to(torch.int64) - usage).clamp(
                min=-(2**31), max=2**31 - 1
            ).to(torch.int32)

        return free_hook

    def _decide_ds(self, ds, pool_names: List[str],
                   over_provision: int = 0) -> ScaleDecision:
        """Sharded decision over an already-packed device shard — the
        common core of decide() and the resident path."""
        device = self.inner.device
        hook = self._free_hook()
        decision = None
        # device merge: rows never leave HBM (the host merge cost
        # ~180 us/rank of serialize + numpy merge + ws reload);
        # needs device collectives (nccl/RCCL) or world 1
        device_merge = (
            hasattr(self.inner, "decide_rows_device")
            and str(device).startswith("cuda")
            and (not self.active or pdist.backend() == "nccl")
        )
        if device_merge:
            # no-sync local build: G stays on device, the FIXED-size
            # gather needs no size exchange, and a rank's local hash
            # overflow rides the poison row so every rank falls back
            # together
            self.inner._enqueue_hash_segments(ds, emit_rows=True)
            t0 = time.perf_counter()
            gathered = pdist.allgather_fixed_device(self.inner.ws_seg_rows)
            self.comm_stats["allgather_s"] += time.perf_counter() - t0
            decision = self.inner.decide_rows_device(
                ds, gathered, pool_names,
                over_provision=over_provision, free_hook=hook,
            )
        if decision is None:
            segs = self.inner.build_segments(ds)
            rows = self._device_segments_to_rows(ds, segs)
            t0 = time.perf_counter()
            rows = pdist.allgather_rows(rows, device=device)
            self.comm_stats["allgather_s"] += time.perf_counter() - t0
            merged = merge_rows(rows)
            decision = self.inner.decide_merged_rows(
                ds, merged, pool_names, over_provision=over_provision,
                free_hook=hook,
            )
            if decision is None:  # merged list exceeds the workspace
                gsegs = self._rows_to_device_segments(merged, device)
                decision = self.inner.decide_device(
                    ds, gsegs, pool_names, over_provision=over_provision,
                    free_hook=hook,
                )
        return decision

    # ------------------------------------------------------------------
    def decide_packed(self, ds, pool_names: List[str],
                      over_provision: int = 0) -> ScaleDecision:
        """Sharded decide over an ALREADY-PACKED device shard (HBM-
        resident benchmarks / callers that manage their own upload):
        local segment build -> fixed-size row gather -> device merge ->
        replicated placement, with the usual agreement check."""
        decision = self._decide_ds(ds, pool_names, over_provision)
        if self.active and self.check_agreement:
            t0 = time.perf_counter()
            self._assert_agreement(decision, len(pool_names), pool_names)
            self.comm_stats["allreduce_s"] += time.perf_counter() - t0
        self.comm_stats["calls"] += 1
        return decision

    # ------------------------------------------------------------------
    def decide_resident(self, cache, base: ClusterSnapshot,
                        over_provision: int = 0) -> ScaleDecision:
        """Sharded RESIDENT decide (VERDICT r01 #1b): each rank's watch
        cache holds its uid-hash pod shard (DeltaSnapshotCache
        .set_shard); dirty rows scatter into the HBM-resident shard
        (kernel E), segments build locally, and the fixed-size row
        gather + device merge produce the identical global decision on
        every rank — no full-snapshot upload at any world size."""
        device = getattr(self.inner, "device", None)
        if (device is None or not hasattr(self.inner, "resident_ds")
                or base.num_pools > 64):
            # CPU engine (gloo test rig) or pool-count cliff: decide
            # over the host-built shard snapshot (same agreement path)
            snap = cache.build_snapshot(base)
            return self.decide(snap, over_provision)
        ds = self.inner.resident_ds(cache, base)
        decision = self._decide_ds(ds, base.pool_names, over_provision)
        if self.active and self.check_agreement:
            t0 = time.perf_counter()
            self._assert_agreement(decision, base.num_pools, base.pool_names)
            self.comm_stats["allreduce_s"] += time.perf_counter() - t0
        self.comm_stats["calls"] += 1
        return decision

    @staticmethod
    def _device_segments_to_rows(ds, segs) -> np.ndarray:
        """Serialize the segment SoA into host rows with ONE device->host
        copy (a per-column .cpu() costs a sync each; this path runs every
        sharded decide)."""
        G = segs.G
        if not G:
            return np.zeros((0, _ROW_COLS), np.int64)
        dev_rows = torch.empty((G, _ROW_COLS), dtype=torch.int64,
                               device=segs.seg_req.device)
        dev_rows[:, :MAX_SELECTOR_PAIRS] = segs.seg_sel_pairs
        dev_rows[:, MAX_SELECTOR_PAIRS] = segs.seg_sel_count.to(torch.int64)
        dev_rows[:, MAX_SELECTOR_PAIRS + 1 : MAX_SELECTOR_PAIRS + 5] = segs.seg_req.to(torch.int64)
        dev_rows[:, MAX_SELECTOR_PAIRS + 5] = segs.seg_count
        dev_rows[:, MAX_SELECTOR_PAIRS + 6] = segs.seg_sel_hash
        dev_rows[:, MAX_SELECTOR_PAIRS + 7] = segs.seg_key
        return dev_rows.cpu().numpy()

    @staticmethod
    def _rows_to_device_segments(rows: np.ndarray, device):
        from ..ops.engine_gpu import DeviceSegments

        G = rows.shape[0]
        e64 = torch.zeros(0, dtype=torch.int64, device=device)
        return DeviceSegments(
            seg_sel_pairs=torch.from_numpy(
                np.ascontiguousarray(rows[:, :MAX_SELECTOR_PAIRS])
            ).to(device),
            seg_sel_count=torch.from_numpy(
                rows[:, MAX_SELECTOR_PAIRS].astype(np.int32)
            ).to(device),
            seg_sel_hash=torch.from_numpy(
                np.ascontiguousarray(rows[:, MAX_SELECTOR_PAIRS + 6])
            ).to(device),
            seg_req=torch.from_numpy(
                np.ascontiguousarray(rows[:, MAX_SELECTOR_PAIRS + 1 : MAX_SELECTOR_PAIRS + 5].astype(np.int32))
            ).to(device),
            seg_count=torch.from_numpy(
                np.ascontiguousarray(rows[:, MAX_SELECTOR_PAIRS + 5])
            ).to(device),
            seg_key=torch.from_numpy(
                np.ascontiguousarray(rows[:, MAX_SELECTOR_PAIRS + 7])
            ).to(device),
            pod_sorted=e64,
            pod_seg=e64,
        )

    def _assert_agreement(self, decision: ScaleDecision, num_pools: int,
                          pool_names: List[str]):
        """All-reduce the per-pool units vector with MIN and MAX; any
        divergence across ranks is a hard error (determinism bug)."""
        K = num_pools
        vec = torch.zeros(K + 3, dtype=torch.int64)
        for k, name in enumerate(pool_names):
            vec[k] = decision.units_needed.get(name, 0)
        vec[K] = decision.pods_placed_existing
        vec[K + 1] = decision.pods_placed_new
        vec[K + 2] = decision.pods_unfulfillable
        device = getattr(self.inner, "device", None)
        if device is not None and str(device).startswith("cuda"):
            vec = vec.to(device)
        lo = vec.clone()
        hi = vec.clone()
        pdist.allreduce_tensor(lo, "min")
        pdist.allreduce_tensor(hi, "max")
        if not torch.equal(lo, hi):
            raise RuntimeError(
                f"sharded decision divergence: min={lo.tolist()} max={hi.tolist()}"
            )

    # ------------------------------------------------------------------
    def maintain(self, snap: ClusterSnapshot, util_threshold_permille: int = 300) -> MaintainReport:
        device = getattr(self.inner, "device", None)
        if isinstance(self.inner, CpuEngine) or device is None:
            sim = OracleSimulator(snap)
            partials = sim.maintain_partials()
            if self.active:
                # RAW partials reduce exactly: SUM for util/counts,
                # MAX for flags; permille division happens AFTER.
                t = {k: torch.from_numpy(v.astype(np.int64)) for k, v in partials.items()}
                pdist.allreduce_tensor(t["util"], "sum")
                pdist.allreduce_tensor(t["blocked"], "max")
                pdist.allreduce_tensor(t["pdb_bad"], "max")
                pdist.allreduce_tensor(t["nontrivial"], "sum")
                pdist.allreduce_tensor(t["drainable"], "sum")
                partials = {k: v.numpy() for k, v in t.items()}
            return sim.finalize_maintain(partials, util_threshold_permille)

        ds = self.inner.packer.pack(snap, device)
        _, tensors = self.inner.maintain_device(ds, util_threshold_permille)
        if self.active:
            pdist.allreduce_tensor(tensors["util"], "sum")
            pdist.allreduce_tensor(tensors["blocked"], "max")
            pdist.allreduce_tensor(tensors["pdb_bad"], "max")
            pdist.allreduce_tensor(tensors["nontrivial"], "sum")
            pdist.allreduce_tensor(tensors["drainable"], "sum")
        rep = self.inner.finalize_maintain(ds, tensors, util_threshold_permille)
        rep.candidates = self.inner._candidates(rep, snap.node_names, snap, util_threshold_permille)
        return rep
