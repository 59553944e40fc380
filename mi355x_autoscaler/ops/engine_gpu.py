"""GPU simulation engine: the HIP kernel pipeline on a device-resident
cluster snapshot.

Pipeline per decide() — all stages on-device, one small readback of the
decision vector at the end:

  upload SoA          host -> HBM (288 GB/GPU: million-pod traces fit)
  classify_keys       kernel C  (radix keys realizing the oracle order)
  sort + unique       device radix sort (torch) -> placement segments
  node_free_subtract  free = allocatable - bound requests (atomics)
  fit_eligibility     kernel A  (LDS-staged label sets, 64-bit ballots)
  seg_pool_route      segment -> agent pool
  segsum_mfma         per-pool pending demand (MFMA segmented sums)
  binpack_place       kernel B  (FFD as uniform scan-threshold steps)
  binpack_units       kernel B' (virtual new nodes -> units per pool)

maintain() runs segsum_mfma (per-node utilization) + kernel D.

Decisions are bit-identical to sim/oracle.py by construction (same
lattice, same order, same integer arithmetic); tests/test_gpu_engine.py
asserts it on hardware.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional

import numpy as np
import torch

from ..sim.decision import DrainCandidate, MaintainReport, ScaleDecision
from ..sim.engine import SimulationEngine
from ..snapshot.encode import ClusterSnapshot, PodFlags
from . import load_ops

_PACK_MAX = 1 << 52
_SEGSUM_CHUNK = 256  # exactness bound for the f32 MFMA limb chains


@dataclass
class DeviceSnapshot:
    """The SoA snapshot resident in HBM."""

    pod_req: torch.Tensor        # [P,4] i32
    pod_flags: torch.Tensor      # [P]   i32
    pod_sel_hash: torch.Tensor   # [P]   i64
    pod_sel_pairs: torch.Tensor  # [P,8] i64
    pod_sel_count: torch.Tensor  # [P]   i32
    pod_node: torch.Tensor       # [P]   i32
    pod_pdb: torch.Tensor        # [P]   i32
    node_alloc: torch.Tensor     # [N,4] i32
    node_labels: torch.Tensor    # [N,32] i64
    node_label_count: torch.Tensor  # [N] i32
    node_unsched: torch.Tensor   # [N] i32
    node_pool: torch.Tensor      # [N] i32
    pool_unit: torch.Tensor      # [K,4] i32
    pool_max: torch.Tensor       # [K] i32
    pool_cur: torch.Tensor       # [K] i32
    pool_labels: torch.Tensor    # [K,32] i64
    pool_label_count: torch.Tensor  # [K] i32
    pdb_allowed: torch.Tensor    # [D] i32

    @property
    def P(self) -> int:
        return int(self.pod_req.shape[0])

    @property
    def N(self) -> int:
        return int(self.node_alloc.shape[0])

    @property
    def K(self) -> int:
        return int(self.pool_unit.shape[0])


def upload_snapshot(snap: ClusterSnapshot, device) -> DeviceSnapshot:
    def i32(a):
        return torch.from_numpy(np.ascontiguousarray(a.astype(np.int32))).to(device, non_blocking=True)

    def i64(a):
        return torch.from_numpy(np.ascontiguousarray(a.astype(np.int64))).to(device, non_blocking=True)

    return DeviceSnapshot(
        pod_req=i32(snap.pod_req),
        pod_flags=i32(snap.pod_flags),
        pod_sel_hash=i64(snap.pod_sel_hash),
        pod_sel_pairs=i64(snap.pod_sel_pairs),
        pod_sel_count=i32(snap.pod_sel_count),
        pod_node=i32(snap.pod_node),
        pod_pdb=i32(snap.pod_pdb),
        node_alloc=i32(snap.node_alloc),
        node_labels=i64(snap.node_labels),
        node_label_count=i32(snap.node_label_count),
        node_unsched=i32(snap.node_unsched),
        node_pool=i32(snap.node_pool),
        pool_unit=i32(snap.pool_unit),
        pool_max=i32(snap.pool_max),
        pool_cur=i32(snap.pool_cur),
        pool_labels=i64(snap.pool_labels),
        pool_label_count=i32(snap.pool_label_count),
        pdb_allowed=i32(snap.pdb_allowed),
    )


@dataclass
class DeviceSegments:
    """Placement segments resident on device (global oracle order)."""

    seg_sel_pairs: torch.Tensor  # [G,8] i64
    seg_sel_count: torch.Tensor  # [G] i32
    seg_sel_hash: torch.Tensor   # [G] i64
    seg_req: torch.Tensor        # [G,4] i32
    seg_count: torch.Tensor      # [G] i64
    pod_sorted: torch.Tensor     # [M] i64 pending pod ids, segment-sorted
    pod_seg: torch.Tensor        # [M] i64 segment id per sorted pod

    @property
    def G(self) -> int:
        return int(self.seg_req.shape[0])


def build_segments_device(ds: DeviceSnapshot, ops) -> DeviceSegments:
    device = ds.pod_req.device
    P = ds.P
    key_sel = torch.empty(P, dtype=torch.int64, device=device)
    key_req = torch.empty(P, dtype=torch.int64, device=device)
    pending = torch.empty(P, dtype=torch.int32, device=device)
    ops.classify_keys(ds.pod_flags, ds.pod_sel_hash, ds.pod_req, key_sel, key_req, pending)

    pend_idx = torch.nonzero(pending, as_tuple=False).flatten()
    M = int(pend_idx.numel())
    if M == 0:
        z64 = torch.zeros(0, dtype=torch.int64, device=device)
        return DeviceSegments(
            seg_sel_pairs=torch.zeros((0, 8), dtype=torch.int64, device=device),
            seg_sel_count=torch.zeros(0, dtype=torch.int32, device=device),
            seg_sel_hash=z64,
            seg_req=torch.zeros((0, 4), dtype=torch.int32, device=device),
            seg_count=z64,
            pod_sorted=z64,
            pod_seg=z64,
        )

    ks = key_sel[pend_idx]
    kr = key_req[pend_idx]
    # (sel asc, req-key asc[=req desc], pod idx asc): two stable passes
    order1 = torch.argsort(kr, stable=True)
    ks1 = ks[order1]
    order2 = torch.argsort(ks1, stable=True)
    order = order1[order2]
    pod_sorted = pend_idx[order].to(torch.int64)

    keys2 = torch.stack([ks[order], kr[order]], dim=1)  # [M,2] sorted rows
    uniq, inverse, counts = torch.unique_consecutive(
        keys2, dim=0, return_inverse=True, return_counts=True
    )
    G = int(uniq.shape[0])
    starts = torch.zeros(G, dtype=torch.int64, device=device)
    if G > 1:
        starts[1:] = torch.cumsum(counts, 0)[:-1]
    first_pod = pod_sorted[starts]

    return DeviceSegments(
        seg_sel_pairs=ds.pod_sel_pairs[first_pod],
        seg_sel_count=ds.pod_sel_count[first_pod],
        seg_sel_hash=ds.pod_sel_hash[first_pod],
        seg_req=ds.pod_req[first_pod],
        seg_count=counts.to(torch.int64),
        pod_sorted=pod_sorted,
        pod_seg=inverse.to(torch.int64),
    )


def segsum_workitems(seg_of: torch.Tensor, num_segments: int):
    """Build (window, start, end) work items for segsum_mfma over an
    ascending segment-id array: windows of 16 segments, chunks of
    <=_SEGSUM_CHUNK elements that never cross a window boundary."""
    device = seg_of.device
    W = (num_segments + 15) // 16
    win_of = torch.div(seg_of, 16, rounding_mode="floor")
    cnt = torch.bincount(win_of, minlength=W)
    starts = torch.zeros(W, dtype=torch.int64, device=device)
    if W > 1:
        starts[1:] = torch.cumsum(cnt, 0)[:-1]
    nchunks = (cnt + _SEGSUM_CHUNK - 1) // _SEGSUM_CHUNK
    T = int(nchunks.sum().item())
    if T == 0:
        e = torch.zeros(0, dtype=torch.int64, device=device)
        return e.to(torch.int32), e, e
    wins = torch.arange(W, device=device)
    wi_win = torch.repeat_interleave(wins, nchunks)
    chunk_base = torch.zeros(W, dtype=torch.int64, device=device)
    if W > 1:
        chunk_base[1:] = torch.cumsum(nchunks, 0)[:-1]
    ci = torch.arange(T, device=device) - torch.repeat_interleave(chunk_base, nchunks)
    wi_start = starts[wi_win] + ci * _SEGSUM_CHUNK
    wi_end = torch.minimum(wi_start + _SEGSUM_CHUNK, starts[wi_win] + cnt[wi_win])
    return wi_win.to(torch.int32), wi_start, wi_end


def segsum(values: torch.Tensor, seg_of_sorted: torch.Tensor, num_segments: int, ops):
    """MFMA segmented sum -> [S,4] int64 (values >= 0, seg ids sorted)."""
    device = values.device
    out = torch.zeros((num_segments, 4), dtype=torch.int64, device=device)
    if values.shape[0] == 0 or num_segments == 0:
        return out
    wi_win, wi_start, wi_end = segsum_workitems(seg_of_sorted, num_segments)
    ops.segsum_mfma(
        values.contiguous(), seg_of_sorted.to(torch.int32).contiguous(),
        wi_win.contiguous(), wi_start.contiguous(), wi_end.contiguous(),
        num_segments, out,
    )
    return out


class GpuEngine(SimulationEngine):
    name = "gpu"

    def __init__(self, device: str = "cuda:0"):
        self.device = torch.device(device)
        if not torch.cuda.is_available():
            raise RuntimeError("GpuEngine requires a ROCm device (torch.cuda unavailable)")
        self.ops = load_ops()
        self._last_demand: Optional[torch.Tensor] = None

    # ------------------------------------------------------------------
    def decide(self, snap: ClusterSnapshot, over_provision: int = 0) -> ScaleDecision:
        ds = upload_snapshot(snap, self.device)
        segs = build_segments_device(ds, self.ops)
        return self.decide_device(ds, segs, snap.pool_names, over_provision)

    def decide_device(
        self,
        ds: DeviceSnapshot,
        segs: DeviceSegments,
        pool_names: List[str],
        over_provision: int = 0,
        seg_count_override: Optional[torch.Tensor] = None,
        free_hook=None,
    ) -> ScaleDecision:
        ops = self.ops
        device = self.device
        N, K, G = ds.N, ds.K, segs.G
        decision = ScaleDecision()
        if G == 0:
            decision.pool_residual_totals = np.zeros((K, 4), np.int64)
            return decision

        seg_count = seg_count_override if seg_count_override is not None else segs.seg_count

        # free capacity (local bound pods; the sharded engine's hook
        # all-reduces usage across ranks so free is globally correct)
        free = ds.node_alloc.clone()
        ops.node_free_subtract(ds.pod_node, ds.pod_req, free)
        if free_hook is not None:
            free = free_hook(free, ds)

        # eligibility + pool routing
        W = (N + 63) // 64
        elig = torch.zeros((G, max(W, 1)), dtype=torch.int64, device=device)
        if N > 0:
            ops.fit_eligibility(
                ds.node_labels, ds.node_label_count, ds.node_unsched,
                segs.seg_sel_pairs.contiguous(), segs.seg_sel_count.contiguous(), elig,
            )
        seg_pool = torch.empty(G, dtype=torch.int32, device=device)
        ops.seg_pool_route(
            segs.seg_sel_pairs.contiguous(), segs.seg_sel_count.contiguous(),
            segs.seg_req.contiguous(), ds.pool_labels, ds.pool_label_count,
            ds.pool_unit, seg_pool,
        )

        # MFMA per-pool pending demand (local-shard pods)
        if segs.pod_sorted.numel() and K > 0:
            pod_pool = seg_pool.to(torch.int64)[segs.pod_seg]
            keep = pod_pool >= 0
            pp = pod_pool[keep]
            vals = ds.pod_req[segs.pod_sorted[keep]]
            order = torch.argsort(pp, stable=True)
            demand = segsum(vals[order].contiguous(), pp[order], K, ops)
        else:
            demand = torch.zeros((K, 4), dtype=torch.int64, device=device)
        self._last_demand = demand

        # FFD placement over existing nodes
        placed = torch.zeros(G, dtype=torch.int64, device=device)
        if N > 0:
            ops.binpack_place(free, elig, segs.seg_req.contiguous(), seg_count.contiguous(), placed)

        # units phase
        max_new = torch.clamp(ds.pool_max - ds.pool_cur, min=0)
        maxu = max(int(max_new.max().item()) if K else 0, 1)
        units_free = torch.zeros((max(K, 1), maxu, 4), dtype=torch.int64, device=device)
        units_cnt = torch.zeros(max(K, 1), dtype=torch.int32, device=device)
        placed_new = torch.zeros(G, dtype=torch.int64, device=device)
        unful = torch.zeros(G, dtype=torch.int64, device=device)
        pool_residual = torch.zeros((max(K, 1), 4), dtype=torch.int64, device=device)
        ops.binpack_units(
            seg_pool, segs.seg_req.contiguous(), seg_count.contiguous(), placed,
            ds.pool_unit, max_new.to(torch.int32).contiguous(),
            units_free, units_cnt, placed_new, unful, pool_residual,
        )

        # single small readback
        placed_h = placed.cpu().numpy()
        placed_new_h = placed_new.cpu().numpy()
        unful_h = unful.cpu().numpy()
        units_h = units_cnt.cpu().numpy()
        max_new_h = max_new.cpu().numpy()
        pool_cur_h = ds.pool_cur.cpu().numpy()
        residual_h = pool_residual.cpu().numpy().astype(np.int64)
        sel_hash_h = segs.seg_sel_hash.cpu().numpy()

        decision.pods_placed_existing = int(placed_h.sum())
        decision.pods_placed_new = int(placed_new_h.sum())
        decision.pods_unfulfillable = int(unful_h.sum())
        for g in np.nonzero(unful_h > 0)[0]:
            decision.unfulfillable_detail.append(
                (int(sel_hash_h[g]) & ((1 << 64) - 1), int(unful_h[g]))
            )
        for k in range(K):
            n_units = int(units_h[k])
            if n_units > 0 and over_provision > 0:
                n_units = min(n_units + over_provision, int(max_new_h[k]))
            if n_units > 0:
                decision.units_needed[pool_names[k]] = n_units
                decision.new_pool_sizes[pool_names[k]] = int(pool_cur_h[k]) + n_units
        decision.pool_residual_totals = residual_h[:K]
        return decision

    # ------------------------------------------------------------------
    def maintain(self, snap: ClusterSnapshot, util_threshold_permille: int = 300) -> MaintainReport:
        ds = upload_snapshot(snap, self.device)
        rep, _ = self.maintain_device(ds, util_threshold_permille)
        rep.candidates = self._candidates(rep, snap.node_names, snap, util_threshold_permille)
        return rep

    def maintain_device(self, ds: DeviceSnapshot, util_threshold_permille: int = 300):
        """Device passes only; returns (MaintainReport without
        candidates, device tensors dict) so the sharded engine can
        all-reduce the partials before finalizing."""
        ops = self.ops
        device = self.device
        N = ds.N
        blocked = torch.zeros(N, dtype=torch.int32, device=device)
        pdb_bad = torch.zeros(N, dtype=torch.int32, device=device)
        nontrivial = torch.zeros(N, dtype=torch.int32, device=device)
        drainable = torch.zeros(N, dtype=torch.int32, device=device)
        pdb_allowed = ds.pdb_allowed if ds.pdb_allowed.numel() else torch.zeros(
            1, dtype=torch.int32, device=device
        )
        ops.drain_flags(ds.pod_node, ds.pod_flags, ds.pod_pdb, pdb_allowed,
                        blocked, pdb_bad, nontrivial, drainable)

        bound = torch.nonzero(ds.pod_node >= 0, as_tuple=False).flatten()
        if bound.numel() and N > 0:
            nodes_of = ds.pod_node[bound].to(torch.int64)
            order = torch.argsort(nodes_of, stable=True)
            util = segsum(ds.pod_req[bound][order].contiguous(), nodes_of[order], N, self.ops)
        else:
            util = torch.zeros((N, 4), dtype=torch.int64, device=device)

        tensors = dict(util=util, blocked=blocked, pdb_bad=pdb_bad,
                       nontrivial=nontrivial, drainable=drainable)
        rep = self.finalize_maintain(ds, tensors, util_threshold_permille)
        return rep, tensors

    def finalize_maintain(self, ds: DeviceSnapshot, tensors, util_threshold_permille: int):
        ops = self.ops
        N = ds.N
        device = self.device
        util_permille = torch.zeros(N, dtype=torch.int32, device=device)
        candidate = torch.zeros(N, dtype=torch.int32, device=device)
        if N > 0:
            ops.drain_finalize(
                tensors["util"], ds.node_alloc, ds.node_pool, tensors["blocked"],
                tensors["pdb_bad"], tensors["nontrivial"],
                util_threshold_permille, util_permille, candidate,
            )
        return MaintainReport(
            node_util_permille=util_permille.cpu().numpy(),
            node_blocked=tensors["blocked"].cpu().numpy().astype(np.uint8),
            node_pdb_ok=(tensors["pdb_bad"].cpu().numpy() == 0).astype(np.uint8),
            node_idle=(tensors["nontrivial"].cpu().numpy() == 0).astype(np.uint8),
            node_drainable_pods=tensors["drainable"].cpu().numpy(),
        )

    @staticmethod
    def _candidates(rep: MaintainReport, node_names, snap, util_threshold_permille):
        out = []
        for j in np.argsort(rep.node_util_permille, kind="stable"):
            j = int(j)
            if rep.node_blocked[j] or not rep.node_pdb_ok[j] or snap.node_pool[j] < 0:
                continue
            if rep.node_util_permille[j] >= util_threshold_permille and not rep.node_idle[j]:
                continue
            out.append(
                DrainCandidate(
                    node_index=j,
                    node_name=node_names[j] if node_names else str(j),
                    util_permille=int(rep.node_util_permille[j]),
                    drainable_pods=int(rep.node_drainable_pods[j]),
                    idle=bool(rep.node_idle[j]),
                )
            )
        return out
