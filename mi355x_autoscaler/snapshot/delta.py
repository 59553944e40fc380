"""Watch-based delta ingest: apply pod watch events to a cached SoA
snapshot instead of re-listing/re-encoding the world every cycle.

The reference re-listed everything per cycle [R: autoscaler/cluster.py
· loop_logic — reconstructed]; at 1M pods the relist dominates wall
time, so this cache is the host half of SURVEY §2.2 row E
(snapshot_diff): events mutate SoA rows in place (tombstoned deletes,
slot reuse, amortized compaction), and the per-cycle delta — the set
of touched row indices — is what the device scatter kernel
(ops/hip/snapshot_diff.hip) applies to the HBM-resident snapshot.

Checkpoint/resume: `save()`/`load()` spill the cached SoA
to an .npz so a 1M-pod replay can resume mid-burst.
"""

from __future__ import annotations

import logging
from typing import Dict, List, Optional, Set, Tuple

import numpy as np

from ..models.kube import KubePod
from .encode import (
    MAX_POD_LABEL_PAIRS,
    MAX_SELECTOR_PAIRS,
    ClusterSnapshot,
    PodFlags,
    encode_snapshot,
)

logger = logging.getLogger(__name__)

_GROW = 1.5
_COMPACT_DEAD_FRACTION = 0.5

_POD_COLS = ("pod_req", "pod_flags", "pod_sel_hash", "pod_sel_pairs",
             "pod_sel_count", "pod_node", "pod_pdb",
             "pod_label_pairs", "pod_label_count")


def _norm_matchers(seq) -> List[Tuple[str, Dict[str, str]]]:
    """Normalize PDB matchers to [(namespace, matchLabels)] — a v1 PDB
    selects pods in ITS OWN namespace only. Bare label dicts (legacy
    call shape) normalize to namespace 'default'."""
    out: List[Tuple[str, Dict[str, str]]] = []
    for s in (seq or []):
        if isinstance(s, dict):
            out.append(("default", dict(s)))
        else:
            ns, labels = s
            out.append((str(ns), dict(labels)))
    return out


class DeltaSnapshotCache:
    """Maintains the pod half of a ClusterSnapshot under watch events.

    Node/pool arrays come from a (cheap) per-cycle re-encode of the few
    node objects; pod columns live here and are patched row-wise.
    """

    def __init__(self):
        self.cap = 0
        self.size = 0
        self.uid_row: Dict[str, int] = {}
        self.free_rows: List[int] = []
        self.dirty: Set[int] = set()
        self.dirty_all = False  # cheaper than a size-wide dirty set
        self.arrays: Dict[str, np.ndarray] = {}
        self.pod_uids: List[str] = []
        self.pod_refs: List[Tuple[str, str]] = []
        self.node_index: Dict[str, int] = {}
        self.pdb_selectors: List[Dict[str, str]] = []
        self.has_label_pairs = False  # set by prime()/load()
        # uid-hash shard filter (sharded resident controller): when
        # set, only pods with shard_for(uid) == shard_rank are cached;
        # watch events for other ranks' pods are dropped at apply.
        self.shard_rank = 0
        self.shard_world = 1

    def set_shard(self, rank: int, world: int):
        """Restrict this cache to one uid-hash shard (prime() must be
        fed the already-sharded snapshot — parallel.elastic
        .shard_snapshot uses the same fnv1a64(uid) %% world hash)."""
        self.shard_rank = int(rank)
        self.shard_world = max(1, int(world))

    def _in_shard(self, uid: str) -> bool:
        if self.shard_world <= 1:
            return True
        from ..parallel.elastic import shard_for

        return shard_for(uid, self.shard_world) == self.shard_rank

    # ------------------------------------------------------------------
    def prime(self, snap: ClusterSnapshot, node_index: Dict[str, int],
              pdb_selectors: Optional[List[Dict[str, str]]] = None):
        """Seed from a full list/encode."""
        self.size = snap.num_pods
        self.cap = max(64, int(self.size * _GROW))
        self.arrays = {}
        # pod_label_* may be absent on the priming snapshot (synthetic
        # SoA traces, shards): cache zeros and refuse reassign_pdbs.
        self.has_label_pairs = (
            snap.pod_label_pairs.shape[0] == self.size)
        for col in _POD_COLS:
            src = getattr(snap, col)
            if src.shape[0] != self.size:  # absent optional column
                src = np.zeros((self.size,) + src.shape[1:], src.dtype)
            shape = (self.cap,) + src.shape[1:]
            a = np.zeros(shape, src.dtype)
            a[: self.size] = src
            if col in ("pod_node", "pod_pdb"):
                a[self.size :] = -1  # padding rows must be UNBOUND
            self.arrays[col] = a
        # pad by ACTUAL list length: snapshots without uid bookkeeping
        # (synthetic SoA) would otherwise leave the lists shorter than
        # cap and row writes past len() would raise
        self.pod_uids = list(snap.pod_uids) + [""] * (self.cap - len(snap.pod_uids))
        self.pod_refs = list(snap.pod_refs) + [("", "")] * (self.cap - len(snap.pod_refs))
        self.uid_row = {u: i for i, u in enumerate(snap.pod_uids) if u}
        self.free_rows = []
        self.dirty = set()
        self.dirty_all = False
        self.node_index = dict(node_index)
        self.pdb_selectors = _norm_matchers(pdb_selectors)

    # ------------------------------------------------------------------
    def _encode_row(self, pod: KubePod):
        """One pod -> lattice row (same rules as encode_snapshot)."""
        from .encode import _hash_pairs, quantize_request

        req = np.array(quantize_request(pod.resources), np.int32)
        pairs, cnt, combined = _hash_pairs(
            pod.selectors, MAX_SELECTOR_PAIRS, poison_overflow=True)
        lpairs, lcnt, _ = _hash_pairs(pod.labels, MAX_POD_LABEL_PAIRS)
        f = 0
        if pod.is_pending_unschedulable:
            f |= PodFlags.PENDING
        if pod.is_mirrored:
            f |= PodFlags.MIRROR
        if pod.is_daemonset:
            f |= PodFlags.DAEMONSET
        if pod.is_critical:
            f |= PodFlags.CRITICAL
        if pod.is_replicated:
            f |= PodFlags.REPLICATED
        if pod.is_drainable:
            f |= PodFlags.DRAINABLE
        if pod.blocks_drain:
            f |= PodFlags.BLOCKS_DRAIN
        if cnt > MAX_SELECTOR_PAIRS:
            f |= PodFlags.SELECTOR_OVERFLOW
        node = -1
        if pod.node_name and pod.node_name in self.node_index:
            node = self.node_index[pod.node_name]
            f |= PodFlags.BOUND
        pdb = -1
        for d, (pns, sel) in enumerate(self.pdb_selectors):
            if (sel and pns == pod.namespace
                    and all(pod.labels.get(k) == v for k, v in sel.items())):
                pdb = d
                break
        return req, pairs, min(cnt, MAX_SELECTOR_PAIRS), combined, f, node, pdb, lpairs, lcnt

    def _write_row(self, row: int, pod: KubePod):
        req, pairs, cnt, combined, f, node, pdb, lpairs, lcnt = self._encode_row(pod)
        a = self.arrays
        a["pod_req"][row] = req
        a["pod_sel_pairs"][row] = pairs
        a["pod_sel_count"][row] = cnt
        a["pod_sel_hash"][row] = combined
        a["pod_flags"][row] = f
        a["pod_node"][row] = node
        a["pod_pdb"][row] = pdb
        a["pod_label_pairs"][row] = lpairs
        a["pod_label_count"][row] = lcnt
        self.pod_uids[row] = pod.uid
        self.pod_refs[row] = (pod.namespace, pod.name)
        self.dirty.add(row)

    def _clear_row(self, row: int):
        for col in _POD_COLS:
            self.arrays[col][row] = 0
        self.arrays["pod_node"][row] = -1
        self.arrays["pod_pdb"][row] = -1
        self.pod_uids[row] = ""
        self.pod_refs[row] = ("", "")
        self.dirty.add(row)

    def _grow(self):
        new_cap = max(64, int(self.cap * _GROW) + 1)
        for col, a in self.arrays.items():
            b = np.zeros((new_cap,) + a.shape[1:], a.dtype)
            b[: self.cap] = a
            if col in ("pod_node", "pod_pdb"):
                b[self.cap :] = -1
            self.arrays[col] = b
        self.pod_uids += [""] * (new_cap - self.cap)
        self.pod_refs += [("", "")] * (new_cap - self.cap)
        self.cap = new_cap

    # ------------------------------------------------------------------
    def apply_event(self, event: Dict):
        etype = event.get("type")
        obj = event.get("object") or {}
        pod = KubePod(obj)
        if not self._in_shard(pod.uid):
            return  # another rank's pod (sharded resident mode)
        if etype in ("ADDED", "MODIFIED"):
            row = self.uid_row.get(pod.uid)
            if row is None:
                if self.free_rows:
                    row = self.free_rows.pop()
                else:
                    if self.size >= self.cap:
                        self._grow()
                    row = self.size
                    self.size += 1
                self.uid_row[pod.uid] = row
            self._write_row(row, pod)
        elif etype == "DELETED":
            row = self.uid_row.pop(pod.uid, None)
            if row is not None:
                self._clear_row(row)
                self.free_rows.append(row)

    def apply_batch(self, decoded: Dict) -> int:
        """Apply a C++-decoded watch batch (decode_watch_events output):
        row resolution runs in C (resolve_batch_rows) when the native
        extension is present, the column writes are vectorized; the
        Python loop below remains the reference/fallback (differential-
        tested). Returns events applied."""
        types = decoded["types"]
        n = int(types.shape[0])
        if n == 0:
            return 0
        if self.shard_world > 1:
            decoded = self._filter_shard(decoded)
            types = decoded["types"]
            n = int(types.shape[0])
            if n == 0:
                return 0
        if self.has_label_pairs and "pod_label_pairs" not in decoded:
            # a batch without label columns (mismatched/absent compiled
            # codec) would leave stale label hashes that reassign_pdbs
            # trusts — disable in-place reassign, forcing the re-prime
            # fallback instead of silently wrong PDB matching
            logger.warning(
                "watch batch lacks pod label columns; disabling in-place "
                "PDB reassignment (re-prime fallback)")
            self.has_label_pairs = False
        resolve = None
        try:
            import torch  # noqa: F401  (extension links against libtorch)

            from mi355x_autoscaler import _hip_ops

            resolve = _hip_ops.resolve_batch_rows
        except (ImportError, AttributeError):
            pass
        if resolve is not None:
            # pre-grow so capacity can never run out mid-batch
            n_upserts = int((types == 0).sum())
            while (self.cap - self.size) + len(self.free_rows) < n_upserts:
                self._grow()
            rows, sel, cleared, new_size = resolve(
                types, decoded["pod_uids"], decoded["pod_refs"],
                self.uid_row, self.free_rows, self.size, self.cap,
                self.pod_uids, self.pod_refs,
            )
            self.size = int(new_size)
            if cleared.size:
                for col in _POD_COLS:
                    self.arrays[col][cleared] = 0
                self.arrays["pod_node"][cleared] = -1
                self.arrays["pod_pdb"][cleared] = -1
                self.dirty.update(cleared.tolist())
            if rows.size:
                a = self.arrays
                a["pod_req"][rows] = decoded["pod_req"][sel]
                a["pod_flags"][rows] = decoded["pod_flags"][sel]
                a["pod_sel_hash"][rows] = decoded["pod_sel_hash"][sel]
                a["pod_sel_pairs"][rows] = decoded["pod_sel_pairs"][sel]
                a["pod_sel_count"][rows] = decoded["pod_sel_count"][sel]
                a["pod_node"][rows] = decoded["pod_node"][sel]
                a["pod_pdb"][rows] = decoded["pod_pdb"][sel]
                if "pod_label_pairs" in decoded:
                    a["pod_label_pairs"][rows] = decoded["pod_label_pairs"][sel]
                    a["pod_label_count"][rows] = decoded["pod_label_count"][sel]
                self.dirty.update(rows.tolist())
            return n
        return self._apply_batch_py(decoded)

    def _filter_shard(self, decoded: Dict) -> Dict:
        """Drop decoded watch rows whose pod uid hashes to another
        rank's shard (sharded resident mode)."""
        from ..parallel.elastic import shard_for

        uids = decoded["pod_uids"]
        keep = np.fromiter(
            (shard_for(u, self.shard_world) == self.shard_rank for u in uids),
            bool, count=len(uids))
        if keep.all():
            return decoded
        idx = np.nonzero(keep)[0]
        out = {}
        for k, v in decoded.items():
            if isinstance(v, np.ndarray) and v.shape[:1] == (len(uids),):
                out[k] = v[idx]
            elif isinstance(v, list) and len(v) == len(uids):
                out[k] = [v[i] for i in idx]
            else:
                out[k] = v
        return out

    def _apply_batch_py(self, decoded: Dict) -> int:
        """Pure-Python reference for apply_batch's row resolution."""
        types = decoded["types"]
        n = int(types.shape[0])
        if n == 0:
            return 0
        uids = decoded["pod_uids"]
        refs = decoded["pod_refs"]
        rows = np.empty(n, np.int64)
        upsert_idx: List[int] = []
        for i in range(n):
            t = types[i]
            uid = uids[i]
            if t == 1:  # DELETED
                row = self.uid_row.pop(uid, None)
                if row is not None:
                    self._clear_row(row)
                    self.free_rows.append(row)
                continue
            if t != 0:
                continue
            row = self.uid_row.get(uid)
            if row is None:
                if self.free_rows:
                    row = self.free_rows.pop()
                else:
                    if self.size >= self.cap:
                        self._grow()
                    row = self.size
                    self.size += 1
                self.uid_row[uid] = row
            rows[len(upsert_idx)] = row
            upsert_idx.append(i)
            self.pod_uids[row] = uid
            self.pod_refs[row] = tuple(refs[i])
            self.dirty.add(row)
        if upsert_idx:
            sel = np.asarray(upsert_idx, np.int64)
            dst = rows[: len(upsert_idx)]
            a = self.arrays
            a["pod_req"][dst] = decoded["pod_req"][sel]
            a["pod_flags"][dst] = decoded["pod_flags"][sel]
            a["pod_sel_hash"][dst] = decoded["pod_sel_hash"][sel]
            a["pod_sel_pairs"][dst] = decoded["pod_sel_pairs"][sel]
            a["pod_sel_count"][dst] = decoded["pod_sel_count"][sel]
            a["pod_node"][dst] = decoded["pod_node"][sel]
            a["pod_pdb"][dst] = decoded["pod_pdb"][sel]
            if "pod_label_pairs" in decoded:
                a["pod_label_pairs"][dst] = decoded["pod_label_pairs"][sel]
                a["pod_label_count"][dst] = decoded["pod_label_count"][sel]
        return n

    def drain_dirty(self) -> np.ndarray:
        """Rows touched since the last call (the device scatter set)."""
        if self.dirty_all:
            self.dirty_all = False
            self.dirty = set()
            return np.arange(self.size, dtype=np.int64)
        rows = np.array(sorted(self.dirty), np.int64)
        self.dirty = set()
        return rows

    # ------------------------------------------------------------------
    def remap_nodes(self, new_node_index: Dict[str, int]) -> int:
        """The node SET changed (scale-up materialized, node deleted):
        rewrite pod_node indices in place instead of re-priming from a
        full pod relist (74 ms at 1M pods). Bound pods whose node
        vanished become unbound and lose the BOUND flag — exactly what
        a relist would encode for a node name missing from the index.
        Returns the number of pod rows whose binding changed."""
        old = self.node_index
        remap = np.full(max(len(old), 1), -1, np.int32)
        for name, i in old.items():
            j = new_node_index.get(name)
            if j is not None:
                remap[i] = j
        pn = self.arrays["pod_node"][: self.size]
        bound_idx = np.nonzero(pn >= 0)[0]
        if bound_idx.size:
            newv = remap[pn[bound_idx]]
            changed = newv != pn[bound_idx]
            nchanged = int(changed.sum())
            if nchanged:
                rows = bound_idx[changed]
                pn[rows] = newv[changed]
                lost = rows[newv[changed] < 0]
                if lost.size:
                    self.arrays["pod_flags"][lost] &= ~int(PodFlags.BOUND)
                if rows.size > max(1024, self.size // 8):
                    self.dirty_all = True
                else:
                    self.dirty.update(int(r) for r in rows)
        else:
            nchanged = 0
        self.node_index = dict(new_node_index)
        return nchanged

    # ------------------------------------------------------------------
    def reassign_pdbs(self, pdb_selectors: List[Dict[str, str]]) -> Optional[int]:
        """The PDB SET changed: re-derive the pod->PDB column in place
        from the cached pod LABEL pair hashes (subset match, same FNV
        convention as selectors) instead of re-priming from a full pod
        relist. Returns rows changed, or None when an exact in-place
        rematch is impossible (no label data cached, or a pod/selector
        wider than the MAX_POD_LABEL_PAIRS lattice) — the caller falls
        back to the re-prime."""
        if not self.has_label_pairs:
            return None
        from .encode import _hash_pairs

        matchers = _norm_matchers(pdb_selectors)
        n = self.size
        lp = self.arrays["pod_label_pairs"][:n]
        lc = self.arrays["pod_label_count"][:n]
        live = np.array([bool(u) for u in self.pod_uids[:n]], bool)
        if live.any() and (lc[live] > MAX_POD_LABEL_PAIRS).any():
            return None  # truncated label sets: can't rematch exactly
        pod_ns = np.array([r[0] for r in self.pod_refs[:n]], object)
        new_pdb = np.full(n, -1, np.int32)
        unassigned = live.copy()
        for d, (pns, sel) in enumerate(matchers):
            # policy/v1: empty matchLabels selects every pod in the
            # PDB's namespace — the pair loop below is vacuous, so m
            # stays unassigned & same-namespace (encode/codec parity)
            pairs, cnt, _ = _hash_pairs(sel, MAX_POD_LABEL_PAIRS)
            if cnt > MAX_POD_LABEL_PAIRS:
                return None
            m = unassigned & (pod_ns == pns)
            for h in pairs[:cnt]:
                if not m.any():
                    break
                m &= (lp == h).any(axis=1)
            new_pdb[m] = d
            unassigned &= ~m
        old = self.arrays["pod_pdb"][:n]
        # tombstones keep their -1 (new_pdb is -1 there: not live)
        changed = np.nonzero(new_pdb != old)[0]
        if changed.size:
            old[changed] = new_pdb[changed]
            if changed.size > max(1024, n // 8):
                self.dirty_all = True
            else:
                self.dirty.update(int(r) for r in changed)
        self.pdb_selectors = matchers
        return int(changed.size)

    @property
    def dead_fraction(self) -> float:
        return len(self.free_rows) / max(1, self.size)

    def maybe_compact(self) -> bool:
        """Drop tombstones when they dominate (rebuild row map)."""
        if self.dead_fraction < _COMPACT_DEAD_FRACTION:
            return False
        live = [i for i in range(self.size) if self.pod_uids[i]]
        for col in _POD_COLS:
            a = self.arrays[col]
            a[: len(live)] = a[live]
            # vacated rows must be inert: flags 0 AND unbound (-1) — a
            # zero pod_node would phantom-bind them to node 0
            a[len(live): self.size] = -1 if col in ("pod_node", "pod_pdb") else 0
        self.pod_uids = [self.pod_uids[i] for i in live] + [""] * (self.cap - len(live))
        self.pod_refs = [self.pod_refs[i] for i in live] + [("", "")] * (self.cap - len(live))
        self.uid_row = {u: i for i, u in enumerate(self.pod_uids[: len(live)]) if u}
        self.size = len(live)
        self.free_rows = []
        self.dirty = set()
        self.dirty_all = True  # full rescatter after compact
        return True

    # ------------------------------------------------------------------
    def build_snapshot(self, base: ClusterSnapshot) -> ClusterSnapshot:
        """Combine cached pod columns with a freshly-encoded node/pool
        side (ClusterSnapshot for the engines; tombstone rows are inert:
        flags=0, node=-1)."""
        import dataclasses

        snap = dataclasses.replace(
            base,
            pod_req=self.arrays["pod_req"][: self.size],
            pod_flags=self.arrays["pod_flags"][: self.size],
            pod_sel_hash=self.arrays["pod_sel_hash"][: self.size],
            pod_sel_pairs=self.arrays["pod_sel_pairs"][: self.size],
            pod_sel_count=self.arrays["pod_sel_count"][: self.size],
            pod_node=self.arrays["pod_node"][: self.size],
            pod_pdb=self.arrays["pod_pdb"][: self.size],
            pod_label_pairs=self.arrays["pod_label_pairs"][: self.size],
            pod_label_count=self.arrays["pod_label_count"][: self.size],
            pod_uids=self.pod_uids[: self.size],
            pod_refs=self.pod_refs[: self.size],
        )
        from .encode import compute_node_free

        compute_node_free(snap)
        return snap

    # ------------------------------------------------------------------
    # checkpoint / resume (snapshot spill)
    def save(self, path: str):
        np.savez_compressed(
            path,
            version=2,  # v2 = + pod_label_pairs/count (v1 loads fine:
            #             load() zero-fills absent columns and disables
            #             in-place PDB reassignment)
            size=self.size,
            uids=np.array(self.pod_uids[: self.size], dtype=object),
            refs=np.array(self.pod_refs[: self.size], dtype=object),
            **{c: self.arrays[c][: self.size] for c in _POD_COLS},
        )

    def load(self, path: str, node_index: Dict[str, int],
             pdb_selectors: Optional[List[Dict[str, str]]] = None):
        data = np.load(path, allow_pickle=True)
        self.size = int(data["size"])
        self.cap = max(64, int(self.size * _GROW))
        self.arrays = {}
        from .encode import MAX_POD_LABEL_PAIRS as _LW

        _DEFAULTS = {"pod_label_pairs": ((_LW,), np.int64),
                     "pod_label_count": ((), np.int32)}
        self.has_label_pairs = "pod_label_pairs" in data
        for col in _POD_COLS:
            if col in data:
                src = data[col]
            else:  # checkpoint predates the column: zeros, reassign off
                tail, dt = _DEFAULTS[col]
                src = np.zeros((self.size,) + tail, dt)
            a = np.zeros((self.cap,) + src.shape[1:], src.dtype)
            a[: self.size] = src
            if col in ("pod_node", "pod_pdb"):
                a[self.size :] = -1
            self.arrays[col] = a
        uids = list(data["uids"])
        refs = [tuple(r) for r in data["refs"]]
        self.pod_uids = uids + [""] * (self.cap - self.size)
        self.pod_refs = refs + [("", "")] * (self.cap - self.size)
        self.uid_row = {u: i for i, u in enumerate(uids) if u}
        self.free_rows = [i for i, u in enumerate(uids) if not u]
        self.dirty = set()
        self.dirty_all = True
        self.node_index = dict(node_index)
        self.pdb_selectors = _norm_matchers(pdb_selectors)
