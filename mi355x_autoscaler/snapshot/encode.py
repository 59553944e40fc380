"""Cluster snapshot -> SoA encoding on a fixed integer lattice.

The reference wraps every pod/node JSON in Python objects and does all
scheduling arithmetic in interpreted dict algebra
[R: autoscaler/kube.py, autoscaler/cluster.py — reconstructed]. This
framework instead encodes the whole cluster into structure-of-arrays
int32/int64 buffers once per cycle; the CPU oracle
(:mod:`mi355x_autoscaler.sim.oracle`) and the HIP kernels
(:mod:`mi355x_autoscaler.ops`) both consume THIS lattice, which is what
makes CPU and GPU decisions bit-identical:

  * cpu      -> millicores, int32 (requests already integral)
  * memory   -> MiB, int32; requests round UP, allocatable rounds DOWN
                (conservative: the simulator never over-packs vs exact
                byte arithmetic)
  * pods,gpu -> counts, int32

Label/selector matching is hashed: every ``key=value`` pair becomes a
64-bit FNV-1a hash; a pod's nodeSelector is a sorted, padded array of
pair hashes and a combined set hash; node labels likewise. Subset match
(every selector pair present in the node's label set) is then pure
integer work — the form kernel A evaluates on-device.

A C++ fast-path encoder (ops/csrc/snapshot_codec.cpp) implements the
same lattice for high-throughput JSON ingest; this module is the
canonical definition plus the numpy reference implementation.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np

from ..models.agent_pool import AgentPool
from ..models.kube import KubeNode, KubePod
from ..models.resources import NUM_RESOURCES, ResourceVec

MIB = 1024 * 1024

# Padded-width limits for hashed label/selector sets. With
# MAX_SELECTOR_PAIRS == MAX_LABEL_PAIRS, selector matching is EXACT for
# every selector the node-label lattice can represent: a selector wider
# than MAX_LABEL_PAIRS can never be a subset of a representable label
# set, so such pods get a poison pair (below) and match nothing —
# flagged SELECTOR_OVERFLOW. Nodes with more labels than MAX_LABEL_PAIRS
# match on the first MAX_LABEL_PAIRS sorted pairs (loud warning).
MAX_SELECTOR_PAIRS = 32
MAX_LABEL_PAIRS = 32

# Pod LABEL pair hashes (host-only column, never uploaded): lets the
# watch cache re-derive the pod->PDB mapping in place when the PDB set
# changes (PDB matchLabels ⊆ pod labels, same FNV-pair subset
# convention as selectors). Pods with more labels than this width are
# flagged via pod_label_count > MAX_POD_LABEL_PAIRS and force the old
# full-re-prime path (exactness over speed on the rare row).
MAX_POD_LABEL_PAIRS = 16

# uint64 0xFFFF…F as int64: sorts LAST in the unsigned pair order, and no
# real FNV-1a label hash equals it in practice, so a selector carrying it
# fails every subset walk (host selector_subset_match and device
# sel_subset alike).
SELECTOR_POISON = -1


_label_overflow_warned = False


def _warn_label_overflow(name: str, cnt: int) -> None:
    """Nodes with more label pairs than the lattice width match on the
    first MAX_LABEL_PAIRS (unsigned-sorted) only — consistent across
    CPU/GPU engines but potentially narrower than exact k8s semantics.
    Surface it loudly once instead of truncating silently."""
    global _label_overflow_warned
    if not _label_overflow_warned:
        import logging

        logging.getLogger(__name__).warning(
            "node %s has %d label pairs (> lattice width %d): selector "
            "matching uses the first %d sorted pairs", name, cnt,
            MAX_LABEL_PAIRS, MAX_LABEL_PAIRS)
        _label_overflow_warned = True

_FNV_OFFSET = 0xCBF29CE484222325
_FNV_PRIME = 0x100000001B3
_MASK64 = (1 << 64) - 1


def fnv1a64(data: bytes) -> int:
    """64-bit FNV-1a. Mirrored bit-for-bit by the device hasher in
    ops/hip/common.h so host- and device-computed hashes agree."""
    h = _FNV_OFFSET
    for b in data:
        h ^= b
        h = (h * _FNV_PRIME) & _MASK64
    return h


def hash_label_pair(key: str, value: str) -> int:
    return fnv1a64(f"{key}\x1f{value}".encode("utf-8"))


def _as_i64(h: int) -> int:
    """Reinterpret uint64 hash as int64 for numpy storage."""
    return h - (1 << 64) if h >= (1 << 63) else h


class PodFlags:
    """Bitfield for the per-pod flags column (uint16)."""

    PENDING = 1 << 0        # unschedulable-pending -> scale-up input
    MIRROR = 1 << 1
    DAEMONSET = 1 << 2
    CRITICAL = 1 << 3
    REPLICATED = 1 << 4
    DRAINABLE = 1 << 5      # may be evicted on scale-down
    BLOCKS_DRAIN = 1 << 6   # pins its node against scale-down
    SELECTOR_OVERFLOW = 1 << 7  # >MAX_SELECTOR_PAIRS pairs: poisoned, matches nothing
    BOUND = 1 << 8          # has spec.nodeName resolving to a known node


def quantize_request(v: ResourceVec) -> Tuple[int, int, int, int]:
    """Request vector -> lattice (memory rounds UP)."""
    return (v.cpu, -(-v.memory // MIB), v.pods, v.gpu)


def quantize_allocatable(v: ResourceVec) -> Tuple[int, int, int, int]:
    """Allocatable vector -> lattice (memory rounds DOWN)."""
    return (v.cpu, v.memory // MIB, v.pods, v.gpu)


@dataclass
class ClusterSnapshot:
    """SoA snapshot of pods, nodes and pools on the integer lattice.

    Array shapes (P pods, N nodes, K pools):
      pod_req        [P, 4] int32     quantized requests
      pod_flags      [P]    int32     PodFlags bits
      pod_sel_hash   [P]    int64     combined selector-set hash (0 = empty)
      pod_sel_pairs  [P, MAX_SELECTOR_PAIRS] int64  sorted, 0-padded
      pod_sel_count  [P]    int32
      pod_node       [P]    int32     bound node index or -1
      pod_pdb        [P]    int32     PDB index or -1
      node_alloc     [N, 4] int32     quantized allocatable
      node_free      [N, 4] int32     alloc minus bound pods (filled by
                                      compute_node_free / kernel)
      node_labels    [N, MAX_LABEL_PAIRS] int64 sorted, 0-padded
      node_label_count [N]  int32
      node_unsched   [N]    int32
      node_pool      [N]    int32     pool index or -1
      pool_unit      [K, 4] int32     capacity of ONE new node
      pool_label_*               prototype labels of a NEW pool node
      pdb_allowed    [D]    int32     disruptionsAllowed per PDB
    """

    pod_req: np.ndarray
    pod_flags: np.ndarray
    pod_sel_hash: np.ndarray
    pod_sel_pairs: np.ndarray
    pod_sel_count: np.ndarray
    pod_node: np.ndarray
    pod_pdb: np.ndarray

    node_alloc: np.ndarray
    node_free: np.ndarray
    node_labels: np.ndarray
    node_label_count: np.ndarray
    node_unsched: np.ndarray
    node_pool: np.ndarray

    pool_unit: np.ndarray
    pool_max: np.ndarray
    pool_cur: np.ndarray
    pool_labels: np.ndarray
    pool_label_count: np.ndarray
    pool_names: List[str] = field(default_factory=list)

    pdb_allowed: np.ndarray = field(default_factory=lambda: np.zeros(0, np.int32))

    # host-only pod LABEL pair hashes (PDB re-matching in the watch
    # cache; NOT part of the device upload). Zero-length when the
    # producer doesn't track them (synthetic SoA traces, shards).
    pod_label_pairs: np.ndarray = field(
        default_factory=lambda: np.zeros((0, MAX_POD_LABEL_PAIRS), np.int64))
    pod_label_count: np.ndarray = field(
        default_factory=lambda: np.zeros(0, np.int32))

    # bookkeeping for decoding decisions back to objects
    pod_uids: List[str] = field(default_factory=list)
    pod_refs: List[Tuple[str, str]] = field(default_factory=list)  # (namespace, name)
    node_names: List[str] = field(default_factory=list)

    @property
    def num_pods(self) -> int:
        return int(self.pod_req.shape[0])

    @property
    def num_nodes(self) -> int:
        return int(self.node_alloc.shape[0])

    @property
    def num_pools(self) -> int:
        return int(self.pool_unit.shape[0])

    def nbytes(self) -> int:
        total = 0
        for f in self.__dataclass_fields__:
            v = getattr(self, f)
            if isinstance(v, np.ndarray):
                total += v.nbytes
        return total


def _hash_pairs(
    mapping: Dict[str, str], width: int, poison_overflow: bool = False
) -> Tuple[np.ndarray, int, int]:
    """dict -> (sorted padded int64 array, count, combined set hash).

    poison_overflow (pod selectors): when the set is wider than the
    lattice, the last slot becomes SELECTOR_POISON so the truncated
    selector can never spuriously subset-match a label set — exact
    "matches nothing" semantics (a >width selector cannot be a subset
    of a <=width label set). Node labels do NOT poison (they truncate
    with a loud warning: labels are the superset side of the walk)."""
    hashes = sorted(hash_label_pair(k, v) for k, v in mapping.items())
    combined = 0
    for h in hashes:
        combined ^= h  # order-independent set hash
    out = np.zeros(width, np.int64)
    cnt = min(len(hashes), width)
    for i in range(cnt):
        out[i] = _as_i64(hashes[i])
    if poison_overflow and len(hashes) > width:
        out[width - 1] = SELECTOR_POISON  # max uint64: stays sorted-last
    return out, len(hashes), _as_i64(combined)


def encode_snapshot(
    pods: Sequence[KubePod],
    nodes: Sequence[KubeNode],
    pools: Dict[str, AgentPool],
    pdbs: Optional[Sequence] = None,
) -> ClusterSnapshot:
    """Encode wrapped k8s objects into the SoA lattice (numpy reference
    implementation; the C++ codec is the high-throughput path)."""
    P, N = len(pods), len(nodes)
    pool_names = sorted(pools.keys())
    pool_index = {name: i for i, name in enumerate(pool_names)}
    K = len(pool_names)

    pod_req = np.zeros((P, NUM_RESOURCES), np.int32)
    pod_flags = np.zeros(P, np.int32)
    pod_sel_hash = np.zeros(P, np.int64)
    pod_sel_pairs = np.zeros((P, MAX_SELECTOR_PAIRS), np.int64)
    pod_sel_count = np.zeros(P, np.int32)
    pod_label_pairs = np.zeros((P, MAX_POD_LABEL_PAIRS), np.int64)
    pod_label_count = np.zeros(P, np.int32)
    pod_node = np.full(P, -1, np.int32)
    pod_pdb = np.full(P, -1, np.int32)

    node_alloc = np.zeros((N, NUM_RESOURCES), np.int32)
    node_labels = np.zeros((N, MAX_LABEL_PAIRS), np.int64)
    node_label_count = np.zeros(N, np.int32)
    node_unsched = np.zeros(N, np.int32)
    node_pool = np.full(N, -1, np.int32)

    node_idx = {n.name: i for i, n in enumerate(nodes)}

    for i, n in enumerate(nodes):
        node_alloc[i] = quantize_allocatable(n.allocatable)
        node_labels[i], cnt, _ = _hash_pairs(n.labels, MAX_LABEL_PAIRS)
        node_label_count[i] = min(cnt, MAX_LABEL_PAIRS)
        if cnt > MAX_LABEL_PAIRS:
            _warn_label_overflow(n.name, cnt)
        node_unsched[i] = 1 if n.unschedulable else 0
        if n.pool_name in pool_index:
            node_pool[i] = pool_index[n.pool_name]

    # PDB matching: a PDB object carries {namespace, selector labels,
    # allowed}. A PDB selects pods in ITS OWN namespace only (v1
    # semantics) — label reuse across namespaces must not cross-guard.
    pdbs = list(pdbs or [])
    pdb_allowed = np.zeros(len(pdbs), np.int32)
    pdb_matchers: List[Tuple[str, Dict[str, str]]] = []
    for d, pdb in enumerate(pdbs):
        pdb_allowed[d] = int(getattr(pdb, "disruptions_allowed", 0))
        pdb_matchers.append((getattr(pdb, "namespace", "default"),
                             dict(getattr(pdb, "match_labels", {}) or {})))

    for i, p in enumerate(pods):
        pod_req[i] = quantize_request(p.resources)
        pairs, cnt, combined = _hash_pairs(
            p.selectors, MAX_SELECTOR_PAIRS, poison_overflow=True)
        lpairs, lcnt, _ = _hash_pairs(p.labels, MAX_POD_LABEL_PAIRS)
        pod_label_pairs[i] = lpairs
        pod_label_count[i] = lcnt  # RAW count: >width flags overflow
        pod_sel_pairs[i] = pairs
        pod_sel_count[i] = min(cnt, MAX_SELECTOR_PAIRS)
        pod_sel_hash[i] = combined
        f = 0
        if p.is_pending_unschedulable:
            f |= PodFlags.PENDING
        if p.is_mirrored:
            f |= PodFlags.MIRROR
        if p.is_daemonset:
            f |= PodFlags.DAEMONSET
        if p.is_critical:
            f |= PodFlags.CRITICAL
        if p.is_replicated:
            f |= PodFlags.REPLICATED
        if p.is_drainable:
            f |= PodFlags.DRAINABLE
        if p.blocks_drain:
            f |= PodFlags.BLOCKS_DRAIN
        if cnt > MAX_SELECTOR_PAIRS:
            f |= PodFlags.SELECTOR_OVERFLOW
        if p.node_name and p.node_name in node_idx:
            pod_node[i] = node_idx[p.node_name]
            f |= PodFlags.BOUND
        pod_flags[i] = f
        for d, (pns, sel) in enumerate(pdb_matchers):
            # policy/v1: an EMPTY selector selects every pod in the
            # PDB's namespace (all() is vacuously true) — matched
            # identically in the C++ codec and delta.reassign_pdbs
            if (pns == p.namespace
                    and all(p.labels.get(k) == v for k, v in sel.items())):
                pod_pdb[i] = d
                break

    pool_unit = np.zeros((K, NUM_RESOURCES), np.int32)
    pool_max = np.zeros(K, np.int32)
    pool_cur = np.zeros(K, np.int32)
    pool_labels = np.zeros((K, MAX_LABEL_PAIRS), np.int64)
    pool_label_count = np.zeros(K, np.int32)
    for name, k in pool_index.items():
        pool = pools[name]
        unit = pool.unit_capacity
        if unit is not None:
            pool_unit[k] = quantize_allocatable(unit)
        pool_max[k] = pool.max_size
        pool_cur[k] = pool.actual_capacity
        pool_labels[k], cnt, _ = _hash_pairs(pool.prototype_labels(), MAX_LABEL_PAIRS)
        pool_label_count[k] = min(cnt, MAX_LABEL_PAIRS)

    snap = ClusterSnapshot(
        pod_req=pod_req,
        pod_flags=pod_flags,
        pod_sel_hash=pod_sel_hash,
        pod_sel_pairs=pod_sel_pairs,
        pod_sel_count=pod_sel_count,
        pod_node=pod_node,
        pod_pdb=pod_pdb,
        node_alloc=node_alloc,
        node_free=node_alloc.copy(),
        node_labels=node_labels,
        node_label_count=node_label_count,
        node_unsched=node_unsched,
        node_pool=node_pool,
        pool_unit=pool_unit,
        pool_max=pool_max,
        pool_cur=pool_cur,
        pool_labels=pool_labels,
        pool_label_count=pool_label_count,
        pool_names=pool_names,
        pdb_allowed=pdb_allowed,
        pod_label_pairs=pod_label_pairs,
        pod_label_count=pod_label_count,
        pod_uids=[p.uid for p in pods],
        pod_refs=[(p.namespace, p.name) for p in pods],
        node_names=[n.name for n in nodes],
    )
    compute_node_free(snap)
    return snap


def compute_node_free(snap: ClusterSnapshot) -> None:
    """node_free = allocatable - sum(requests of bound pods).

    numpy reference of the device-side segmented subtraction (the GPU
    path computes this with the CSR utilization kernel).
    """
    free = snap.node_alloc.astype(np.int64).copy()
    bound = snap.pod_node >= 0
    if bound.any():
        np.subtract.at(free, snap.pod_node[bound], snap.pod_req[bound].astype(np.int64))
    snap.node_free = np.clip(free, np.iinfo(np.int32).min, np.iinfo(np.int32).max).astype(np.int32)


def selector_subset_match(
    sel_pairs: np.ndarray, sel_count: int, label_pairs: np.ndarray, label_count: int
) -> bool:
    """True iff every selector pair hash appears in the label hash set.
    Both arrays sorted ascending AS UNSIGNED 64-bit (the producer sorts
    the raw FNV hashes before the two's-complement int64 cast), so the
    two-pointer walk must compare unsigned too. Mirrors the device
    predicate (common.h sel_subset)."""
    _M = (1 << 64) - 1
    li = 0
    for si in range(sel_count):
        s = int(sel_pairs[si]) & _M
        while li < label_count and (int(label_pairs[li]) & _M) < s:
            li += 1
        if li >= label_count or (int(label_pairs[li]) & _M) != s:
            return False
    return True
