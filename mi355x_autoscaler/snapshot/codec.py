"""Fast-ingest wrapper around the native snapshot codec.

``encode_snapshot_fast`` = node/pool side via the Python encoder (node
counts are small) + pod columns via the C++ SAX codec
(ops/csrc/snapshot_codec.cpp) operating directly on the PodList JSON
bytes from the API server — the reference's per-pod Python wrapping
[R: autoscaler/kube.py] is the ingest bottleneck at large P and this
path removes it. Falls back to the Python encoder when the extension
is not built (CPU-only dev boxes).
"""

from __future__ import annotations

import json
from typing import Dict, Optional, Sequence


from ..models.agent_pool import AgentPool
from ..models.kube import KubeNode
from .encode import ClusterSnapshot, compute_node_free, encode_snapshot


def codec_available() -> bool:
    try:
        import torch  # noqa: F401
        from mi355x_autoscaler import _hip_ops

        return hasattr(_hip_ops, "encode_pods_json")
    except ImportError:
        return False


def encode_snapshot_fast(
    pods_body: bytes,
    nodes: Sequence[KubeNode],
    pools: Dict[str, AgentPool],
    pdbs: Optional[Sequence] = None,
) -> ClusterSnapshot:
    """PodList JSON bytes + wrapped nodes -> ClusterSnapshot."""
    if not codec_available():
        from ..models.kube import KubePod

        pods_json = json.loads(pods_body).get("items", [])
        return encode_snapshot([KubePod(p) for p in pods_json], nodes, pools, pdbs)

    import torch  # noqa: F401  (loads libc10 for the extension)
    from mi355x_autoscaler import _hip_ops

    # node/pool side via the canonical encoder (zero pods)
    base = encode_snapshot([], nodes, pools, pdbs)
    node_index = {n.name: i for i, n in enumerate(nodes)}
    pdb_selectors = [
        (getattr(p, "namespace", "default"),
         dict(getattr(p, "match_labels", {}) or {}))
        for p in (pdbs or [])
    ]
    cols = _hip_ops.encode_pods_json(bytes(pods_body), node_index, pdb_selectors)

    snap = ClusterSnapshot(
        pod_req=cols["pod_req"],
        pod_flags=cols["pod_flags"],
        pod_sel_hash=cols["pod_sel_hash"],
        pod_sel_pairs=cols["pod_sel_pairs"],
        pod_sel_count=cols["pod_sel_count"],
        pod_node=cols["pod_node"],
        pod_pdb=cols["pod_pdb"],
        pod_label_pairs=cols["pod_label_pairs"],
        pod_label_count=cols["pod_label_count"],
        node_alloc=base.node_alloc,
        node_free=base.node_alloc.copy(),
        node_labels=base.node_labels,
        node_label_count=base.node_label_count,
        node_unsched=base.node_unsched,
        node_pool=base.node_pool,
        pool_unit=base.pool_unit,
        pool_max=base.pool_max,
        pool_cur=base.pool_cur,
        pool_labels=base.pool_labels,
        pool_label_count=base.pool_label_count,
        pool_names=base.pool_names,
        pdb_allowed=base.pdb_allowed,
        pod_uids=list(cols["pod_uids"]),
        pod_refs=[tuple(r) for r in cols["pod_refs"]],
        node_names=base.node_names,
    )
    compute_node_free(snap)
    return snap
