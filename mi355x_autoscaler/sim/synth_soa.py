"""Vectorized synthetic snapshot generator — builds the SoA lattice
directly (no per-pod JSON), so million-pod traces (BASELINE config #5)
are generated in milliseconds. The node/pool side goes through the
real encode path; pod columns are synthesized with numpy and use the
SAME hashes/flags the JSON path would produce (asserted in tests)."""

from __future__ import annotations

from typing import Sequence, Tuple

import numpy as np

from ..models.agent_pool import group_nodes_into_pools
from ..models.kube import KubeNode
from ..snapshot.encode import (
    MAX_SELECTOR_PAIRS,
    ClusterSnapshot,
    PodFlags,
    encode_snapshot,
    hash_label_pair,
)
from .synth import make_node_json


def _as_i64(h: int) -> int:
    return h - (1 << 64) if h >= (1 << 63) else h


def synth_burst_snapshot(
    n_pending: int,
    n_nodes_per_pool: Sequence[int] = (40, 30, 30),
    pools: Sequence[Tuple[str, str]] = (
        ("agentpool1", "Standard_D2_v2"),
        ("bigpool", "Standard_D4_v2"),
        ("gpupool", "Standard_ND96_MI355X_v1"),
    ),
    gpu_fraction: float = 0.35,
    n_shapes: int = 16,
    bound_pods_per_node: int = 0,
    seed: int = 0,
) -> ClusterSnapshot:
    rng = np.random.default_rng(seed)

    nodes_json = []
    for (pool, sku), count in zip(pools, n_nodes_per_pool):
        gpu = 8 if ("MI355X" in sku or "NC" in sku or "ND" in sku) else 0
        cpu = 96000 if gpu else 2000
        mem = 1966080 if gpu else 7168
        for i in range(count):
            nodes_json.append(
                make_node_json(pool, i, instance_type=sku, cpu_milli=cpu,
                               mem_mib=mem, gpu=gpu, pods=250 if gpu else 110)
            )
    nodes = [KubeNode(n) for n in nodes_json]
    pools_map = group_nodes_into_pools(nodes)
    base = encode_snapshot([], nodes, pools_map)

    # shape table (mirrors sim/synth.py burst shapes)
    n_shapes = max(2, min(n_shapes, 64))
    shape_cpu = rng.choice([100, 250, 500, 1000, 2000], n_shapes).astype(np.int32)
    shape_mem = rng.choice([128, 256, 512, 1024, 4096], n_shapes).astype(np.int32)
    wants_gpu = rng.random(n_shapes) < gpu_fraction
    shape_gpu = np.where(wants_gpu, rng.choice([1, 2, 4, 8], n_shapes), 0).astype(np.int32)

    gpu_sel_hash = hash_label_pair("accelerator", "amd-mi355x")
    gpu_sel_i64 = _as_i64(gpu_sel_hash)

    P_bound = bound_pods_per_node * len(nodes)
    P = n_pending + P_bound
    pod_req = np.zeros((P, 4), np.int32)
    pod_flags = np.zeros(P, np.int32)
    pod_sel_hash = np.zeros(P, np.int64)
    pod_sel_pairs = np.zeros((P, MAX_SELECTOR_PAIRS), np.int64)
    pod_sel_count = np.zeros(P, np.int32)
    pod_node = np.full(P, -1, np.int32)
    pod_pdb = np.full(P, -1, np.int32)

    sh = rng.integers(0, n_shapes, n_pending)
    pod_req[:n_pending, 0] = shape_cpu[sh]
    pod_req[:n_pending, 1] = shape_mem[sh]
    pod_req[:n_pending, 2] = 1
    pod_req[:n_pending, 3] = shape_gpu[sh]
    is_gpu = shape_gpu[sh] > 0
    pod_sel_hash[:n_pending] = np.where(is_gpu, gpu_sel_i64, 0)
    pod_sel_pairs[:n_pending, 0] = np.where(is_gpu, gpu_sel_i64, 0)
    pod_sel_count[:n_pending] = is_gpu.astype(np.int32)
    pod_flags[:n_pending] = (
        PodFlags.PENDING | PodFlags.REPLICATED | PodFlags.DRAINABLE
    )

    if P_bound:
        b = slice(n_pending, P)
        pod_req[b, 0] = rng.choice([50, 100, 200], P_bound)
        pod_req[b, 1] = rng.choice([64, 128, 256], P_bound)
        pod_req[b, 2] = 1
        pod_node[b] = np.repeat(np.arange(len(nodes), dtype=np.int32), bound_pods_per_node)
        pod_flags[b] = PodFlags.REPLICATED | PodFlags.DRAINABLE | PodFlags.BOUND

    snap = ClusterSnapshot(
        pod_req=pod_req,
        pod_flags=pod_flags,
        pod_sel_hash=pod_sel_hash,
        pod_sel_pairs=pod_sel_pairs,
        pod_sel_count=pod_sel_count,
        pod_node=pod_node,
        pod_pdb=pod_pdb,
        node_alloc=base.node_alloc,
        node_free=base.node_free.copy(),
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
        pdb_allowed=np.zeros(0, np.int32),
        pod_uids=[],
        pod_refs=[],
        node_names=base.node_names,
    )
    from ..snapshot.encode import compute_node_free

    compute_node_free(snap)
    return snap
