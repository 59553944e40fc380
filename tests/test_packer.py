"""_PinnedPacker narrow-width selector staging (CPU; the GPU runs the
identical code path plus the pinned->HBM DMA).

The host lattice keeps pod_sel_pairs at [P, MAX_SELECTOR_PAIRS(=32)]
but the packer stages only the used width W = max(sel_count) and
expands into a persistent zero-padded [P, 32] tensor — these tests pin
the exactness of that round trip across width changes, overflow
(poison) rows, and shape changes.
[R: no reference equivalent — upload-path optimization of the new
framework's device snapshot]
"""

import numpy as np
import torch

from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.models.kube import KubeNode, KubePod
from mi355x_autoscaler.ops.engine_gpu import _PinnedPacker
from mi355x_autoscaler.sim.synth import make_node_json, make_pod_json
from mi355x_autoscaler.snapshot.encode import (
    MAX_SELECTOR_PAIRS,
    encode_snapshot,
)


def _snap(widths):
    nodes = [KubeNode(make_node_json("agentpool1", 0))]
    pods = []
    for i, w in enumerate(widths):
        sel = {f"k{j:02d}": f"v{j}" for j in range(w)}
        pods.append(KubePod(make_pod_json(
            f"p{i}", cpu_milli=100, pending=True,
            node_selector=sel if w else None)))
    return encode_snapshot(pods, nodes, group_nodes_into_pools(nodes))


def _pairs_equal(views, snap):
    got = views["pod_sel_pairs"]
    assert got.shape == (snap.num_pods, MAX_SELECTOR_PAIRS)
    np.testing.assert_array_equal(got.cpu().numpy(), snap.pod_sel_pairs)


def test_narrow_width_expands_exactly():
    snap = _snap([0, 1, 3, 3])
    packer = _PinnedPacker()
    views = packer.pack_views(snap, "cpu")
    _pairs_equal(views, snap)


def test_width_shrink_zero_fills_stale_columns():
    packer = _PinnedPacker()
    wide = _snap([5, 5])   # W=5
    _pairs_equal(packer.pack_views(wide, "cpu"), wide)
    narrow = _snap([2, 2])  # W=2: cols 2..4 of the persistent tensor
    _pairs_equal(packer.pack_views(narrow, "cpu"), narrow)  # must rezero
    wider = _snap([7, 7])   # grow again
    _pairs_equal(packer.pack_views(wider, "cpu"), wider)


def test_overflow_poison_row_forces_full_width():
    # >32-pair selector: count == 32, so the staged width is the full
    # lattice and the poison column (last) survives the round trip
    snap = _snap([33, 2])
    packer = _PinnedPacker()
    views = packer.pack_views(snap, "cpu")
    _pairs_equal(views, snap)
    from mi355x_autoscaler.snapshot.encode import SELECTOR_POISON

    assert int(views["pod_sel_pairs"][0, -1]) == SELECTOR_POISON


def test_pod_count_change_reallocates():
    packer = _PinnedPacker()
    a = _snap([3] * 4)
    _pairs_equal(packer.pack_views(a, "cpu"), a)
    b = _snap([3] * 9)
    _pairs_equal(packer.pack_views(b, "cpu"), b)
    c = _snap([1])
    _pairs_equal(packer.pack_views(c, "cpu"), c)


def test_all_views_match_host_lattice():
    """Every packed column (not just the narrowed one) round-trips."""
    snap = _snap([0, 2, 4])
    views = _PinnedPacker().pack_views(snap, "cpu")
    for name in ("pod_req", "pod_flags", "pod_sel_hash", "pod_sel_count",
                 "pod_node", "node_alloc", "node_labels", "pool_labels"):
        np.testing.assert_array_equal(
            views[name].cpu().numpy().reshape(getattr(snap, name).shape),
            getattr(snap, name), err_msg=name)
