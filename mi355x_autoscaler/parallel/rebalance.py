"""Snapshot shard rebalancing.

Pods are sharded across ranks by uid-hash; churn (bursts landing on
one namespace, drains, watch deltas) can skew shard sizes, and the
controller's per-cycle cost is max-over-ranks. This module measures
the skew with one tiny all-gather and, past a threshold, exchanges
surplus pod SoA rows with `all_to_all_single` — the bandwidth-bound
collective of the design: on MI355X it runs over RCCL/xGMI, on the
CPU test rig over gloo. All-to-all is PAIRWISE p2p, so a single call
already drives every xGMI link concurrently (each rank streams
distinct slices to up to 7 peers at ~153 GB/s each) — unlike ring
collectives, no multi-ring chunking is needed for link engagement.
Row payloads are the packed int64 pod columns (~200 MB for a 1M-pod
rebalance); overlapping the host pack/unpack with the exchange is the
remaining (unmeasured, multi-GPU-gated) tuning headroom.

The decision layer never depends on WHICH rank owns a pod (segment
counts are all-reduced), so rebalancing is pure load-balancing — the
invariance is what tests/test_rebalance.py asserts.
"""

from __future__ import annotations

from typing import List, Tuple

import numpy as np
import torch

from ..snapshot.encode import MAX_SELECTOR_PAIRS, ClusterSnapshot
from . import dist as pdist

# packed row: req(4) + flags + sel_hash + sel_pairs(MAX_SELECTOR_PAIRS)
# + sel_count + node + pdb int64 columns (uids/refs in a side channel)
_SEL0 = 6                              # first selector-pair column
_TAIL = _SEL0 + MAX_SELECTOR_PAIRS     # sel_count, node, pdb follow
_ROW64 = _TAIL + 3


def pack_rows(snap: ClusterSnapshot, idx: np.ndarray) -> np.ndarray:
    rows = np.zeros((idx.size, _ROW64), np.int64)
    rows[:, 0:4] = snap.pod_req[idx]
    rows[:, 4] = snap.pod_flags[idx]
    rows[:, 5] = snap.pod_sel_hash[idx]
    rows[:, _SEL0:_TAIL] = snap.pod_sel_pairs[idx]
    rows[:, _TAIL] = snap.pod_sel_count[idx]
    rows[:, _TAIL + 1] = snap.pod_node[idx]
    rows[:, _TAIL + 2] = snap.pod_pdb[idx]  # pdb list replicated across ranks
    return rows


def unpack_rows(rows: np.ndarray):
    return dict(
        pod_req=rows[:, 0:4].astype(np.int32),
        pod_flags=rows[:, 4].astype(np.int32),
        pod_sel_hash=rows[:, 5].copy(),
        pod_sel_pairs=rows[:, _SEL0:_TAIL].copy(),
        pod_sel_count=rows[:, _TAIL].astype(np.int32),
        pod_node=rows[:, _TAIL + 1].astype(np.int32),
        pod_pdb=rows[:, _TAIL + 2].astype(np.int32),
    )


def rebalance_shard(
    snap: ClusterSnapshot,
    skew_threshold: float = 1.5,
    device=None,
) -> Tuple[ClusterSnapshot, bool]:
    """Even out shard sizes across the world. Returns (snapshot,
    rebalanced?). No-op when world==1 or skew below threshold.

    Surplus ranks send their TAIL rows to deficit ranks (deterministic:
    every rank computes the same plan from the gathered counts)."""
    world = pdist.world_size()
    if world <= 1:
        return snap, False
    rank = pdist.rank()
    import torch.distributed as dist

    counts = torch.zeros(world, dtype=torch.int64)
    counts[rank] = snap.num_pods
    pdist.allreduce_tensor(counts, "sum")
    counts_np = counts.numpy()
    total = int(counts_np.sum())
    if total == 0:
        return snap, False
    mx, mn = counts_np.max(), max(counts_np.min(), 1)
    if mx / mn < skew_threshold and (mx - mn) < 4096:
        return snap, False

    # deterministic plan: target = balanced split; surplus ranks ship
    # tail rows to deficit ranks in rank order.
    target = np.full(world, total // world, np.int64)
    target[: total % world] += 1
    surplus = counts_np - target  # >0 sender, <0 receiver
    send_mat = np.zeros((world, world), np.int64)  # [src, dst]
    givers = [(r, int(s)) for r, s in enumerate(surplus) if s > 0]
    takers = [(r, int(-s)) for r, s in enumerate(surplus) if s < 0]
    gi = ti = 0
    while gi < len(givers) and ti < len(takers):
        g, gn = givers[gi]
        t, tn = takers[ti]
        move = min(gn, tn)
        send_mat[g, t] += move
        gn -= move
        tn -= move
        givers[gi] = (g, gn)
        takers[ti] = (t, tn)
        if gn == 0:
            gi += 1
        if tn == 0:
            ti += 1

    send_counts = send_mat[rank]          # rows I send to each dst
    recv_counts = send_mat[:, rank]       # rows I receive from each src

    # carve my tail into the outgoing slices (rank order of dst)
    n_send = int(send_counts.sum())
    keep = snap.num_pods - n_send
    out_rows = pack_rows(snap, np.arange(keep, snap.num_pods))
    send_t = torch.from_numpy(np.ascontiguousarray(out_rows)).flatten()
    recv_t = torch.zeros(int(recv_counts.sum()) * _ROW64, dtype=torch.int64)
    use_cuda = device is not None and str(device).startswith("cuda")
    if use_cuda:
        send_t = send_t.to(device)
        recv_t = recv_t.to(device)
    dist.all_to_all_single(
        recv_t, send_t,
        output_split_sizes=[int(c) * _ROW64 for c in recv_counts],
        input_split_sizes=[int(c) * _ROW64 for c in send_counts],
    )
    new_rows = recv_t.cpu().numpy().reshape(-1, _ROW64)

    # uid/ref side-channel (object lists; tiny relative to the rows)
    sent_uids = snap.pod_uids[keep:]
    sent_refs = snap.pod_refs[keep:]
    uid_blobs: List = [None] * world
    dist.all_gather_object(uid_blobs, (rank, send_counts.tolist(), sent_uids, sent_refs))
    my_uids: List[str] = []
    my_refs: List[Tuple[str, str]] = []
    for src in range(world):
        _, sc, uids, refs = uid_blobs[src]
        # rows src sent to me are the slice after those to lower ranks
        before = int(np.sum(sc[:rank]))
        mine = int(sc[rank])
        my_uids.extend(uids[before : before + mine])
        my_refs.extend(refs[before : before + mine])

    cols = unpack_rows(new_rows)
    import dataclasses

    merged = dataclasses.replace(
        snap,
        pod_req=np.concatenate([snap.pod_req[:keep], cols["pod_req"]]),
        pod_flags=np.concatenate([snap.pod_flags[:keep], cols["pod_flags"]]),
        pod_sel_hash=np.concatenate([snap.pod_sel_hash[:keep], cols["pod_sel_hash"]]),
        pod_sel_pairs=np.concatenate([snap.pod_sel_pairs[:keep], cols["pod_sel_pairs"]]),
        pod_sel_count=np.concatenate([snap.pod_sel_count[:keep], cols["pod_sel_count"]]),
        pod_node=np.concatenate([snap.pod_node[:keep], cols["pod_node"]]),
        pod_pdb=np.concatenate([snap.pod_pdb[:keep], cols["pod_pdb"]]),
        # label pairs are NOT shipped (pod_pdb already travels; the
        # delta cache is never primed from a rebalanced shard) — mark
        # the column absent instead of leaving a wrong-length array
        pod_label_pairs=snap.pod_label_pairs[:0],
        pod_label_count=snap.pod_label_count[:0],
        pod_uids=snap.pod_uids[:keep] + my_uids,
        pod_refs=snap.pod_refs[:keep] + my_refs,
    )
    from ..snapshot.encode import compute_node_free

    compute_node_free(merged)
    return merged, True
