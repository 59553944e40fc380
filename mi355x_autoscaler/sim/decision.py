"""Decision data structures shared by the CPU oracle, the GPU engine and
the RCCL-sharded controller."""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import numpy as np


@dataclass
class Segment:
    """A run of interchangeable pending pods: same selector set, same
    quantized request vector. The unit of placement work — FFD placement
    is exactly a sequence of uniform placements, one per segment, in
    (group_hash asc, request desc) order."""

    sel_hash: int                 # int64 view of the selector-set hash
    sel_pairs: np.ndarray         # [MAX_SELECTOR_PAIRS] int64, sorted
    sel_count: int
    req: np.ndarray               # [4] int32 quantized request
    count: int                    # number of pods in the segment
    req_key: int = 0              # pack_req_key(req, critical) — the merge key
    pod_indices: Optional[np.ndarray] = None  # global pod ids (local shard)


@dataclass
class ScaleDecision:
    """The global scale-up decision: identical on every rank after the
    RCCL all-reduce; what the actuator turns into an ARM redeploy."""

    units_needed: Dict[str, int] = field(default_factory=dict)   # pool -> +nodes
    new_pool_sizes: Dict[str, int] = field(default_factory=dict)  # pool -> absolute
    pods_placed_existing: int = 0      # fit on current nodes
    pods_placed_new: int = 0           # fit only on new nodes
    pods_unfulfillable: int = 0        # no pool can ever host them
    unfulfillable_detail: List[Tuple[int, int]] = field(default_factory=list)
    # per-pool residual resource totals [K, 4] int64 (MFMA segsum output)
    pool_residual_totals: Optional[np.ndarray] = None

    def wants_scale_up(self) -> bool:
        return any(v > 0 for v in self.units_needed.values())


@dataclass
class DrainCandidate:
    node_index: int
    node_name: str
    util_permille: int           # max-axis utilization, 0..1000
    drainable_pods: int
    idle: bool


@dataclass
class MaintainReport:
    """Scale-down scoring output for one cycle."""

    node_util_permille: np.ndarray      # [N] int32
    node_blocked: np.ndarray            # [N] uint8: a resident pod pins it
    node_pdb_ok: np.ndarray             # [N] uint8: all resident pods' PDBs allow
    node_idle: np.ndarray               # [N] uint8: no non-mirror/ds pods
    node_drainable_pods: np.ndarray     # [N] int32
    candidates: List[DrainCandidate] = field(default_factory=list)
