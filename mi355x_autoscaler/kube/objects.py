"""Small typed objects that are not pod/node (those live in models/kube.py)."""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict


@dataclass
class PodDisruptionBudget:
    """The slice of a v1 PodDisruptionBudget the drain scorer needs.

    The reference predates PDB-aware draining (it only checked
    owner/mirror/daemonset [R: autoscaler/cluster.py · maintain —
    reconstructed]); BASELINE.json config #4 requires it, so the drain
    kernel consumes ``disruptions_allowed`` per PDB.
    """

    name: str
    match_labels: Dict[str, str] = field(default_factory=dict)
    disruptions_allowed: int = 0
    namespace: str = "default"

    @classmethod
    def from_json(cls, obj: Dict) -> "PodDisruptionBudget":
        meta = obj.get("metadata", {})
        spec = obj.get("spec", {})
        status = obj.get("status", {})
        return cls(
            name=meta.get("name", ""),
            namespace=meta.get("namespace", "default"),
            match_labels=(spec.get("selector") or {}).get("matchLabels") or {},
            disruptions_allowed=int(status.get("disruptionsAllowed", 0)),
        )
