"""Cloud-provider interface: what the control loop needs from "Azure".

The reference hard-wired azure-mgmt SDK calls inside ``EngineScaler``
[R: autoscaler/engine_scaler.py, autoscaler/scaler.py — reconstructed].
Here the actuation surface is a small interface with two
implementations: the in-process fake (benchmark target, cloud/fake.py)
and the API-shape-compatible Azure stub (cloud/azure_stub.py).
"""

from __future__ import annotations

import abc
import time
from dataclasses import dataclass, field
from typing import Dict, Optional


@dataclass
class ScaleEvent:
    kind: str                      # "scale_up" | "delete_node"
    ts: float = field(default_factory=time.time)
    pool_sizes: Optional[Dict[str, int]] = None
    node_name: Optional[str] = None
    dry_run: bool = False


class CloudProvider(abc.ABC):
    """Actuation boundary. Not on the decision hot path."""

    @abc.abstractmethod
    def scale_pools(self, new_pool_sizes: Dict[str, int], dry_run: bool = False) -> bool:
        """Re-submit the cluster deployment with the given absolute pool
        sizes. Must be idempotent: submitting identical sizes is a no-op
        [R: engine_scaler.py · scale — reconstructed]."""

    @abc.abstractmethod
    def delete_node_resources(self, node_name: str, dry_run: bool = False) -> bool:
        """Delete the VM + NIC + OS disk backing a node
        [R: engine_scaler.py · delete_resources_for_node]."""

    @abc.abstractmethod
    def current_pool_sizes(self) -> Dict[str, int]:
        """Pool sizes as the cloud believes them (deployment record)."""

    def vm_exists(self, instance_id: str) -> Optional[bool]:
        """Whether the backing VM still exists; None = unknown (the
        instance-terminated node state falls back to 'busy' handling).
        [R: ec2 ancestor's instance-terminated state — reconstructed]"""
        return None
