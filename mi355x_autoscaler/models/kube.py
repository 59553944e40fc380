"""Kubernetes object wrappers: pods and nodes.

Re-creation of the reference's ``KubePod`` / ``KubeNode``
[R: autoscaler/kube.py — reconstructed, unverified], built over plain
k8s-API JSON dicts instead of pykube objects (no pykube offline; the
in-repo REST client in :mod:`mi355x_autoscaler.kube.client` supplies
the dicts and performs the mutations).
"""

from __future__ import annotations

import re
from datetime import datetime, timezone
from typing import Dict, Optional

from .resources import ResourceVec

# ACS-Engine agent node naming convention the reference relies on to map
# nodes to agent pools: k8s-<poolname>-<cluster-suffix>-<index>
# [R: autoscaler/cluster.py · pool grouping — reconstructed]
_ACS_NODE_RE = re.compile(r"^k8s-(?P<pool>[A-Za-z0-9]+)-(?P<suffix>[0-9a-fA-F]+)-(?P<index>\d+)$")

# Annotations the reference checks [R: autoscaler/kube.py — reconstructed]
MIRROR_ANNOTATION = "kubernetes.io/config.mirror"
CRITICAL_ANNOTATION = "scheduler.alpha.kubernetes.io/critical-pod"
# Opt-out annotation for scale-down draining (cluster-autoscaler convention,
# kept so operators can pin pods).
SAFE_TO_EVICT_ANNOTATION = "cluster-autoscaler.kubernetes.io/safe-to-evict"

_REPLICATED_KINDS = {
    "ReplicationController",
    "ReplicaSet",
    "StatefulSet",
    "Job",
    "DeploymentConfig",
}


def parse_k8s_time(s: Optional[str]) -> Optional[float]:
    """RFC3339 -> unix seconds."""
    if not s:
        return None
    try:
        return datetime.strptime(s, "%Y-%m-%dT%H:%M:%SZ").replace(tzinfo=timezone.utc).timestamp()
    except ValueError:
        try:
            return datetime.fromisoformat(s.replace("Z", "+00:00")).timestamp()
        except ValueError:
            return None


class KubePod:
    """Wrapper over a pod JSON dict.

    [R: autoscaler/kube.py · class KubePod — reconstructed]
    """

    def __init__(self, obj: Dict):
        self.obj = obj
        meta = obj.get("metadata", {})
        self.uid: str = meta.get("uid", "")
        self.name: str = meta.get("name", "")
        self.namespace: str = meta.get("namespace", "default")
        self.annotations: Dict[str, str] = meta.get("annotations") or {}
        self.labels: Dict[str, str] = meta.get("labels") or {}
        self.owner_references = meta.get("ownerReferences") or []
        spec = obj.get("spec", {})
        self.node_name: Optional[str] = spec.get("nodeName")
        # nodeSelector is THE selector the fit simulation matches
        # [R: autoscaler/kube.py · KubePod.selectors — reconstructed]
        self.selectors: Dict[str, str] = spec.get("nodeSelector") or {}
        self.resources = self._sum_container_requests(spec)
        status = obj.get("status", {})
        self.phase: str = status.get("phase", "")
        self._conditions = status.get("conditions") or []
        self.creation_ts = parse_k8s_time(meta.get("creationTimestamp"))
        self.start_ts = parse_k8s_time(status.get("startTime"))

    @staticmethod
    def _sum_container_requests(spec: Dict) -> ResourceVec:
        """Sum every container's resources.requests; a pod itself counts
        1 against the node's pod capacity.
        [R: autoscaler/kube.py · KubePod.resources — reconstructed]"""
        total = ResourceVec(pods=1)
        for c in spec.get("containers", []) or []:
            req = (c.get("resources") or {}).get("requests") or {}
            total = total + ResourceVec.from_requests(req)
        # initContainers run serially: per-axis max against the main sum
        for c in spec.get("initContainers", []) or []:
            req = (c.get("resources") or {}).get("requests") or {}
            v = ResourceVec.from_requests(req)
            total = ResourceVec(
                cpu=max(total.cpu, v.cpu),
                memory=max(total.memory, v.memory),
                pods=total.pods,
                gpu=max(total.gpu, v.gpu),
            )
        return total

    # -- classification --------------------------------------------------
    @property
    def is_pending_unschedulable(self) -> bool:
        """PodScheduled=False with reason Unschedulable — the trigger for
        scale-up. [R: autoscaler/cluster.py · pod partitioning]"""
        if self.phase != "Pending" or self.node_name:
            return False
        for cond in self._conditions:
            if cond.get("type") == "PodScheduled":
                return cond.get("status") == "False" and cond.get("reason") == "Unschedulable"
        return False

    @property
    def is_mirrored(self) -> bool:
        return MIRROR_ANNOTATION in self.annotations

    @property
    def is_critical(self) -> bool:
        return self.annotations.get(CRITICAL_ANNOTATION, "").lower() in ("true", "'true'")

    @property
    def is_daemonset(self) -> bool:
        return any(o.get("kind") == "DaemonSet" for o in self.owner_references)

    @property
    def is_replicated(self) -> bool:
        """Owned by a controller that will reschedule it — precondition
        for safe eviction. [R: autoscaler/kube.py · is_replicated]"""
        return any(o.get("kind") in _REPLICATED_KINDS or o.get("kind") == "DaemonSet"
                   for o in self.owner_references)

    @property
    def is_drainable(self) -> bool:
        """Pod may be evicted when its node is scaled down."""
        if self.annotations.get(SAFE_TO_EVICT_ANNOTATION, "").lower() == "false":
            return False
        return (self.is_replicated or self.is_mirrored or self.is_daemonset) and not self.is_critical

    @property
    def blocks_drain(self) -> bool:
        """Pod that pins its node: not controller-owned, or critical."""
        if self.is_mirrored or self.is_daemonset:
            return False  # restarts with the node; does not pin it
        return not self.is_drainable

    def delete(self, kube, grace_seconds: int = 30) -> None:
        """[R: autoscaler/kube.py · KubePod.delete — reconstructed]"""
        kube.delete_pod(self.namespace, self.name, grace_seconds=grace_seconds)

    def __repr__(self):
        return f"KubePod({self.namespace}/{self.name})"


class KubeNode:
    """Wrapper over a node JSON dict.

    [R: autoscaler/kube.py · class KubeNode — reconstructed]
    """

    def __init__(self, obj: Dict):
        self.obj = obj
        meta = obj.get("metadata", {})
        self.uid: str = meta.get("uid", "")
        self.name: str = meta.get("name", "")
        self.labels: Dict[str, str] = meta.get("labels") or {}
        self.annotations: Dict[str, str] = meta.get("annotations") or {}
        spec = obj.get("spec", {})
        self.unschedulable: bool = bool(spec.get("unschedulable", False))
        self.provider_id: str = spec.get("providerID", "")
        status = obj.get("status", {})
        self.capacity = ResourceVec.from_node_status(status.get("capacity") or {})
        self.allocatable = ResourceVec.from_node_status(
            status.get("allocatable") or status.get("capacity") or {}
        )
        self.creation_ts = parse_k8s_time(meta.get("creationTimestamp"))
        m = _ACS_NODE_RE.match(self.name)
        self.pool_name: Optional[str] = m.group("pool") if m else None
        self.pool_index: Optional[int] = int(m.group("index")) if m else None
        # azure:///subscriptions/.../virtualMachines/<vm-name>
        # [R: autoscaler/kube.py · KubeNode.instance_id — reconstructed]
        self.instance_id: str = self.provider_id.rsplit("/", 1)[-1] if self.provider_id else self.name

    @property
    def instance_type(self) -> Optional[str]:
        return (
            self.labels.get("node.kubernetes.io/instance-type")
            or self.labels.get("beta.kubernetes.io/instance-type")
        )

    @property
    def is_ready(self) -> bool:
        for cond in (self.obj.get("status", {}).get("conditions") or []):
            if cond.get("type") == "Ready":
                return cond.get("status") == "True"
        return False

    # -- actuation verbs (the reference put these on KubeNode; here
    # they are thin delegations to the injected API client so the data
    # object stays transport-free)
    # [R: autoscaler/kube.py · cordon/uncordon/drain/delete — reconstructed]
    def cordon(self, kube) -> None:
        kube.set_unschedulable(self.name, True)
        self.unschedulable = True

    def uncordon(self, kube) -> None:
        kube.set_unschedulable(self.name, False)
        self.unschedulable = False

    def drain(self, kube, pods, grace_seconds: int = 30) -> int:
        """Evict the given resident pods (mirror/daemonset pods are
        skipped, as `kubectl drain` does). Returns pods deleted."""
        n = 0
        for p in pods:
            if p.is_mirrored or p.is_daemonset:
                continue
            kube.delete_pod(p.namespace, p.name, grace_seconds=grace_seconds)
            n += 1
        return n

    def delete(self, kube) -> None:
        kube.delete_node(self.name)

    def __repr__(self):
        return f"KubeNode({self.name})"
