"""Agent-pool model.

Re-creation of the reference's ``AgentPool``
[R: autoscaler/agent_pool.py — reconstructed, unverified]: a named
ACS-Engine agent pool (set of nodes sharing a VM size), its current
actual size, the predicted capacity of one NEW node of the pool
(``unit_capacity``), and the pool's max size (ACS caps pools around
100 agents).
"""

from __future__ import annotations

from typing import Dict, List, Optional

from .capacity import get_capacity
from .kube import KubeNode
from .resources import ResourceVec

# ACS-Engine caps agent pools; the reference treated ~100 as the hard
# ceiling [R: README.md known limitations — reconstructed].
DEFAULT_MAX_POOL_SIZE = 100

# "k8s-master-<suffix>-<idx>" matches the agent naming convention but is
# the control plane — never an agent pool to scale (the reference's
# template processing strips master resources for the same reason).
ALWAYS_IGNORED_POOLS = frozenset({"master"})


class AgentPool:
    def __init__(
        self,
        name: str,
        instance_type: Optional[str] = None,
        nodes: Optional[List[KubeNode]] = None,
        max_size: int = DEFAULT_MAX_POOL_SIZE,
    ):
        self.name = name
        self.nodes: List[KubeNode] = list(nodes or [])
        self._instance_type = instance_type
        self.max_size = max_size

    # -- identity --------------------------------------------------------
    @property
    def instance_type(self) -> Optional[str]:
        if self._instance_type:
            return self._instance_type
        for n in self.nodes:
            it = n.instance_type
            if it:
                return it
        return None

    # -- sizing ----------------------------------------------------------
    @property
    def actual_capacity(self) -> int:
        """Current number of nodes in the pool.
        [R: autoscaler/agent_pool.py · actual_capacity — reconstructed]"""
        return len(self.nodes)

    @property
    def unit_capacity(self) -> Optional[ResourceVec]:
        """Schedulable capacity ONE new node of this pool will provide.

        Prefers the live allocatable of an existing pool node (ground
        truth) and falls back to the static SKU table for empty pools.
        [R: autoscaler/agent_pool.py · unit_capacity — reconstructed]
        """
        for n in self.nodes:
            if not n.allocatable.is_zero():
                return n.allocatable
        return get_capacity(self.instance_type)

    @property
    def unschedulable_nodes(self) -> List[KubeNode]:
        return [n for n in self.nodes if n.unschedulable]

    @property
    def schedulable_nodes(self) -> List[KubeNode]:
        return [n for n in self.nodes if not n.unschedulable]

    # -- label matching ---------------------------------------------------
    def prototype_labels(self) -> Dict[str, str]:
        """Labels a NEW node of this pool is expected to carry — used to
        decide whether a pending pod's nodeSelector could ever match a
        node this pool would create. Union of an existing node's labels,
        or the minimal synthesized set for an empty pool."""
        for n in self.nodes:
            if n.labels:
                return dict(n.labels)
        labels = {"agentpool": self.name}
        if self.instance_type:
            labels["beta.kubernetes.io/instance-type"] = self.instance_type
            labels["node.kubernetes.io/instance-type"] = self.instance_type
        return labels

    def selector_matches(self, selectors: Dict[str, str]) -> bool:
        labels = self.prototype_labels()
        return all(labels.get(k) == v for k, v in selectors.items())

    def __repr__(self):
        return f"AgentPool({self.name}, n={self.actual_capacity}, sku={self.instance_type})"


def group_nodes_into_pools(
    nodes: List[KubeNode],
    ignore_pools: Optional[List[str]] = None,
    max_size: int = DEFAULT_MAX_POOL_SIZE,
) -> Dict[str, AgentPool]:
    """Group nodes into AgentPools by the ACS node-name convention
    ``k8s-<pool>-<suffix>-<idx>``; nodes that don't match the convention
    (e.g. masters) are skipped.
    [R: autoscaler/cluster.py · pool grouping — reconstructed]
    """
    ignore = set(ignore_pools or []) | ALWAYS_IGNORED_POOLS
    pools: Dict[str, AgentPool] = {}
    for node in nodes:
        pool = node.pool_name
        if pool is None or pool in ignore:
            continue
        if pool not in pools:
            pools[pool] = AgentPool(pool, max_size=max_size)
        pools[pool].nodes.append(node)
    for p in pools.values():
        p.nodes.sort(key=lambda n: (n.pool_index if n.pool_index is not None else 0, n.name))
    return pools
