"""In-process fake cloud provider.

Replaces the reference's mocked Azure SDK clients in tests and serves
as the benchmark actuation target (BASELINE.json: "synthetic cluster
state"). Records every call; optionally materializes the new nodes into
a FakeKubeState so the next reconcile cycle observes them (closing the
scale-up loop end-to-end without Azure).
"""

from __future__ import annotations

import logging
import threading
import time
from typing import Dict, List, Optional

from .provider import CloudProvider, ScaleEvent

logger = logging.getLogger(__name__)


class FakeCloudProvider(CloudProvider):
    def __init__(
        self,
        initial_pool_sizes: Optional[Dict[str, int]] = None,
        kube_state=None,                    # FakeKubeState to materialize into
        pool_templates: Optional[Dict[str, dict]] = None,  # pool -> make_node_json kwargs
        node_boot_delay_s: float = 0.0,
    ):
        self.lock = threading.RLock()
        self.pool_sizes: Dict[str, int] = dict(initial_pool_sizes or {})
        self.kube_state = kube_state
        self.pool_templates = pool_templates or {}
        self.node_boot_delay_s = node_boot_delay_s
        self.events: List[ScaleEvent] = []
        self.fail_next: int = 0
        self.terminated_vms: set = set()

    def _maybe_fail(self):
        if self.fail_next > 0:
            self.fail_next -= 1
            raise RuntimeError("injected cloud fault")

    def scale_pools(self, new_pool_sizes: Dict[str, int], dry_run: bool = False) -> bool:
        with self.lock:
            self._maybe_fail()
            changed = {
                p: s for p, s in new_pool_sizes.items() if self.pool_sizes.get(p) != s
            }
            self.events.append(
                ScaleEvent(kind="scale_up", pool_sizes=dict(new_pool_sizes), dry_run=dry_run)
            )
            if dry_run or not changed:
                return bool(changed)
            for pool, size in changed.items():
                old = self.pool_sizes.get(pool, 0)
                self.pool_sizes[pool] = size
                if self.kube_state is not None and size > old:
                    self._materialize_nodes(pool, old, size)
            return True

    def _materialize_nodes(self, pool: str, old: int, new: int):
        from ..sim.synth import make_node_json

        tmpl = dict(self.pool_templates.get(pool, {}))
        # avoid index collisions with existing nodes of the pool
        existing = [
            n for n in self.kube_state.nodes.values()
            if n["metadata"]["labels"].get("agentpool") == pool
        ]
        base = max(
            [int(n["metadata"]["name"].rsplit("-", 1)[-1]) for n in existing] + [-1]
        )
        if self.node_boot_delay_s:
            time.sleep(self.node_boot_delay_s)
        with self.kube_state.lock:
            for i in range(new - old):
                obj = make_node_json(pool, base + 1 + i, **tmpl)
                # via add_node so node watchers see the ADDED event
                self.kube_state.add_node(obj)

    def delete_node_resources(self, node_name: str, dry_run: bool = False) -> bool:
        with self.lock:
            self._maybe_fail()
            self.events.append(ScaleEvent(kind="delete_node", node_name=node_name, dry_run=dry_run))
            if dry_run:
                return True
            self.terminated_vms.add(node_name)
            # pool bookkeeping: k8s-<pool>-<suffix>-<idx>
            parts = node_name.split("-")
            if len(parts) >= 4 and parts[0] == "k8s":
                pool = parts[1]
                if pool in self.pool_sizes and self.pool_sizes[pool] > 0:
                    self.pool_sizes[pool] -= 1
            return True

    def current_pool_sizes(self) -> Dict[str, int]:
        with self.lock:
            return dict(self.pool_sizes)

    def vm_exists(self, instance_id: str) -> bool:
        with self.lock:
            return instance_id not in self.terminated_vms
