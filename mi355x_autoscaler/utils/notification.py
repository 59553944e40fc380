"""Slack notifications on scale events.

[R: autoscaler/notification.py · class Notifier — reconstructed,
unverified]: incoming-webhook POSTs on scale-up, failed-to-scale and
scale-down events, formatting the affected pods. Additionally every
event is appended to a structured in-memory event log (SURVEY §5
observability plan).
"""

from __future__ import annotations

import json
import logging
import time
from typing import Dict, List, Optional

logger = logging.getLogger(__name__)


class Notifier:
    def __init__(self, hook_url: Optional[str] = None, dry_run: bool = False):
        self.hook_url = hook_url
        self.dry_run = dry_run
        self.events: List[Dict] = []

    def _post(self, text: str):
        if not self.hook_url or self.dry_run:
            return
        try:
            import requests

            requests.post(self.hook_url, data=json.dumps({"text": text}), timeout=5)
        except Exception:  # notification failure must never kill the loop
            logger.exception("slack notification failed")

    def _record(self, kind: str, **fields):
        evt = {"kind": kind, "ts": time.time(), **fields}
        self.events.append(evt)
        logger.info("event %s: %s", kind, fields)

    def notify_scale_up(self, units_needed: Dict[str, int], pods_pending: int):
        self._record("scale_up", units=units_needed, pods_pending=pods_pending)
        lines = ", ".join(f"{pool}: +{n}" for pool, n in units_needed.items())
        self._post(f":chart_with_upwards_trend: scaling up ({lines}) for {pods_pending} pending pods")

    def notify_failed_to_scale(self, selector_hash: int, pods: int,
                               example_pod: str = ""):
        """[R: reference formatted the pending pod list into the Slack
        message; the lattice carries hashes, so we name an example pod
        from the group instead.]"""
        self._record("failed_to_scale", selector_hash=selector_hash, pods=pods,
                     example_pod=example_pod)
        eg = f", e.g. {example_pod}" if example_pod else ""
        self._post(
            f":warning: failed to scale: {pods} pods (selector group "
            f"{selector_hash:#x}{eg}) fit no pool or pool at max"
        )

    def notify_scale_down(self, node_name: str, drained_pods: int):
        self._record("scale_down", node=node_name, drained_pods=drained_pods)
        self._post(f":chart_with_downwards_trend: drained and removed node {node_name} ({drained_pods} pods)")

    def notify_error(self, message: str):
        self._record("error", message=message)
        self._post(f":rotating_light: autoscaler error: {message}")
