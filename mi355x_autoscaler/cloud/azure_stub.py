"""API-shape-compatible Azure backend stub.

The reference drove azure-mgmt-resource/compute/network SDK clients
[R: autoscaler/engine_scaler.py — reconstructed]. Those SDKs are not
installable offline, so this backend preserves the EngineScaler call
shape (fetch deployment -> rewrite template -> create_or_update;
delete VM -> NIC -> disk) against injected client objects. Pass real
azure-mgmt clients in a connected environment, or leave None to get a
loud error — the benchmarks run on the fake provider, which
BASELINE.json sanctions.
"""

from __future__ import annotations

import logging
from typing import Dict

from .provider import CloudProvider
from .template_processing import prepare_scale_deployment

logger = logging.getLogger(__name__)


class AzureEngineProvider(CloudProvider):
    DEPLOYMENT_NAME = "azuredeploy"  # acs-engine's canonical deployment name

    def __init__(
        self,
        resource_group: str,
        resource_client=None,
        compute_client=None,
        network_client=None,
    ):
        self.resource_group = resource_group
        self.resource_client = resource_client
        self.compute_client = compute_client
        self.network_client = network_client

    def _require(self, client, what: str):
        if client is None:
            raise RuntimeError(
                f"Azure backend requires a {what} client; run with --backend fake offline"
            )
        return client

    def _get_deployment(self):
        rc = self._require(self.resource_client, "resource")
        return rc.deployments.get(self.resource_group, self.DEPLOYMENT_NAME)

    def current_pool_sizes(self) -> Dict[str, int]:
        dep = self._get_deployment()
        params = dep.properties.parameters or {}
        sizes = {}
        for key, val in params.items():
            if key.endswith("Count"):
                pool = key[: -len("Count")]
                sizes[pool] = int(val["value"] if isinstance(val, dict) else val)
        return sizes

    def scale_pools(self, new_pool_sizes: Dict[str, int], dry_run: bool = False) -> bool:
        rc = self._require(self.resource_client, "resource")
        dep = self._get_deployment()
        template = rc.deployments.export_template(
            self.resource_group, self.DEPLOYMENT_NAME
        ).template
        current = self.current_pool_sizes()
        tmpl, params = prepare_scale_deployment(
            template, dep.properties.parameters or {}, new_pool_sizes, current
        )
        if dry_run:
            logger.info("[dry-run] would redeploy with pool sizes %s", new_pool_sizes)
            return True
        poller = rc.deployments.begin_create_or_update(
            self.resource_group,
            self.DEPLOYMENT_NAME,
            {"properties": {"template": tmpl, "parameters": params, "mode": "Incremental"}},
        )
        # async poller: VM boot takes minutes and is NOT part of the
        # decision-latency metric; we return once accepted.
        logger.info("submitted scale deployment: %s", new_pool_sizes)
        return poller is not None

    def delete_node_resources(self, node_name: str, dry_run: bool = False) -> bool:
        """VM -> NIC -> OS disk, in order [R: engine_scaler.py]."""
        if dry_run:
            logger.info("[dry-run] would delete VM/NIC/disk for %s", node_name)
            return True
        cc = self._require(self.compute_client, "compute")
        nc = self._require(self.network_client, "network")
        vm = cc.virtual_machines.get(self.resource_group, node_name)
        cc.virtual_machines.begin_delete(self.resource_group, node_name).result()
        for nic_ref in vm.network_profile.network_interfaces:
            nic_name = nic_ref.id.rsplit("/", 1)[-1]
            nc.network_interfaces.begin_delete(self.resource_group, nic_name).result()
        disk = vm.storage_profile.os_disk
        if disk and disk.managed_disk:
            cc.disks.begin_delete(self.resource_group, disk.name).result()
        return True
