"""AzureEngineProvider driven through RECORDED ARM exchanges.

The azure-mgmt SDKs are not installable offline, so these recorded
client doubles reproduce the exact SDK surface the provider calls
(deployments.get / export_template / begin_create_or_update, VM get +
VM/NIC/disk begin_delete) against golden acs-engine-shaped payloads,
and assert on the full request bodies — the scale-up template surgery
and the scale-down delete ordering are exercised end to end, not
mocked away. [R: autoscaler/engine_scaler.py · Scaler/EngineScaler —
reconstructed, mount empty; SURVEY §2.1 row 5 / §3.3-3.4]
"""

import copy
from types import SimpleNamespace as NS

import pytest

from mi355x_autoscaler.cloud.azure_stub import AzureEngineProvider

RG = "my-acs-rg"

# golden: acs-engine-style deployment (trimmed to the fields the
# provider touches — master VM/NIC/extension resources, two agent
# pools with Count/Offset parameters)
GOLDEN_TEMPLATE = {
    "$schema": "https://schema.management.azure.com/schemas/2015-01-01/deploymentTemplate.json",
    "parameters": {
        "agentpool1Count": {"type": "int"},
        "agentpool1Offset": {"type": "int", "defaultValue": 0},
        "gpupoolCount": {"type": "int"},
        "gpupoolOffset": {"type": "int", "defaultValue": 0},
        "masterCount": {"type": "int"},
    },
    "resources": [
        {
            "type": "Microsoft.Compute/virtualMachines",
            "name": "[concat('k8s-master-', copyIndex())]",
            "dependsOn": ["[concat('k8s-master-nic-', copyIndex())]"],
        },
        {
            "type": "Microsoft.Network/networkInterfaces",
            "name": "[concat('k8s-master-nic-', copyIndex())]",
        },
        {
            "type": "Microsoft.Compute/virtualMachines/extensions",
            "name": "[concat('k8s-master-', copyIndex(), '/cse-master')]",
        },
        {
            "type": "Microsoft.Compute/virtualMachines",
            "name": "[concat('k8s-agentpool1-', copyIndex(variables('agentpool1Offset')))]",
            "dependsOn": [
                "[concat('k8s-agentpool1-nic-', copyIndex())]",
                "[concat('k8s-master-', 0)]",
            ],
        },
        {
            "type": "Microsoft.Network/networkInterfaces",
            "name": "[concat('k8s-agentpool1-nic-', copyIndex())]",
        },
        {
            "type": "Microsoft.Compute/virtualMachines",
            "name": "[concat('k8s-gpupool-', copyIndex(variables('gpupoolOffset')))]",
        },
    ],
}

GOLDEN_PARAMETERS = {
    "agentpool1Count": {"value": 3},
    "gpupoolCount": {"value": 2},
    "masterCount": {"value": 1},
}


class _Poller:
    def __init__(self, log, tag):
        self._log = log
        self._tag = tag

    def result(self):
        self._log.append(("poller.result", self._tag))
        return NS(name=self._tag)


class RecordedDeployments:
    def __init__(self, log):
        self.log = log
        self.parameters = copy.deepcopy(GOLDEN_PARAMETERS)
        self.submitted = []

    def get(self, rg, name):
        self.log.append(("deployments.get", rg, name))
        return NS(properties=NS(parameters=copy.deepcopy(self.parameters)))

    def export_template(self, rg, name):
        self.log.append(("deployments.export_template", rg, name))
        return NS(template=copy.deepcopy(GOLDEN_TEMPLATE))

    def begin_create_or_update(self, rg, name, body):
        self.log.append(("deployments.begin_create_or_update", rg, name))
        self.submitted.append(copy.deepcopy(body))
        # the ARM record: a successful submit updates the stored params
        self.parameters.update(body["properties"]["parameters"])
        return _Poller(self.log, f"deploy:{name}")


class RecordedCompute:
    def __init__(self, log):
        self.log = log
        self.vms = {
            "k8s-agentpool1-12345678-2": NS(
                network_profile=NS(network_interfaces=[
                    NS(id=f"/subscriptions/s/resourceGroups/{RG}/providers/"
                          "Microsoft.Network/networkInterfaces/"
                          "k8s-agentpool1-nic-2")
                ]),
                storage_profile=NS(os_disk=NS(
                    name="k8s-agentpool1-12345678-2_OsDisk",
                    managed_disk=NS(id="disk-id"),
                )),
            )
        }
        self.virtual_machines = NS(
            get=self._vm_get, begin_delete=self._vm_delete)
        self.disks = NS(begin_delete=self._disk_delete)

    def _vm_get(self, rg, name):
        self.log.append(("virtual_machines.get", rg, name))
        return self.vms[name]

    def _vm_delete(self, rg, name):
        self.log.append(("virtual_machines.begin_delete", rg, name))
        return _Poller(self.log, f"vm:{name}")

    def _disk_delete(self, rg, name):
        self.log.append(("disks.begin_delete", rg, name))
        return _Poller(self.log, f"disk:{name}")


class RecordedNetwork:
    def __init__(self, log):
        self.log = log
        self.network_interfaces = NS(begin_delete=self._nic_delete)

    def _nic_delete(self, rg, name):
        self.log.append(("network_interfaces.begin_delete", rg, name))
        return _Poller(self.log, f"nic:{name}")


@pytest.fixture()
def arm():
    log = []
    deployments = RecordedDeployments(log)
    provider = AzureEngineProvider(
        RG,
        resource_client=NS(deployments=deployments),
        compute_client=RecordedCompute(log),
        network_client=RecordedNetwork(log),
    )
    return provider, deployments, log


def test_current_pool_sizes_from_deployment_record(arm):
    provider, deployments, log = arm
    sizes = provider.current_pool_sizes()
    assert sizes == {"agentpool1": 3, "gpupool": 2, "master": 1}
    assert ("deployments.get", RG, "azuredeploy") in log


def test_scale_up_full_exchange(arm):
    """fetch deployment -> export template -> strip masters -> rewrite
    Count/Offset -> create_or_update: the complete reference scale-up
    exchange, asserted on the submitted ARM body."""
    provider, deployments, log = arm
    assert provider.scale_pools({"agentpool1": 5})

    assert [e[0] for e in log] == [
        "deployments.get",            # provider._get_deployment
        "deployments.export_template",
        "deployments.get",            # current_pool_sizes
        "deployments.begin_create_or_update",
    ]
    body = deployments.submitted[-1]["properties"]
    assert deployments.submitted[-1]["properties"]["mode"] == "Incremental"

    # template surgery: every master resource gone, agents intact,
    # master dependsOn references dropped
    names = [r["name"] for r in body["template"]["resources"]]
    assert not any("master" in n for n in names)
    assert any("agentpool1" in n for n in names)
    assert any("gpupool" in n for n in names)
    agent_vm = next(r for r in body["template"]["resources"]
                    if "agentpool1-" in r["name"] and r["type"].endswith("virtualMachines"))
    assert all("master" not in d for d in agent_vm.get("dependsOn", []))

    # parameters: Count raised, Offset pinned at the CURRENT size so
    # existing agents are not redeployed; untouched pool unchanged
    params = body["parameters"]
    assert params["agentpool1Count"]["value"] == 5
    assert params["agentpool1Offset"]["value"] == 3
    assert params["gpupoolCount"]["value"] == 2

    # the ARM record advanced: a second read sees the new size
    assert provider.current_pool_sizes()["agentpool1"] == 5


def test_scale_up_resubmit_is_idempotent(arm):
    """Re-submitting the same target sizes must produce an equivalent
    deployment (same Count; Offset follows the recorded state) — the
    reference relied on same-name deployment overwrite semantics
    [R: engine_scaler.py; SURVEY §3.5 idempotency]."""
    provider, deployments, log = arm
    provider.scale_pools({"agentpool1": 5})
    provider.scale_pools({"agentpool1": 5})
    first, second = deployments.submitted
    assert second["properties"]["parameters"]["agentpool1Count"]["value"] == 5
    # offset now equals the already-recorded 5: nothing redeploys
    assert second["properties"]["parameters"]["agentpool1Offset"]["value"] == 5
    assert first["properties"]["template"] == second["properties"]["template"]


def test_scale_down_delete_ordering(arm):
    """delete VM -> NIC -> OS disk, in the reference's order, each
    waited to completion [R: engine_scaler.py delete path]."""
    provider, deployments, log = arm
    assert provider.delete_node_resources("k8s-agentpool1-12345678-2")
    ops = [e for e in log if e[0].startswith(("virtual_machines", "network_interfaces", "disks"))]
    assert ops == [
        ("virtual_machines.get", RG, "k8s-agentpool1-12345678-2"),
        ("virtual_machines.begin_delete", RG, "k8s-agentpool1-12345678-2"),
        ("network_interfaces.begin_delete", RG, "k8s-agentpool1-nic-2"),
        ("disks.begin_delete", RG, "k8s-agentpool1-12345678-2_OsDisk"),
    ]
    # every delete was polled to completion before the next started
    kinds = [e[0] for e in log]
    assert kinds.index("poller.result") < kinds.index("network_interfaces.begin_delete")


def test_dry_run_submits_nothing(arm):
    provider, deployments, log = arm
    assert provider.scale_pools({"agentpool1": 9}, dry_run=True)
    assert deployments.submitted == []
    assert provider.delete_node_resources("k8s-agentpool1-12345678-2",
                                          dry_run=True)
    assert not any(e[0].startswith("virtual_machines") for e in log)


def test_missing_clients_fail_loudly():
    provider = AzureEngineProvider(RG)
    with pytest.raises(RuntimeError, match="resource"):
        provider.current_pool_sizes()
    with pytest.raises(RuntimeError, match="compute"):
        provider.delete_node_resources("k8s-agentpool1-12345678-2")
