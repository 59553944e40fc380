from mi355x_autoscaler.cloud.template_processing import (
    prepare_scale_deployment,
    set_pool_counts,
    strip_master_resources,
)


TEMPLATE = {
    "resources": [
        {"name": "[variables('masterVMName')]", "type": "Microsoft.Compute/virtualMachines"},
        {"name": "k8s-master-nic", "type": "Microsoft.Network/networkInterfaces"},
        {
            "name": "[concat(variables('agentpool1VMNamePrefix'), copyIndex())]",
            "type": "Microsoft.Compute/virtualMachines",
            "dependsOn": ["[variables('masterVMName')]", "agentpool1-nic"],
        },
    ]
}

PARAMS = {
    "agentpool1Count": {"value": 3},
    "agentpool1Offset": {"value": 0},
    "gpupoolCount": {"value": 1},
}


def test_strip_master_resources():
    out = strip_master_resources(TEMPLATE)
    names = [r["name"] for r in out["resources"]]
    assert all("master" not in n and "Master" not in n for n in names)
    assert len(out["resources"]) == 1
    # master dependencies removed from surviving resources
    assert out["resources"][0]["dependsOn"] == ["agentpool1-nic"]
    # input untouched (pure function)
    assert len(TEMPLATE["resources"]) == 3


def test_set_pool_counts_and_offsets():
    out = set_pool_counts(PARAMS, {"agentpool1": 5}, {"agentpool1": 3})
    assert out["agentpool1Count"]["value"] == 5
    assert out["agentpool1Offset"]["value"] == 3  # existing agents not redeployed
    assert out["gpupoolCount"]["value"] == 1  # untouched
    assert PARAMS["agentpool1Count"]["value"] == 3  # pure


def test_prepare_scale_deployment():
    tmpl, params = prepare_scale_deployment(TEMPLATE, PARAMS, {"agentpool1": 4}, {"agentpool1": 3})
    assert len(tmpl["resources"]) == 1
    assert params["agentpool1Count"]["value"] == 4
