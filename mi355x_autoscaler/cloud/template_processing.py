"""ARM deployment-template rewriting for ACS-Engine scale-up.

Pure-function re-creation of the reference's template surgery
[R: autoscaler/template_processing.py — reconstructed, unverified]:
before re-submitting the cluster's ARM deployment with a larger agent
count, the template must be stripped so ONLY the new agent VMs deploy:

  * delete master VM / master NIC / first-boot (extension) resources
  * set each scaled pool's ``<pool>Count`` parameter to the new size
  * set each pool's ``<pool>Offset`` parameter to the current size so
    existing agents keep their indices and are not redeployed

Operates on plain dicts (the template JSON); no Azure SDK involved, so
it is unit-testable offline and shared by the fake and Azure backends.
"""

from __future__ import annotations

import copy
from typing import Dict, List, Tuple

_MASTER_MARKERS = ("master", "Master")
_FIRSTBOOT_TYPES = (
    "Microsoft.Compute/virtualMachines/extensions",
)


def _is_master_resource(resource: Dict) -> bool:
    name = str(resource.get("name", ""))
    return any(m in name for m in _MASTER_MARKERS)


def strip_master_resources(template: Dict) -> Dict:
    """Remove master-VM and first-boot resources so a redeploy cannot
    touch the control plane. Returns a new template dict."""
    out = copy.deepcopy(template)
    resources: List[Dict] = out.get("resources", [])
    kept = []
    for res in resources:
        if _is_master_resource(res):
            continue
        if res.get("type") in _FIRSTBOOT_TYPES and _is_master_resource(res):
            continue
        # drop dependencies on removed master resources
        deps = res.get("dependsOn")
        if deps:
            res["dependsOn"] = [d for d in deps if not any(m in d for m in _MASTER_MARKERS)]
        kept.append(res)
    out["resources"] = kept
    return out


def set_pool_counts(
    parameters: Dict,
    new_sizes: Dict[str, int],
    current_sizes: Dict[str, int],
) -> Dict:
    """Rewrite ``<pool>Count`` / ``<pool>Offset`` parameters.

    ARM parameter files store values as {"value": X}; accept both bare
    and wrapped forms.
    """
    out = copy.deepcopy(parameters)

    def _set(key: str, value: int):
        if key in out and isinstance(out[key], dict) and "value" in out[key]:
            out[key]["value"] = value
        else:
            out[key] = {"value": value}

    for pool, size in new_sizes.items():
        _set(f"{pool}Count", size)
        # ARM deploys agent indices [offset, count): offset = current
        # so only NEW VMs deploy; min() makes an equal-size resubmit a
        # true no-op (idempotent retry)
        _set(f"{pool}Offset", min(current_sizes.get(pool, 0), size))
    return out


def prepare_scale_deployment(
    template: Dict,
    parameters: Dict,
    new_sizes: Dict[str, int],
    current_sizes: Dict[str, int],
) -> Tuple[Dict, Dict]:
    """The full pre-submit pipeline [R: engine_scaler.py · scale]."""
    return (
        strip_master_resources(template),
        set_pool_counts(parameters, new_sizes, current_sizes),
    )
