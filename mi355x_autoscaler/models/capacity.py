"""Static SKU -> capacity table for predicting new-node capacity.

Re-creation of the reference's ``autoscaler/capacity.py``
[R: autoscaler/capacity.py — reconstructed, unverified]: the autoscaler
must know what a node of a pool's VM size WILL provide before the node
exists, to convert residual pending pods into "units needed".

MI355X-native delta: GPU rows use the ``amd.com/gpu`` axis, and the
table adds MI355X SKUs (8 x MI355X, 288 GB HBM3E each). Memory values
are the *allocatable* estimate (capacity minus a kube-reserved slice),
mirroring how the reference's table under-promised slightly so
predictions never exceeded reality.
"""

from __future__ import annotations

from typing import Dict, Optional

from .resources import ResourceVec

GIB = 1024**3

# Fraction of raw memory assumed reserved for system/kubelet when
# predicting a future node's allocatable.
_RESERVED_MEM_FRACTION = 0.04
_DEFAULT_PODS_PER_NODE = 110


def _mk(cores: int, mem_gib: float, gpu: int = 0, pods: int = _DEFAULT_PODS_PER_NODE) -> ResourceVec:
    mem = int(mem_gib * GIB * (1.0 - _RESERVED_MEM_FRACTION))
    return ResourceVec(cpu=cores * 1000, memory=mem, pods=pods, gpu=gpu)


# Azure VM sizes the reference's table covered (D/DS v2, F, NC families)
# plus the MI355X ND-class rows this framework targets.
CAPACITY: Dict[str, ResourceVec] = {
    # General purpose [R: autoscaler/capacity.py — reconstructed]
    "Standard_D1_v2": _mk(1, 3.5),
    "Standard_D2_v2": _mk(2, 7),
    "Standard_D3_v2": _mk(4, 14),
    "Standard_D4_v2": _mk(8, 28),
    "Standard_D5_v2": _mk(16, 56),
    "Standard_D11_v2": _mk(2, 14),
    "Standard_D12_v2": _mk(4, 28),
    "Standard_D13_v2": _mk(8, 56),
    "Standard_D14_v2": _mk(16, 112),
    "Standard_D15_v2": _mk(20, 140),
    "Standard_DS2_v2": _mk(2, 7),
    "Standard_DS3_v2": _mk(4, 14),
    "Standard_DS4_v2": _mk(8, 28),
    "Standard_DS5_v2": _mk(16, 56),
    "Standard_F2": _mk(2, 4),
    "Standard_F4": _mk(4, 8),
    "Standard_F8": _mk(8, 16),
    "Standard_F16": _mk(16, 32),
    # Legacy NVIDIA NC rows (kept so reference-era configs parse; the
    # GPU axis is unified — see models/resources.py).
    "Standard_NC6": _mk(6, 56, gpu=1),
    "Standard_NC12": _mk(12, 112, gpu=2),
    "Standard_NC24": _mk(24, 224, gpu=4),
    "Standard_NC24r": _mk(24, 224, gpu=4),
    # MI355X rows (new): 8 x MI355X per node, 288 GB HBM3E per GPU.
    "Standard_ND96_MI355X_v1": _mk(96, 2048, gpu=8, pods=250),
    "Standard_ND48_MI355X_v1": _mk(48, 1024, gpu=4, pods=200),
    "Standard_ND24_MI355X_v1": _mk(24, 512, gpu=2, pods=150),
    "Standard_ND12_MI355X_v1": _mk(12, 256, gpu=1, pods=110),
}


def get_capacity(instance_type: Optional[str]) -> Optional[ResourceVec]:
    """Capacity of one new node of this VM size, or None if unknown.

    [R: autoscaler/capacity.py · get_capacity — reconstructed]
    """
    if instance_type is None:
        return None
    return CAPACITY.get(instance_type)
