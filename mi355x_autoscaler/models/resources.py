"""Resource-vector algebra for scheduling arithmetic.

Re-creation of the reference's ``KubeResource`` dict-algebra
[R: autoscaler/kube.py · class KubeResource — reconstructed, unverified]:
parses Kubernetes quantity strings (``500m``, ``128Mi``, ``2Gi``, plain
ints), supports ``+``/``-``/scalar ``*`` and partial-order comparisons,
and tracks the four resources the autoscaler schedules on.

MI355X-native deltas vs the reference:
  * the GPU resource key is ``amd.com/gpu`` (reference used
    ``alpha.kubernetes.io/nvidia-gpu``); the legacy NVIDIA keys are
    accepted on input and folded into the same "gpu" axis so traces
    written for the reference still parse.
  * all arithmetic is exact integers (milli-CPU, bytes, counts) — the
    device snapshot quantizes to the fixed int32 lattice defined in
    :mod:`mi355x_autoscaler.snapshot` and the CPU oracle consumes the
    same lattice, so CPU and GPU decisions are bit-identical.
"""

from __future__ import annotations

from typing import Iterable, Mapping

# Canonical internal axes, in fixed order. This order is ALSO the
# resource-row order of the device SoA snapshot (snapshot/encode.py) and
# of every HIP kernel — do not reorder.
AXES = ("cpu", "memory", "pods", "gpu")
CPU, MEMORY, PODS, GPU = range(4)
NUM_RESOURCES = len(AXES)

GPU_RESOURCE_KEY = "amd.com/gpu"
# Accepted aliases folded into the "gpu" axis.
_GPU_ALIASES = (
    GPU_RESOURCE_KEY,
    "alpha.kubernetes.io/nvidia-gpu",
    "nvidia.com/gpu",
)

_BINARY_SUFFIX = {
    "Ki": 1024,
    "Mi": 1024**2,
    "Gi": 1024**3,
    "Ti": 1024**4,
    "Pi": 1024**5,
    "Ei": 1024**6,
}
_DECIMAL_SUFFIX = {
    "n": 1e-9,
    "u": 1e-6,
    "m": 1e-3,
    "k": 1e3,
    "M": 1e6,
    "G": 1e9,
    "T": 1e12,
    "P": 1e15,
    "E": 1e18,
}


def parse_quantity(value) -> float:
    """Parse a Kubernetes quantity string into a float of base units.

    ``"500m"`` -> 0.5, ``"128Mi"`` -> 134217728, ``"2"`` -> 2.0.
    [R: autoscaler/kube.py · KubeResource parsing — reconstructed]
    """
    if isinstance(value, (int, float)):
        return float(value)
    s = str(value).strip()
    if not s:
        return 0.0
    for suf, mult in _BINARY_SUFFIX.items():
        if s.endswith(suf):
            return float(s[: -len(suf)]) * mult
    # decimal suffixes are single-char; check longest-match irrelevant
    suf = s[-1]
    if suf in _DECIMAL_SUFFIX and not suf.isdigit():
        return float(s[:-1]) * _DECIMAL_SUFFIX[suf]
    return float(s)


def parse_cpu_milli(value) -> int:
    """CPU request -> integer millicores (round up: requests never shrink)."""
    f = parse_quantity(value) * 1000.0
    return int(f + 0.5) if f > 0 else 0


def parse_memory_bytes(value) -> int:
    f = parse_quantity(value)
    return int(f + 0.5) if f > 0 else 0


class ResourceVec:
    """Exact integer resource vector over the fixed AXES.

    Units: cpu = millicores, memory = bytes, pods/gpu = counts.
    """

    __slots__ = ("cpu", "memory", "pods", "gpu")

    def __init__(self, cpu: int = 0, memory: int = 0, pods: int = 0, gpu: int = 0):
        self.cpu = int(cpu)
        self.memory = int(memory)
        self.pods = int(pods)
        self.gpu = int(gpu)

    # -- construction ----------------------------------------------------
    @classmethod
    def from_requests(cls, requests: Mapping, pods: int = 0) -> "ResourceVec":
        """Build from a k8s ``resources.requests`` mapping."""
        cpu = parse_cpu_milli(requests.get("cpu", 0))
        mem = parse_memory_bytes(requests.get("memory", 0))
        gpu = 0
        for key in _GPU_ALIASES:
            if key in requests:
                gpu += int(parse_quantity(requests[key]))
        return cls(cpu=cpu, memory=mem, pods=pods, gpu=gpu)

    @classmethod
    def from_node_status(cls, status_block: Mapping) -> "ResourceVec":
        """Build from a node ``status.capacity``/``status.allocatable`` map."""
        cpu = parse_cpu_milli(status_block.get("cpu", 0))
        mem = parse_memory_bytes(status_block.get("memory", 0))
        pods = int(parse_quantity(status_block.get("pods", 0)))
        gpu = 0
        for key in _GPU_ALIASES:
            if key in status_block:
                gpu += int(parse_quantity(status_block[key]))
        return cls(cpu=cpu, memory=mem, pods=pods, gpu=gpu)

    # -- algebra ---------------------------------------------------------
    def __add__(self, other: "ResourceVec") -> "ResourceVec":
        return ResourceVec(
            self.cpu + other.cpu,
            self.memory + other.memory,
            self.pods + other.pods,
            self.gpu + other.gpu,
        )

    def __sub__(self, other: "ResourceVec") -> "ResourceVec":
        return ResourceVec(
            self.cpu - other.cpu,
            self.memory - other.memory,
            self.pods - other.pods,
            self.gpu - other.gpu,
        )

    def __mul__(self, k: int) -> "ResourceVec":
        return ResourceVec(self.cpu * k, self.memory * k, self.pods * k, self.gpu * k)

    __rmul__ = __mul__

    def fits_in(self, free: "ResourceVec") -> bool:
        """True iff every axis of self is <= the corresponding axis of free.

        This is the partial-order ``<=`` the reference's fit check uses
        [R: autoscaler/kube.py · KubeResource.__le__ — reconstructed].
        """
        return (
            self.cpu <= free.cpu
            and self.memory <= free.memory
            and self.pods <= free.pods
            and self.gpu <= free.gpu
        )

    def any_negative(self) -> bool:
        return self.cpu < 0 or self.memory < 0 or self.pods < 0 or self.gpu < 0

    def is_zero(self) -> bool:
        return self.cpu == 0 and self.memory == 0 and self.pods == 0 and self.gpu == 0

    def __eq__(self, other) -> bool:
        return (
            isinstance(other, ResourceVec)
            and self.cpu == other.cpu
            and self.memory == other.memory
            and self.pods == other.pods
            and self.gpu == other.gpu
        )

    def __hash__(self):
        return hash((self.cpu, self.memory, self.pods, self.gpu))

    def as_tuple(self):
        return (self.cpu, self.memory, self.pods, self.gpu)

    def __repr__(self):
        return (
            f"ResourceVec(cpu={self.cpu}m, mem={self.memory}B, "
            f"pods={self.pods}, gpu={self.gpu})"
        )


def sum_vecs(vecs: Iterable[ResourceVec]) -> ResourceVec:
    total = ResourceVec()
    for v in vecs:
        total = total + v
    return total
