"""Simulation-engine interface + CPU engine.

The control loop talks to a ``SimulationEngine``; the CPU engine wraps
the oracle directly, the GPU engine (ops/engine_gpu.py) runs the HIP
kernel pipeline on device, and the sharded engine
(parallel/engine_sharded.py) wraps either with the RCCL decision
agreement. All three produce bit-identical decisions on the same
snapshot — that property is CI-tested.
"""

from __future__ import annotations

from typing import Optional

from ..snapshot.encode import ClusterSnapshot
from .decision import MaintainReport, ScaleDecision
from .oracle import OracleSimulator, build_segments


class SimulationEngine:
    name = "base"

    def decide(self, snap: ClusterSnapshot, over_provision: int = 0) -> ScaleDecision:
        raise NotImplementedError

    def maintain(self, snap: ClusterSnapshot, util_threshold_permille: int = 300) -> MaintainReport:
        raise NotImplementedError


class CpuEngine(SimulationEngine):
    """Oracle-backed engine — the reference-equivalent slow path
    [R: autoscaler/cluster.py · fulfill_pending/maintain semantics —
    reconstructed], and the stand-in 'published number' per
    BASELINE.md."""

    name = "cpu"

    def decide(self, snap: ClusterSnapshot, over_provision: int = 0) -> ScaleDecision:
        sim = OracleSimulator(snap, over_provision=over_provision)
        return sim.fulfill_pending(build_segments(snap))

    def maintain(self, snap: ClusterSnapshot, util_threshold_permille: int = 300) -> MaintainReport:
        sim = OracleSimulator(snap)
        return sim.maintain_report(util_threshold_permille)


def make_engine(backend: str = "cpu", device: Optional[str] = None) -> SimulationEngine:
    """Factory: 'cpu' | 'gpu' (HIP kernels, requires ROCm device)."""
    if backend == "cpu":
        return CpuEngine()
    if backend == "gpu":
        from ..ops.engine_gpu import GpuEngine

        return GpuEngine(device=device or "cuda:0")
    raise ValueError(f"unknown backend {backend!r}")
