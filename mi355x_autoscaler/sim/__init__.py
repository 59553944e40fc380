from .decision import DrainCandidate, ScaleDecision, Segment  # noqa: F401
from .oracle import OracleSimulator, build_segments  # noqa: F401
