from .notification import Notifier  # noqa: F401
from .metrics import CycleMetrics, PhaseTimer  # noqa: F401
