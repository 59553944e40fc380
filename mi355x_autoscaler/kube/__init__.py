from .objects import PodDisruptionBudget  # noqa: F401
from .client import KubeApiClient  # noqa: F401
