from .resources import ResourceVec, NUM_RESOURCES, AXES  # noqa: F401
from .kube import KubePod, KubeNode  # noqa: F401
from .capacity import get_capacity, CAPACITY  # noqa: F401
from .agent_pool import AgentPool, group_nodes_into_pools  # noqa: F401
