from .loop import Cluster, ClusterOptions  # noqa: F401
