from . import dist  # noqa: F401
from .engine_sharded import ShardedEngine, merge_rows  # noqa: F401
