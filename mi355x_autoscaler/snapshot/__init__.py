from .encode import (  # noqa: F401
    ClusterSnapshot,
    PodFlags,
    encode_snapshot,
    fnv1a64,
    hash_label_pair,
    MIB,
)
