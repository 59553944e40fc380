"""Synthetic cluster-state generator.

Produces pod/node JSON dicts shaped like the k8s API objects the
reference's tests fabricated [R: test/test_cluster.py fixtures —
reconstructed], parameterized for the BASELINE.json benchmark configs
(100-pod plumbing, 1k GPU pods, 10k mixed burst, 500-node scale-down,
1M-pod burst replay). Deterministic given a seed.
"""

from __future__ import annotations

import random
from typing import Dict, List, Optional, Sequence, Tuple

from ..models.kube import KubeNode, KubePod

_SUFFIX = "12345678"


def make_node_json(
    pool: str,
    index: int,
    instance_type: str = "Standard_D2_v2",
    cpu_milli: int = 2000,
    mem_mib: int = 7 * 1024,
    gpu: int = 0,
    pods: int = 110,
    labels: Optional[Dict[str, str]] = None,
    unschedulable: bool = False,
    creation_ts: str = "2026-01-01T00:00:00Z",
) -> Dict:
    name = f"k8s-{pool}-{_SUFFIX}-{index}"
    all_labels = {
        "agentpool": pool,
        "kubernetes.io/hostname": name,
        "beta.kubernetes.io/instance-type": instance_type,
        "node.kubernetes.io/instance-type": instance_type,
        "beta.kubernetes.io/os": "linux",
    }
    if gpu > 0:
        all_labels["accelerator"] = "amd-mi355x"
    if labels:
        all_labels.update(labels)
    cap = {
        "cpu": f"{cpu_milli}m",
        "memory": f"{mem_mib}Mi",
        "pods": str(pods),
    }
    if gpu > 0:
        cap["amd.com/gpu"] = str(gpu)
    return {
        "metadata": {
            "name": name,
            "uid": f"node-{pool}-{index}",
            "labels": all_labels,
            "creationTimestamp": creation_ts,
        },
        "spec": {
            "providerID": (
                "azure:///subscriptions/sub/resourceGroups/rg/providers/"
                f"Microsoft.Compute/virtualMachines/{name}"
            ),
            **({"unschedulable": True} if unschedulable else {}),
        },
        "status": {
            "capacity": cap,
            "allocatable": cap,
            "conditions": [{"type": "Ready", "status": "True"}],
        },
    }


def make_pod_json(
    name: str,
    namespace: str = "default",
    cpu_milli: int = 100,
    mem_mib: int = 128,
    gpu: int = 0,
    node_selector: Optional[Dict[str, str]] = None,
    node_name: Optional[str] = None,
    pending: bool = False,
    owner_kind: Optional[str] = "ReplicaSet",
    annotations: Optional[Dict[str, str]] = None,
    labels: Optional[Dict[str, str]] = None,
) -> Dict:
    requests = {"cpu": f"{cpu_milli}m", "memory": f"{mem_mib}Mi"}
    if gpu > 0:
        requests["amd.com/gpu"] = str(gpu)
    status: Dict = {"phase": "Pending" if pending else "Running"}
    if pending:
        status["conditions"] = [
            {"type": "PodScheduled", "status": "False", "reason": "Unschedulable"}
        ]
    obj: Dict = {
        "metadata": {
            "name": name,
            "namespace": namespace,
            "uid": f"pod-{namespace}-{name}",
            "annotations": annotations or {},
            "labels": labels or {},
        },
        "spec": {
            "containers": [
                {"name": "main", "resources": {"requests": requests}}
            ],
        },
        "status": status,
    }
    if owner_kind:
        obj["metadata"]["ownerReferences"] = [
            {"kind": owner_kind, "name": f"{name}-owner", "uid": f"own-{name}"}
        ]
    if node_selector:
        obj["spec"]["nodeSelector"] = node_selector
    if node_name:
        obj["spec"]["nodeName"] = node_name
    return obj


# ---------------------------------------------------------------------------
# benchmark-shaped traces
# ---------------------------------------------------------------------------

def burst_trace(
    n_pending: int,
    n_nodes_per_pool: Sequence[int] = (3,),
    pools: Sequence[Tuple[str, str]] = (("agentpool1", "Standard_D2_v2"),),
    gpu_fraction: float = 0.0,
    seed: int = 0,
    bound_pods_per_node: int = 0,
) -> Tuple[List[Dict], List[Dict]]:
    """A pending-pod burst: N pools of nodes plus ``n_pending``
    unschedulable pods with randomized (seeded) request shapes.
    Returns (pod_dicts, node_dicts)."""
    rng = random.Random(seed)
    nodes: List[Dict] = []
    for (pool, sku), count in zip(pools, n_nodes_per_pool):
        gpu = 8 if "MI355X" in sku or "NC" in sku or "ND" in sku else 0
        cpu = 96000 if gpu else 2000
        mem = 1966080 if gpu else 7168
        for i in range(count):
            nodes.append(
                make_node_json(pool, i, instance_type=sku, cpu_milli=cpu,
                               mem_mib=mem, gpu=gpu, pods=250 if gpu else 110)
            )

    pods: List[Dict] = []
    # a few distinct deployment shapes (segments) rather than per-pod noise:
    # bursts in practice are replicas of a handful of specs.
    shapes = []
    for s in range(max(2, min(16, n_pending // 64 + 2))):
        wants_gpu = rng.random() < gpu_fraction
        shapes.append(
            dict(
                cpu_milli=rng.choice([100, 250, 500, 1000, 2000]),
                mem_mib=rng.choice([128, 256, 512, 1024, 4096]),
                gpu=rng.choice([1, 2, 4, 8]) if wants_gpu else 0,
                node_selector={"accelerator": "amd-mi355x"} if wants_gpu else None,
            )
        )
    for i in range(n_pending):
        sh = shapes[rng.randrange(len(shapes))]
        pods.append(
            make_pod_json(f"burst-{i}", pending=True, **sh)
        )
    if bound_pods_per_node:
        for nd in nodes:
            nn = nd["metadata"]["name"]
            for b in range(bound_pods_per_node):
                pods.append(
                    make_pod_json(
                        f"bound-{nn}-{b}", cpu_milli=rng.choice([50, 100, 200]),
                        mem_mib=rng.choice([64, 128, 256]), node_name=nn,
                    )
                )
    return pods, nodes


def scaledown_trace(
    n_nodes: int = 500,
    pods_per_node: int = 3,
    util_levels=(0, 1, 2, 5, 8),
    pdb_fraction: float = 0.2,
    seed: int = 0,
) -> Tuple[List[Dict], List[Dict], List]:
    """Scale-down scenario: many under-utilized MI355X nodes with a mix
    of drainable / pinned / PDB-guarded pods. Returns (pods, nodes, pdbs)."""
    from ..kube.objects import PodDisruptionBudget

    rng = random.Random(seed)
    nodes = [
        make_node_json("gpupool", i, instance_type="Standard_ND96_MI355X_v1",
                       cpu_milli=96000, mem_mib=1966080, gpu=8, pods=250)
        for i in range(n_nodes)
    ]
    pods: List[Dict] = []
    pdbs = [
        PodDisruptionBudget(
            name="guard", match_labels={"app": "guarded"}, disruptions_allowed=0
        )
    ]
    for i, nd in enumerate(nodes):
        nn = nd["metadata"]["name"]
        n_pods = util_levels[i % len(util_levels)]
        for b in range(min(n_pods, pods_per_node * 4)):
            guarded = rng.random() < pdb_fraction
            pinned = (not guarded) and rng.random() < 0.05
            pods.append(
                make_pod_json(
                    f"w-{i}-{b}",
                    cpu_milli=rng.choice([500, 1000, 4000]),
                    mem_mib=rng.choice([512, 2048, 8192]),
                    gpu=rng.choice([0, 1]),
                    node_name=nn,
                    owner_kind=None if pinned else "ReplicaSet",
                    labels={"app": "guarded"} if guarded else {"app": "w"},
                )
            )
    return pods, nodes, pdbs


def wrap(pods_json: List[Dict], nodes_json: List[Dict]):
    return [KubePod(p) for p in pods_json], [KubeNode(n) for n in nodes_json]
