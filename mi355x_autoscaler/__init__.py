"""MI355X-native Kubernetes GPU-node autoscaler.

Brand-new framework with the capabilities and CLI surface of
wbuchwalter/Kubernetes-acs-engine-autoscaler, re-architected for AMD
Instinct MI355X: the scheduling simulation (pod->node fit, bin-packing,
drain scoring) runs as hand-written CDNA4 HIP kernels over a
device-resident SoA cluster snapshot, sharded across up to 8 GPUs with
RCCL-over-xGMI agreement on a single global scale decision.
"""

__version__ = "0.1.0"
