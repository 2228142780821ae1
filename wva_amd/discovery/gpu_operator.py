"""Accelerator discovery from GPU-operator / GFD node labels.

Parity: reference internal/discovery/k8s_with_gpu_operator.go:14-228 and
interface.go:6-27. amd.com-first on MI355X (ROCm gpu-operator labels
`amd.com/gpu.product`, `amd.com/gpu.memory`; allocatable `amd.com/gpu`),
with nvidia.com / intel.com kept for heterogeneous clusters (BASELINE
config #3 MI355X-vs-MI300X).

Capacity:  per-vendor node scan on `<vendor>/gpu.product` + `.memory`
           labels, allocatable `<vendor>/gpu` counts; WVA_NODE_SELECTOR
           sharding honored.
Usage:     node → GPU-type map, summing pod GPU requests (regular
           containers summed; init containers max'ed — k8s effective
           request semantics).

MI355X xGMI note: discovery is per-node, so an 8×MI355X hive shows up as
one node with allocatable amd.com/gpu=8. max_hive_by_type() exposes the
largest per-node pool per type so the limiter can check TP=8 single-node
feasibility.
"""
from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Dict, List, Optional

from .. import constants as C
from ..kube.fake import FakeCluster
from ..kube.objects import Container, Node, Pod
from ..utils.logging import get_logger

log = get_logger("discovery")


@dataclass
class AcceleratorModelInfo:
    count: int = 0
    memory: str = ""  # raw label value, MiB (e.g. "294912" for 288 GiB)


# node name → {full product name → AcceleratorModelInfo}
NodeInventory = Dict[str, Dict[str, AcceleratorModelInfo]]


def normalize_accelerator_name(full_name: str) -> str:
    """"AMD-Instinct-MI355X-288GB" → "MI355X"; "NVIDIA-A100-PCIE-80GB" → "A100".

    Parity: reference pipeline/type_inventory.go:23-65, extended with the
    AMD "Instinct" infix used by the ROCm gpu-operator.
    """
    if "-" not in full_name:
        return full_name
    parts = full_name.split("-")
    if len(parts) < 2:
        return full_name
    vendor = parts[0].upper()
    if vendor == "NVIDIA":
        return parts[1]
    if vendor == "AMD":
        # AMD-MI300X-192G → MI300X ; AMD-Instinct-MI355X-288GB → MI355X
        if parts[1].lower() == "instinct" and len(parts) >= 3:
            return parts[2]
        return parts[1]
    if vendor == "INTEL":
        if len(parts) >= 3:
            return f"{parts[1]}-{parts[2]}"
        return parts[1]
    return parts[1]


class K8sGpuOperatorDiscovery:
    """CapacityDiscovery + UsageDiscovery over GFD node labels."""

    def __init__(self, cluster: FakeCluster, node_selector: Optional[Dict[str, str]] = None):
        self.cluster = cluster
        if node_selector is None:
            raw = os.environ.get("WVA_NODE_SELECTOR", "")
            node_selector = {}
            for part in raw.split(","):
                if "=" in part:
                    k, _, v = part.partition("=")
                    node_selector[k.strip()] = v.strip()
        self.node_selector = node_selector

    # --- CapacityDiscovery ---

    def discover(self) -> NodeInventory:
        """Node → accelerator product → (count, memory)."""
        inventory: NodeInventory = {}
        nodes: List[Node] = self.cluster.list(
            "Node", label_selector=self.node_selector or None
        )
        for node in nodes:
            per_node: Dict[str, AcceleratorModelInfo] = {}
            for vendor in C.GPU_VENDORS:
                product = node.labels.get(vendor + C.GPU_PRODUCT_LABEL_SUFFIX)
                if not product:
                    continue
                memory = node.labels.get(vendor + C.GPU_MEMORY_LABEL_SUFFIX, "")
                count_str = node.allocatable.get(vendor + C.GPU_RESOURCE_SUFFIX, "0")
                try:
                    count = int(count_str)
                except ValueError:
                    count = 0
                if count <= 0:
                    continue
                info = per_node.setdefault(product, AcceleratorModelInfo(memory=memory))
                info.count += count
            if per_node:
                inventory[node.name] = per_node
        return inventory

    # --- UsageDiscovery ---

    def discover_usage(self) -> Dict[str, int]:
        """Accelerator type (normalized) → GPUs requested by running pods."""
        # Build node → normalized type map first
        node_type: Dict[str, str] = {}
        for node_name, accels in self.discover().items():
            # A node exposes one GPU product in practice; pick the first.
            for product in accels:
                node_type[node_name] = normalize_accelerator_name(product)
                break

        used: Dict[str, int] = {}
        pods: List[Pod] = self.cluster.list("Pod")
        for pod in pods:
            if pod.status.phase not in ("Running", "Pending"):
                continue
            acc_type = node_type.get(pod.node_name)
            if acc_type is None:
                continue
            gpus = self._pod_gpu_request(pod)
            if gpus > 0:
                used[acc_type] = used.get(acc_type, 0) + gpus
        return used

    @staticmethod
    def _pod_gpu_request(pod: Pod) -> int:
        def container_gpus(c: Container) -> int:
            total = 0
            for vendor in C.GPU_VENDORS:
                v = c.requests.get(vendor + C.GPU_RESOURCE_SUFFIX, "0")
                try:
                    total += int(v)
                except ValueError:
                    pass
            return total

        regular = sum(container_gpus(c) for c in pod.containers)
        init_max = max((container_gpus(c) for c in pod.init_containers), default=0)
        return max(regular, init_max)

    # --- MI355X xGMI hive feasibility ---

    def max_hive_by_type(self) -> Dict[str, int]:
        """Largest single-node allocatable GPU pool per normalized type.

        A TP=8 MI355X variant (gpus_per_replica=8) is only schedulable if
        some node has >= 8 allocatable GPUs of that type; the xGMI fabric is
        intra-node (7 links × ~153 GB/s per GPU), so replicas never span
        nodes.
        """
        result: Dict[str, int] = {}
        for _, accels in self.discover().items():
            for product, info in accels.items():
                short = normalize_accelerator_name(product)
                result[short] = max(result.get(short, 0), info.count)
        return result
