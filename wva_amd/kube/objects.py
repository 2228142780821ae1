"""Lightweight Kubernetes object model.

The reference is a controller-runtime program operating on apps/v1, core/v1
and gateway-api-inference-extension objects. This framework models the
subset of those objects it actually reads/writes as plain dataclasses; the
FakeCluster (kube/fake.py) serves them in-memory and a REST client can map
them onto a real API server.

Fields mirror the Kubernetes JSON schema (camelCase on the wire).
"""
from __future__ import annotations

import copy
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from ..api.types import ObjectMeta


@dataclass
class EnvVar:
    name: str
    value: str = ""


@dataclass
class Container:
    name: str = "main"
    image: str = ""
    command: List[str] = field(default_factory=list)
    args: List[str] = field(default_factory=list)
    env: List[EnvVar] = field(default_factory=list)
    # resource requests/limits: {"amd.com/gpu": "8", "cpu": "4", ...}
    requests: Dict[str, str] = field(default_factory=dict)
    limits: Dict[str, str] = field(default_factory=dict)


@dataclass
class PodTemplateSpec:
    labels: Dict[str, str] = field(default_factory=dict)
    annotations: Dict[str, str] = field(default_factory=dict)
    containers: List[Container] = field(default_factory=list)
    init_containers: List[Container] = field(default_factory=list)
    node_selector: Dict[str, str] = field(default_factory=dict)


@dataclass
class DeploymentStatus:
    replicas: int = 0
    ready_replicas: int = 0
    available_replicas: int = 0
    updated_replicas: int = 0


@dataclass
class Deployment:
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    replicas: int = 1  # spec.replicas
    selector: Dict[str, str] = field(default_factory=dict)  # spec.selector.matchLabels
    template: PodTemplateSpec = field(default_factory=PodTemplateSpec)
    status: DeploymentStatus = field(default_factory=DeploymentStatus)

    kind: str = "Deployment"
    api_version: str = "apps/v1"

    @property
    def name(self) -> str:
        return self.metadata.name

    @property
    def namespace(self) -> str:
        return self.metadata.namespace

    def deepcopy(self) -> "Deployment":
        return copy.deepcopy(self)


@dataclass
class PodStatus:
    phase: str = "Running"
    ready: bool = True
    pod_ip: str = ""


@dataclass
class Pod:
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    containers: List[Container] = field(default_factory=list)
    init_containers: List[Container] = field(default_factory=list)
    node_name: str = ""
    status: PodStatus = field(default_factory=PodStatus)

    kind: str = "Pod"
    api_version: str = "v1"

    @property
    def name(self) -> str:
        return self.metadata.name

    @property
    def namespace(self) -> str:
        return self.metadata.namespace

    def is_ready(self) -> bool:
        return self.status.phase == "Running" and self.status.ready


@dataclass
class Node:
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    # allocatable resources, e.g. {"amd.com/gpu": "8"}
    allocatable: Dict[str, str] = field(default_factory=dict)
    capacity: Dict[str, str] = field(default_factory=dict)

    kind: str = "Node"
    api_version: str = "v1"

    @property
    def name(self) -> str:
        return self.metadata.name

    @property
    def labels(self) -> Dict[str, str]:
        return self.metadata.labels


@dataclass
class ConfigMap:
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    data: Dict[str, str] = field(default_factory=dict)

    kind: str = "ConfigMap"
    api_version: str = "v1"

    @property
    def name(self) -> str:
        return self.metadata.name

    @property
    def namespace(self) -> str:
        return self.metadata.namespace


@dataclass
class Secret:
    """core/v1 Secret subset — the EPP metrics-reader token source
    (reference pod_scraping_source.go:300-331). `data` holds DECODED
    string values; the serde layer base64-encodes on the wire like the
    API server does."""

    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    data: Dict[str, str] = field(default_factory=dict)
    type: str = "Opaque"

    kind: str = "Secret"
    api_version: str = "v1"

    @property
    def name(self) -> str:
        return self.metadata.name

    @property
    def namespace(self) -> str:
        return self.metadata.namespace


@dataclass
class ServicePort:
    name: str = ""
    port: int = 0
    target_port: int = 0


@dataclass
class Service:
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    selector: Dict[str, str] = field(default_factory=dict)
    ports: List[ServicePort] = field(default_factory=list)

    kind: str = "Service"
    api_version: str = "v1"

    @property
    def name(self) -> str:
        return self.metadata.name

    @property
    def namespace(self) -> str:
        return self.metadata.namespace


@dataclass
class InferencePool:
    """gateway-api-inference-extension InferencePool (v1 or v1alpha2).

    Parity: reference internal/utils/pool/pool.go:40-148 converts either
    version to an internal EndpointPool; we model the common fields.
    """

    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    selector: Dict[str, str] = field(default_factory=dict)
    target_port: int = 8000
    # extensionRef / endpoint picker service
    epp_service_name: str = ""

    kind: str = "InferencePool"
    api_version: str = "inference.networking.k8s.io/v1"

    @property
    def name(self) -> str:
        return self.metadata.name

    @property
    def namespace(self) -> str:
        return self.metadata.namespace


@dataclass
class EndpointPicker:
    service_name: str = ""
    namespace: str = ""
    metrics_port_number: int = 9090


@dataclass
class EndpointPool:
    """Internal representation of an InferencePool (pool.go:34)."""

    name: str = ""
    namespace: str = ""
    selector: Dict[str, str] = field(default_factory=dict)
    endpoint_picker: EndpointPicker = field(default_factory=EndpointPicker)


@dataclass
class ServiceMonitor:
    """monitoring.coreos.com/v1 ServiceMonitor marker object (watched for
    deletion only — losing it breaks Prometheus scraping of vLLM pods,
    reference variantautoscaling_controller.go:330-367)."""

    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    selector: Dict[str, str] = field(default_factory=dict)

    kind: str = "ServiceMonitor"
    api_version: str = "monitoring.coreos.com/v1"

    @property
    def name(self) -> str:
        return self.metadata.name

    @property
    def namespace(self) -> str:
        return self.metadata.namespace


@dataclass
class Namespace:
    """core/v1 Namespace — read for the exclusion annotation
    `wva.llmd.ai/exclude` (reference predicates.go:184-243)."""

    metadata: ObjectMeta = field(default_factory=ObjectMeta)

    kind: str = "Namespace"
    api_version: str = "v1"

    @property
    def name(self) -> str:
        return self.metadata.name


@dataclass
class Lease:
    """coordination.k8s.io/v1 Lease used for leader election."""

    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    holder_identity: str = ""
    lease_duration_seconds: float = 60
    acquire_time: Optional[Any] = None
    renew_time: Optional[Any] = None

    kind: str = "Lease"
    api_version: str = "coordination.k8s.io/v1"
