"""Kubernetes wire (JSON) serialization for the kube object model.

Maps the dataclasses in kube/objects.py onto real Kubernetes manifests
(apps/v1 Deployment, core/v1 Pod/Node/ConfigMap/Service,
coordination.k8s.io/v1 Lease, monitoring.coreos.com/v1 ServiceMonitor,
inference.networking.k8s.io InferencePool, llmd.ai/v1alpha1
VariantAutoscaling) so the REST cluster client (kube/rest.py) can speak
to an actual API server with the same object model the FakeCluster
serves in-memory. VariantAutoscaling already carries its own
to_dict/from_dict (api/types.py); everything else is defined here.

Only the field subset the controller reads/writes is mapped — unknown
fields from the server are dropped on decode and absent on encode, which
is safe because the client always PATCHes (server-side apply semantics
are not required for this controller's write set: status, scale,
metadata).
"""
from __future__ import annotations

from typing import Any, Callable, Dict, Optional, Tuple

from ..api.types import ObjectMeta, VariantAutoscaling, parse_rfc3339, rfc3339
from .objects import (
    ConfigMap,
    Namespace,
    Container,
    Deployment,
    DeploymentStatus,
    EnvVar,
    InferencePool,
    Lease,
    Node,
    Pod,
    PodStatus,
    PodTemplateSpec,
    Service,
    ServiceMonitor,
    ServicePort,
)


def _meta_to(meta: ObjectMeta) -> Dict[str, Any]:
    d = meta.to_dict()
    if meta.resource_version:
        d["resourceVersion"] = str(meta.resource_version)
    return d


def _meta_from(d: Dict[str, Any]) -> ObjectMeta:
    meta = ObjectMeta.from_dict(d or {})
    rv = (d or {}).get("resourceVersion")
    if rv is not None:
        try:
            meta.resource_version = int(rv)
        except ValueError:
            meta.resource_version = 0
    return meta


def _container_to(c: Container) -> Dict[str, Any]:
    d: Dict[str, Any] = {"name": c.name}
    if c.image:
        d["image"] = c.image
    if c.command:
        d["command"] = list(c.command)
    if c.args:
        d["args"] = list(c.args)
    if c.env:
        d["env"] = [{"name": e.name, "value": e.value} for e in c.env]
    resources: Dict[str, Any] = {}
    if c.requests:
        resources["requests"] = dict(c.requests)
    if c.limits:
        resources["limits"] = dict(c.limits)
    if resources:
        d["resources"] = resources
    return d


def _container_from(d: Dict[str, Any]) -> Container:
    res = d.get("resources") or {}
    return Container(
        name=d.get("name", "main"),
        image=d.get("image", ""),
        command=list(d.get("command") or []),
        args=list(d.get("args") or []),
        env=[
            EnvVar(name=e.get("name", ""), value=e.get("value", ""))
            for e in (d.get("env") or [])
        ],
        requests={k: str(v) for k, v in (res.get("requests") or {}).items()},
        limits={k: str(v) for k, v in (res.get("limits") or {}).items()},
    )


def _template_to(t: PodTemplateSpec) -> Dict[str, Any]:
    spec: Dict[str, Any] = {
        "containers": [_container_to(c) for c in t.containers],
    }
    if t.init_containers:
        spec["initContainers"] = [_container_to(c) for c in t.init_containers]
    if t.node_selector:
        spec["nodeSelector"] = dict(t.node_selector)
    meta: Dict[str, Any] = {}
    if t.labels:
        meta["labels"] = dict(t.labels)
    if t.annotations:
        meta["annotations"] = dict(t.annotations)
    return {"metadata": meta, "spec": spec}


def _template_from(d: Dict[str, Any]) -> PodTemplateSpec:
    meta = d.get("metadata") or {}
    spec = d.get("spec") or {}
    return PodTemplateSpec(
        labels=dict(meta.get("labels") or {}),
        annotations=dict(meta.get("annotations") or {}),
        containers=[_container_from(c) for c in (spec.get("containers") or [])],
        init_containers=[
            _container_from(c) for c in (spec.get("initContainers") or [])
        ],
        node_selector=dict(spec.get("nodeSelector") or {}),
    )


# --- Deployment ---

def deployment_to_dict(o: Deployment) -> Dict[str, Any]:
    return {
        "apiVersion": o.api_version,
        "kind": o.kind,
        "metadata": _meta_to(o.metadata),
        "spec": {
            "replicas": o.replicas,
            "selector": {"matchLabels": dict(o.selector)},
            "template": _template_to(o.template),
        },
        "status": {
            "replicas": o.status.replicas,
            "readyReplicas": o.status.ready_replicas,
            "availableReplicas": o.status.available_replicas,
            "updatedReplicas": o.status.updated_replicas,
        },
    }


def deployment_from_dict(d: Dict[str, Any]) -> Deployment:
    spec = d.get("spec") or {}
    status = d.get("status") or {}
    return Deployment(
        metadata=_meta_from(d.get("metadata") or {}),
        replicas=int(spec.get("replicas", 1)),
        selector=dict((spec.get("selector") or {}).get("matchLabels") or {}),
        template=_template_from(spec.get("template") or {}),
        status=DeploymentStatus(
            replicas=int(status.get("replicas", 0) or 0),
            ready_replicas=int(status.get("readyReplicas", 0) or 0),
            available_replicas=int(status.get("availableReplicas", 0) or 0),
            updated_replicas=int(status.get("updatedReplicas", 0) or 0),
        ),
    )


# --- Pod ---

def pod_to_dict(o: Pod) -> Dict[str, Any]:
    conditions = []
    if o.status.ready:
        conditions.append({"type": "Ready", "status": "True"})
    return {
        "apiVersion": o.api_version,
        "kind": o.kind,
        "metadata": _meta_to(o.metadata),
        "spec": {
            "containers": [_container_to(c) for c in o.containers],
            "initContainers": [_container_to(c) for c in o.init_containers],
            "nodeName": o.node_name,
        },
        "status": {
            "phase": o.status.phase,
            "podIP": o.status.pod_ip,
            "conditions": conditions,
        },
    }


def pod_from_dict(d: Dict[str, Any]) -> Pod:
    spec = d.get("spec") or {}
    status = d.get("status") or {}
    ready = any(
        c.get("type") == "Ready" and c.get("status") == "True"
        for c in (status.get("conditions") or [])
    )
    return Pod(
        metadata=_meta_from(d.get("metadata") or {}),
        containers=[_container_from(c) for c in (spec.get("containers") or [])],
        init_containers=[
            _container_from(c) for c in (spec.get("initContainers") or [])
        ],
        node_name=spec.get("nodeName", ""),
        status=PodStatus(
            phase=status.get("phase", "Running"),
            ready=ready,
            pod_ip=status.get("podIP", ""),
        ),
    )


# --- Node ---

def node_to_dict(o: Node) -> Dict[str, Any]:
    return {
        "apiVersion": o.api_version,
        "kind": o.kind,
        "metadata": _meta_to(o.metadata),
        "status": {
            "allocatable": dict(o.allocatable),
            "capacity": dict(o.capacity),
        },
    }


def node_from_dict(d: Dict[str, Any]) -> Node:
    status = d.get("status") or {}
    return Node(
        metadata=_meta_from(d.get("metadata") or {}),
        allocatable={
            k: str(v) for k, v in (status.get("allocatable") or {}).items()
        },
        capacity={k: str(v) for k, v in (status.get("capacity") or {}).items()},
    )


# --- Namespace ---

def namespace_to_dict(o: Namespace) -> Dict[str, Any]:
    return {
        "apiVersion": o.api_version,
        "kind": o.kind,
        "metadata": _meta_to(o.metadata),
    }


def namespace_from_dict(d: Dict[str, Any]) -> Namespace:
    return Namespace(metadata=_meta_from(d.get("metadata") or {}))


# --- ConfigMap ---

def configmap_to_dict(o: ConfigMap) -> Dict[str, Any]:
    return {
        "apiVersion": o.api_version,
        "kind": o.kind,
        "metadata": _meta_to(o.metadata),
        "data": dict(o.data),
    }


def configmap_from_dict(d: Dict[str, Any]) -> ConfigMap:
    return ConfigMap(
        metadata=_meta_from(d.get("metadata") or {}),
        data={k: str(v) for k, v in (d.get("data") or {}).items()},
    )


# --- Secret ---

def secret_to_dict(o) -> Dict[str, Any]:
    import base64

    return {
        "apiVersion": o.api_version,
        "kind": o.kind,
        "metadata": _meta_to(o.metadata),
        "type": o.type,
        # API-server wire format: base64-encoded values under `data`
        "data": {
            k: base64.b64encode(str(v).encode()).decode()
            for k, v in o.data.items()
        },
    }


def secret_from_dict(d: Dict[str, Any]):
    import base64

    from .objects import Secret

    data: Dict[str, str] = {}
    for k, v in (d.get("data") or {}).items():
        try:
            data[k] = base64.b64decode(str(v)).decode()
        except Exception:  # noqa: BLE001 — tolerate unencoded test data
            data[k] = str(v)
    # stringData convenience field (API server merges it into data)
    for k, v in (d.get("stringData") or {}).items():
        data[k] = str(v)
    return Secret(
        metadata=_meta_from(d.get("metadata") or {}),
        data=data,
        type=d.get("type", "Opaque"),
    )


# --- Service ---

def service_to_dict(o: Service) -> Dict[str, Any]:
    return {
        "apiVersion": o.api_version,
        "kind": o.kind,
        "metadata": _meta_to(o.metadata),
        "spec": {
            "selector": dict(o.selector),
            "ports": [
                {"name": p.name, "port": p.port, "targetPort": p.target_port}
                for p in o.ports
            ],
        },
    }


def service_from_dict(d: Dict[str, Any]) -> Service:
    spec = d.get("spec") or {}
    return Service(
        metadata=_meta_from(d.get("metadata") or {}),
        selector=dict(spec.get("selector") or {}),
        ports=[
            ServicePort(
                name=p.get("name", ""),
                port=int(p.get("port", 0) or 0),
                target_port=int(p.get("targetPort", 0) or 0),
            )
            for p in (spec.get("ports") or [])
        ],
    )


# --- InferencePool (v1 and v1alpha2 read paths; reference pool.go:40-148) ---

def inferencepool_to_dict(o: InferencePool) -> Dict[str, Any]:
    return {
        "apiVersion": o.api_version,
        "kind": o.kind,
        "metadata": _meta_to(o.metadata),
        "spec": {
            "selector": {"matchLabels": dict(o.selector)},
            "targetPortNumber": o.target_port,
            "extensionRef": {"name": o.epp_service_name},
        },
    }


def inferencepool_from_dict(d: Dict[str, Any]) -> InferencePool:
    spec = d.get("spec") or {}
    sel = spec.get("selector") or {}
    # v1 uses selector.matchLabels; v1alpha2 used a bare label map
    match = sel.get("matchLabels") if "matchLabels" in sel else sel
    ext = spec.get("extensionRef") or {}
    return InferencePool(
        metadata=_meta_from(d.get("metadata") or {}),
        selector={k: str(v) for k, v in (match or {}).items()},
        target_port=int(spec.get("targetPortNumber", 8000) or 8000),
        epp_service_name=ext.get("name", ""),
        api_version=d.get("apiVersion", "inference.networking.k8s.io/v1"),
    )


# --- ServiceMonitor ---

def servicemonitor_to_dict(o: ServiceMonitor) -> Dict[str, Any]:
    return {
        "apiVersion": o.api_version,
        "kind": o.kind,
        "metadata": _meta_to(o.metadata),
        "spec": {"selector": {"matchLabels": dict(o.selector)}},
    }


def servicemonitor_from_dict(d: Dict[str, Any]) -> ServiceMonitor:
    spec = d.get("spec") or {}
    return ServiceMonitor(
        metadata=_meta_from(d.get("metadata") or {}),
        selector=dict((spec.get("selector") or {}).get("matchLabels") or {}),
    )


# --- Lease ---
# The LeaderElector works in epoch-float seconds (lease arithmetic);
# the wire format is RFC3339 MicroTime. Encode accepts either; decode
# returns epoch floats.

def _lease_time_to(v: Any) -> Optional[str]:
    if v is None:
        return None
    if isinstance(v, (int, float)):
        from datetime import datetime, timezone

        return rfc3339(datetime.fromtimestamp(float(v), timezone.utc))
    return rfc3339(v)


def _lease_time_from(s: Optional[str]) -> Optional[float]:
    dt = parse_rfc3339(s)
    return dt.timestamp() if dt is not None else None


def lease_to_dict(o: Lease) -> Dict[str, Any]:
    return {
        "apiVersion": o.api_version,
        "kind": o.kind,
        "metadata": _meta_to(o.metadata),
        "spec": {
            "holderIdentity": o.holder_identity,
            "leaseDurationSeconds": int(o.lease_duration_seconds),
            "acquireTime": _lease_time_to(o.acquire_time),
            "renewTime": _lease_time_to(o.renew_time),
        },
    }


def lease_from_dict(d: Dict[str, Any]) -> Lease:
    spec = d.get("spec") or {}
    return Lease(
        metadata=_meta_from(d.get("metadata") or {}),
        holder_identity=spec.get("holderIdentity", ""),
        lease_duration_seconds=float(spec.get("leaseDurationSeconds", 60) or 60),
        acquire_time=_lease_time_from(spec.get("acquireTime")),
        renew_time=_lease_time_from(spec.get("renewTime")),
    )


# --- VariantAutoscaling (delegates to api/types.py) ---

def va_to_dict(o: VariantAutoscaling) -> Dict[str, Any]:
    d = o.to_dict()
    meta = d.setdefault("metadata", {})
    if o.metadata.resource_version:
        meta["resourceVersion"] = str(o.metadata.resource_version)
    return d


def va_from_dict(d: Dict[str, Any]) -> VariantAutoscaling:
    o = VariantAutoscaling.from_dict(d)
    rv = (d.get("metadata") or {}).get("resourceVersion")
    if rv is not None:
        try:
            o.metadata.resource_version = int(rv)
        except ValueError:
            o.metadata.resource_version = 0
    return o


# kind → (encode, decode, api path info)
# path info: (api prefix, group/version, plural, namespaced)
SERDE: Dict[str, Tuple[Callable, Callable, Tuple[str, str, bool]]] = {
    "Deployment": (
        deployment_to_dict, deployment_from_dict,
        ("apis/apps/v1", "deployments", True),
    ),
    "Pod": (pod_to_dict, pod_from_dict, ("api/v1", "pods", True)),
    "Node": (node_to_dict, node_from_dict, ("api/v1", "nodes", False)),
    "Namespace": (
        namespace_to_dict, namespace_from_dict, ("api/v1", "namespaces", False)
    ),
    "ConfigMap": (
        configmap_to_dict, configmap_from_dict, ("api/v1", "configmaps", True)
    ),
    "Secret": (
        secret_to_dict, secret_from_dict, ("api/v1", "secrets", True)
    ),
    "Service": (
        service_to_dict, service_from_dict, ("api/v1", "services", True)
    ),
    "InferencePool": (
        inferencepool_to_dict, inferencepool_from_dict,
        ("apis/inference.networking.k8s.io/v1", "inferencepools", True),
    ),
    "ServiceMonitor": (
        servicemonitor_to_dict, servicemonitor_from_dict,
        ("apis/monitoring.coreos.com/v1", "servicemonitors", True),
    ),
    "Lease": (
        lease_to_dict, lease_from_dict,
        ("apis/coordination.k8s.io/v1", "leases", True),
    ),
    "VariantAutoscaling": (
        va_to_dict, va_from_dict,
        ("apis/llmd.ai/v1alpha1", "variantautoscalings", True),
    ),
}


def encode(obj: Any) -> Dict[str, Any]:
    kind = getattr(obj, "kind", type(obj).__name__)
    return SERDE[kind][0](obj)


def decode(kind: str, d: Dict[str, Any]) -> Any:
    # harden against malformed wire payloads (a proxy error body, a
    # truncated chunk): the per-kind decoders expect mapping sections,
    # and a non-dict there must degrade to "absent", not AttributeError
    # deep inside a watch pump
    if not isinstance(d, dict):
        d = {}
    else:
        cleaned = None
        for section in ("metadata", "spec", "status"):
            if section in d and not isinstance(d[section], dict):
                if cleaned is None:
                    cleaned = dict(d)
                cleaned.pop(section)
        if cleaned is not None:
            d = cleaned
    return SERDE[kind][1](d)


def resource_path(kind: str, namespace: Optional[str], name: Optional[str] = None) -> str:
    prefix, plural, namespaced = SERDE[kind][2]
    parts = ["", prefix]
    if namespaced and namespace:
        parts += ["namespaces", namespace]
    parts.append(plural)
    if name:
        parts.append(name)
    return "/".join(parts)
