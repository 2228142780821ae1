"""llmd.ai/v1alpha1 VariantAutoscaling API types.

Parity: reference api/v1alpha1/variantautoscaling_types.go:9-141. The JSON
wire schema (field names, defaults, validation) is identical; the in-memory
representation is Python dataclasses instead of Go structs + deepcopy-gen.

Serialization round-trips through `to_dict()` / `from_dict()` which produce
exactly the CRD's JSON shape, so manifests written for the reference CRD
load unchanged.
"""
from __future__ import annotations

import copy
import re
from dataclasses import dataclass, field
from datetime import datetime, timezone
from typing import Any, Dict, List, Optional

API_VERSION = "llmd.ai/v1alpha1"
KIND = "VariantAutoscaling"
SHORT_NAME = "va"

# Condition types (variantautoscaling_types.go:102-111)
TYPE_TARGET_RESOLVED = "TargetResolved"
TYPE_METRICS_AVAILABLE = "MetricsAvailable"
TYPE_OPTIMIZATION_READY = "OptimizationReady"
# Scale-from-zero engine condition (scalefromzero/engine.go)
TYPE_SCALE_FROM_ZERO_MODE = "ScaleFromZeroMode"

# Condition reasons (variantautoscaling_types.go:113-141)
REASON_METRICS_FOUND = "MetricsFound"
REASON_METRICS_MISSING = "MetricsMissing"
REASON_METRICS_STALE = "MetricsStale"
REASON_PROMETHEUS_ERROR = "PrometheusError"
REASON_OPTIMIZATION_SUCCEEDED = "OptimizationSucceeded"
REASON_OPTIMIZATION_FAILED = "OptimizationFailed"
REASON_METRICS_UNAVAILABLE = "MetricsUnavailable"
REASON_INVALID_CONFIGURATION = "InvalidConfiguration"
REASON_SKIPPED_PROCESSING = "SkippedProcessing"
REASON_TARGET_FOUND = "TargetFound"
REASON_TARGET_NOT_FOUND = "TargetNotFound"

_VARIANT_COST_RE = re.compile(r"^\d+(\.\d+)?$")


def utcnow() -> datetime:
    return datetime.now(timezone.utc)


def rfc3339(ts: Optional[datetime]) -> Optional[str]:
    if ts is None:
        return None
    return ts.astimezone(timezone.utc).strftime("%Y-%m-%dT%H:%M:%SZ")


def parse_rfc3339(s: Optional[str]) -> Optional[datetime]:
    if not s:
        return None
    return datetime.strptime(s, "%Y-%m-%dT%H:%M:%SZ").replace(tzinfo=timezone.utc)


@dataclass
class ObjectMeta:
    """Minimal metav1.ObjectMeta subset used by the framework."""

    name: str = ""
    namespace: str = ""
    labels: Dict[str, str] = field(default_factory=dict)
    annotations: Dict[str, str] = field(default_factory=dict)
    generation: int = 1
    resource_version: int = 0
    uid: str = ""
    creation_timestamp: Optional[datetime] = None
    deletion_timestamp: Optional[datetime] = None
    owner_references: List[Dict[str, Any]] = field(default_factory=list)

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {"name": self.name, "namespace": self.namespace}
        if self.labels:
            d["labels"] = dict(self.labels)
        if self.annotations:
            d["annotations"] = dict(self.annotations)
        d["generation"] = self.generation
        if self.uid:
            d["uid"] = self.uid
        if self.creation_timestamp:
            d["creationTimestamp"] = rfc3339(self.creation_timestamp)
        if self.deletion_timestamp:
            d["deletionTimestamp"] = rfc3339(self.deletion_timestamp)
        if self.owner_references:
            d["ownerReferences"] = copy.deepcopy(self.owner_references)
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ObjectMeta":
        return cls(
            name=d.get("name", ""),
            namespace=d.get("namespace", ""),
            labels=dict(d.get("labels") or {}),
            annotations=dict(d.get("annotations") or {}),
            generation=int(d.get("generation", 1)),
            uid=d.get("uid", ""),
            creation_timestamp=parse_rfc3339(d.get("creationTimestamp")),
            deletion_timestamp=parse_rfc3339(d.get("deletionTimestamp")),
            owner_references=list(d.get("ownerReferences") or []),
        )


@dataclass
class CrossVersionObjectReference:
    """autoscaling/v1 CrossVersionObjectReference (HPA-style scale target)."""

    kind: str = "Deployment"
    name: str = ""
    api_version: str = "apps/v1"

    def to_dict(self) -> Dict[str, Any]:
        return {"kind": self.kind, "name": self.name, "apiVersion": self.api_version}

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "CrossVersionObjectReference":
        return cls(
            kind=d.get("kind", "Deployment"),
            name=d.get("name", ""),
            api_version=d.get("apiVersion", "apps/v1"),
        )


@dataclass
class VariantAutoscalingSpec:
    """Spec: scaleTargetRef (required), modelID (required), variantCost
    (string decimal, default "10.0", pattern ^\\d+(\\.\\d+)?$)."""

    scale_target_ref: CrossVersionObjectReference = field(
        default_factory=CrossVersionObjectReference
    )
    model_id: str = ""
    variant_cost: str = "10.0"

    def validate(self) -> List[str]:
        errs = []
        if not self.scale_target_ref.name:
            errs.append("spec.scaleTargetRef.name is required")
        if not self.model_id:
            errs.append("spec.modelID is required")
        if self.variant_cost and not _VARIANT_COST_RE.match(self.variant_cost):
            errs.append("spec.variantCost must match ^\\d+(\\.\\d+)?$")
        return errs

    def cost(self) -> float:
        """Parsed variant cost, falling back to the default on bad input
        (reference engine.go parses spec.variantCost per-tick)."""
        try:
            return float(self.variant_cost)
        except (TypeError, ValueError):
            return 10.0

    def to_dict(self) -> Dict[str, Any]:
        return {
            "scaleTargetRef": self.scale_target_ref.to_dict(),
            "modelID": self.model_id,
            "variantCost": self.variant_cost,
        }

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "VariantAutoscalingSpec":
        return cls(
            scale_target_ref=CrossVersionObjectReference.from_dict(
                d.get("scaleTargetRef") or {}
            ),
            model_id=d.get("modelID", ""),
            variant_cost=d.get("variantCost", "10.0") or "10.0",
        )


@dataclass
class Condition:
    """metav1.Condition subset: type, status, reason, message,
    lastTransitionTime, observedGeneration."""

    type: str = ""
    status: str = "Unknown"  # "True" | "False" | "Unknown"
    reason: str = ""
    message: str = ""
    last_transition_time: Optional[datetime] = None
    observed_generation: int = 0

    def to_dict(self) -> Dict[str, Any]:
        return {
            "type": self.type,
            "status": self.status,
            "reason": self.reason,
            "message": self.message,
            "lastTransitionTime": rfc3339(self.last_transition_time),
            "observedGeneration": self.observed_generation,
        }

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "Condition":
        return cls(
            type=d.get("type", ""),
            status=d.get("status", "Unknown"),
            reason=d.get("reason", ""),
            message=d.get("message", ""),
            last_transition_time=parse_rfc3339(d.get("lastTransitionTime")),
            observed_generation=int(d.get("observedGeneration", 0)),
        )


@dataclass
class OptimizedAlloc:
    """Target optimized allocation (variantautoscaling_types.go:47-58)."""

    last_run_time: Optional[datetime] = None
    accelerator: str = ""
    num_replicas: int = 0

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {
            "accelerator": self.accelerator,
            "numReplicas": self.num_replicas,
        }
        if self.last_run_time is not None:
            d["lastRunTime"] = rfc3339(self.last_run_time)
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "OptimizedAlloc":
        return cls(
            last_run_time=parse_rfc3339(d.get("lastRunTime")),
            accelerator=d.get("accelerator", ""),
            num_replicas=int(d.get("numReplicas", 0)),
        )


@dataclass
class ActuationStatus:
    applied: bool = False

    def to_dict(self) -> Dict[str, Any]:
        return {"applied": self.applied}

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ActuationStatus":
        return cls(applied=bool(d.get("applied", False)))


@dataclass
class VariantAutoscalingStatus:
    desired_optimized_alloc: OptimizedAlloc = field(default_factory=OptimizedAlloc)
    actuation: ActuationStatus = field(default_factory=ActuationStatus)
    conditions: List[Condition] = field(default_factory=list)

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {"actuation": self.actuation.to_dict()}
        alloc = self.desired_optimized_alloc
        # omit the zero-valued alloc: a real stored CR has no
        # desiredOptimizedAlloc until the first valid status write (the
        # CRD requires accelerator minLength=2, so the zero struct could
        # never be stored), and the reference's #731 patch-base trick
        # exists precisely to exclude the zero struct from merge patches
        # (variantautoscaling_controller.go:244-252)
        if alloc.accelerator or alloc.num_replicas or alloc.last_run_time:
            d["desiredOptimizedAlloc"] = alloc.to_dict()
        if self.conditions:
            d["conditions"] = [c.to_dict() for c in self.conditions]
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "VariantAutoscalingStatus":
        return cls(
            desired_optimized_alloc=OptimizedAlloc.from_dict(
                d.get("desiredOptimizedAlloc") or {}
            ),
            actuation=ActuationStatus.from_dict(d.get("actuation") or {}),
            conditions=[Condition.from_dict(c) for c in d.get("conditions") or []],
        )


@dataclass
class VariantAutoscaling:
    """The VariantAutoscaling custom resource (shortName `va`)."""

    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: VariantAutoscalingSpec = field(default_factory=VariantAutoscalingSpec)
    status: VariantAutoscalingStatus = field(default_factory=VariantAutoscalingStatus)

    api_version: str = API_VERSION
    kind: str = KIND

    # --- scale target getters (variantautoscaling_types.go:143-156) ---
    def get_scale_target_api(self) -> str:
        return self.spec.scale_target_ref.api_version

    def get_scale_target_name(self) -> str:
        return self.spec.scale_target_ref.name

    def get_scale_target_kind(self) -> str:
        return self.spec.scale_target_ref.kind

    @property
    def name(self) -> str:
        return self.metadata.name

    @property
    def namespace(self) -> str:
        return self.metadata.namespace

    def full_name(self) -> str:
        """`name:namespace` display form (reference internal/utils/utils.go:318)."""
        return f"{self.metadata.name}:{self.metadata.namespace}"

    def key(self) -> str:
        return f"{self.metadata.namespace}/{self.metadata.name}"

    def deepcopy(self) -> "VariantAutoscaling":
        return copy.deepcopy(self)

    def to_dict(self) -> Dict[str, Any]:
        return {
            "apiVersion": self.api_version,
            "kind": self.kind,
            "metadata": self.metadata.to_dict(),
            "spec": self.spec.to_dict(),
            "status": self.status.to_dict(),
        }

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "VariantAutoscaling":
        return cls(
            metadata=ObjectMeta.from_dict(d.get("metadata") or {}),
            spec=VariantAutoscalingSpec.from_dict(d.get("spec") or {}),
            status=VariantAutoscalingStatus.from_dict(d.get("status") or {}),
            api_version=d.get("apiVersion", API_VERSION),
            kind=d.get("kind", KIND),
        )
