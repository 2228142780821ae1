from .types import (  # noqa: F401
    ActuationStatus,
    Condition,
    CrossVersionObjectReference,
    ObjectMeta,
    OptimizedAlloc,
    VariantAutoscaling,
    VariantAutoscalingSpec,
    VariantAutoscalingStatus,
)
from . import conditions  # noqa: F401
