from .variantautoscaling import VariantAutoscalingReconciler  # noqa: F401
from .configmap import ConfigMapReconciler  # noqa: F401
from .inferencepool import InferencePoolReconciler  # noqa: F401
