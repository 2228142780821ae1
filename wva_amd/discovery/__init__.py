from .gpu_operator import (  # noqa: F401
    AcceleratorModelInfo,
    K8sGpuOperatorDiscovery,
    normalize_accelerator_name,
)
